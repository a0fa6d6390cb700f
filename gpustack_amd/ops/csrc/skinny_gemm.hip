// Split-K skinny GEMM for decode-shaped projections on MI355X (gfx950).
//
// C[M, N] = X[M, K] · W[N, K]^T, bf16 in / bf16 out, f32 accumulate.
// Decode GEMMs (M = batch rows 256-1024, N = 4-28k) leave hipBLASLt ~2.6x
// off the weight-stream roofline at M<=256 because (N/BN)x(M/BM) tiles
// cannot fill 256 CUs; this kernel splits K so the grid covers the chip
// and streams W exactly once.
//
// Structure (guide §5 "minimum 2-phase" recipe, m97-style):
//  - 256x128 tile, BK=64, 8 waves; double-buffered LDS (96 KiB)
//  - global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//    with the ((row&7)<<4) XOR swizzle applied on the PRE-SWIZZLED GLOBAL
//    SOURCE (guide m173: gload_lds writes linearly, so the source
//    permutation carries the swizzle; ds_read applies the same XOR)
//  - mfma_f32_16x16x32_bf16; per-wave output 128x32 (16 fragments)
//  - SPLITK partials in f32 workspace + a reduce/convert kernel
//    (two-pass: deterministic, no atomics)
#include "common.h"

namespace {

constexpr int BM = 256;
constexpr int BN = 128;
constexpr int BK = 64;
constexpr int SG_WAVES = 8;
constexpr int SG_THREADS = SG_WAVES * WAVE_SIZE;

typedef __attribute__((ext_vector_type(8))) short s16x8;

DEVICE_INLINE f32x4 sg_mfma(u16x8 a, u16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(s16x8, a), __builtin_bit_cast(s16x8, b), c, 0, 0, 0);
}

// LDS tiles are [rows][BK] bf16 = rows x 128 bytes; XOR-swizzle byte
// offsets within each row to kill the 16-way ds_read_b128 bank conflict.
DEVICE_INLINE int sg_swz(int row, int colbyte) {
  return row * (BK * 2) + (colbyte ^ ((row & 7) << 4));
}

template <bool WRITE_PARTIAL>
__global__ __launch_bounds__(SG_THREADS) void skinny_gemm_kernel(
    void* __restrict__ out,                 // bf16 [M,N] or f32 [S,M,N]
    const unsigned short* __restrict__ x,   // [M, K]
    const unsigned short* __restrict__ w,   // [N, K]
    int M, int N, int K, int splitk) {
  const int mt = (M + BM - 1) / BM;
  const int nt = N / BN;
  int tid = blockIdx.x;
  const int split = tid % splitk;
  tid /= splitk;
  const int m_tile = tid % mt;
  const int n_tile = tid / mt;
  const int m0 = m_tile * BM;
  const int n0 = n_tile * BN;
  const int kchunk = K / splitk;         // multiple of BK (host-checked)
  const int k0 = split * kchunk;
  const int nk = kchunk / BK;

  __shared__ unsigned short Xl[2][BM * BK];
  __shared__ unsigned short Wl[2][BN * BK];

  const int t = threadIdx.x;
  const int lane = t & (WAVE_SIZE - 1);
  const int wid = t / WAVE_SIZE;
  const int wm = wid >> 2;               // 0..1  (row half)
  const int wn = wid & 3;                // 0..3  (col quarter)
  const int lc = lane & 15;
  const int lg = lane >> 4;

  // staging geometry: each thread owns fixed linear LDS chunks of 16 B.
  // X tile: 256*128 B = 32 KiB -> 2048 chunks -> 4 per thread
  // W tile: 128*128 B = 16 KiB -> 2048/2    -> 2 per thread
  // linear LDS byte L of a chunk maps to logical (row = L/128,
  // colbyte = (L%128) ^ swz(row)); the global source reads that element.
  auto stage = [&](int buf, int kb) {
    const long kbase = k0 + (long)kb * BK;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int L = (t + r * SG_THREADS) * 16;     // linear byte offset
      const int row = L / (BK * 2);
      const int colbyte = (L % (BK * 2)) ^ ((row & 7) << 4);
      int grow = m0 + row;
      if (grow >= M) grow = M - 1;                 // clamp (masked at write)
      const unsigned short* src = x + (long)grow * K + kbase + colbyte / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (reinterpret_cast<char*>(Xl[buf]) + L),
          16, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      const int L = (t + r * SG_THREADS) * 16;
      const int row = L / (BK * 2);
      const int colbyte = (L % (BK * 2)) ^ ((row & 7) << 4);
      const unsigned short* src = w + (long)(n0 + row) * K + kbase + colbyte / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (reinterpret_cast<char*>(Wl[buf]) + L),
          16, 0, 0);
    }
  };

  f32x4 acc[8][2];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int buf = 0;
  for (int kb = 0; kb < nk; ++kb) {
    if (kb + 1 < nk) stage(buf ^ 1, kb + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      u16x8 bfrag[2];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int row = wn * 32 + ni * 16 + lc;      // W row (output col)
        const int colbyte = (kk * 32 + lg * 8) * 2;
        bfrag[ni] = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<char*>(Wl[buf]) + sg_swz(row, colbyte));
      }
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        const int row = wm * 128 + mi * 16 + lc;     // X row (output row)
        const int colbyte = (kk * 32 + lg * 8) * 2;
        const u16x8 a = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<char*>(Xl[buf]) + sg_swz(row, colbyte));
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = sg_mfma(a, bfrag[ni], acc[mi][ni]);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    buf ^= 1;
  }

  // epilogue: D row = (l>>4)*4 + r, col = l&15 per 16x16 fragment
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = m0 + wm * 128 + mi * 16 + lg * 4 + r;
        const int gcol = n0 + wn * 32 + ni * 16 + lc;
        if (grow >= M) continue;
        if (WRITE_PARTIAL) {
          float* p = reinterpret_cast<float*>(out);
          p[((long)split * M + grow) * N + gcol] = acc[mi][ni][r];
        } else {
          unsigned short* o = reinterpret_cast<unsigned short*>(out);
          o[(long)grow * N + gcol] = f2bf(acc[mi][ni][r]);
        }
      }
    }
  }
}

// partial[S, M, N] f32 -> out[M, N] bf16
__global__ void sg_reduce_kernel(unsigned short* __restrict__ out,
                                 const float* __restrict__ partial,
                                 long MN, int splitk) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < MN;
       i += (long)gridDim.x * blockDim.x) {
    float a = 0.f;
    for (int s = 0; s < splitk; ++s) a += partial[s * MN + i];
    out[i] = f2bf(a);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// v2: 8-phase counted-vmcnt schedule (guide §5 256^2 template, re-derived).
//
// 256x256 tile, BK=64, 8 waves each owning a 256x32 C strip. Per K-tile:
// 4 phases; phase q computes C rows [64q, 64q+64) (16 MFMA) while staging
// one half-tile of the NEXT K-tile (order B0,B1,A0,A1). Counted waits —
// vmcnt(4) after phase 1, vmcnt(2) after phase 3 — keep 2-4 global_load_lds
// in flight across barriers (never drained to 0 in the loop); safety is by
// construction: every wave issues the same load sequence, so its own
// vmcnt(N) + the phase barrier guarantees the half-tiles older than N are
// visible to ALL waves before any wave reads them.
// ---------------------------------------------------------------------------
namespace {

constexpr int V2_BM = 256;
constexpr int V2_BN = 256;

template <bool WRITE_PARTIAL>
__global__ __launch_bounds__(SG_THREADS) void skinny_gemm_v2_kernel(
    void* __restrict__ out,
    const unsigned short* __restrict__ x,   // [M, K]
    const unsigned short* __restrict__ w,   // [N, K]
    int M, int N, int K, int splitk) {
  const int mt = (M + V2_BM - 1) / V2_BM;
  const int nt = N / V2_BN;
  int tid = blockIdx.x;
  const int split = tid % splitk;
  tid /= splitk;
  const int m0 = (tid % mt) * V2_BM;
  const int n0 = (tid / mt) * V2_BN;
  const int kchunk = K / splitk;
  const int k0 = split * kchunk;
  const int nk = kchunk / BK;

  __shared__ unsigned short Xl[2][V2_BM * BK];
  __shared__ unsigned short Wl[2][V2_BN * BK];

  const int t = threadIdx.x;
  const int lane = t & (WAVE_SIZE - 1);
  const int wid = t / WAVE_SIZE;          // wave's 32-col C strip
  const int lc = lane & 15;
  const int lg = lane >> 4;

  // one half-tile (128 rows x 64 cols = 16 KiB) = 2 chunks of 16 B per
  // thread; linear LDS byte L -> logical (row = L/128, colbyte ^ swz(row))
  auto stage_half = [&](unsigned short* lds_base, const unsigned short* gsrc,
                        int row0, int rowmax, long ld) {
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      const int L = (t + r * SG_THREADS) * 16;
      const int row = L / (BK * 2);
      const int colbyte = (L % (BK * 2)) ^ ((row & 7) << 4);
      int grow = row0 + row;
      if (grow > rowmax) grow = rowmax;
      const unsigned short* src = gsrc + (long)grow * ld + colbyte / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (reinterpret_cast<char*>(lds_base) + (row0 % V2_BM) * (BK * 2) + L),
          16, 0, 0);
    }
  };
  // stage order within a tile's 4 phases: B0, B1, A0, A1
  auto stage_phase = [&](int buf, int kb, int q) {
    const long kb0 = k0 + (long)kb * BK;
    if (q == 0)      stage_half(Wl[buf], w + kb0, n0,       N - 1, K);
    else if (q == 1) stage_half(Wl[buf], w + kb0, n0 + 128, N - 1, K);
    else if (q == 2) stage_half(Xl[buf], x + kb0, m0,       M - 1, K);
    else             stage_half(Xl[buf], x + kb0, m0 + 128, M - 1, K);
  };

  f32x4 acc[16][2];
#pragma unroll
  for (int mi = 0; mi < 16; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: full tile 0
#pragma unroll
  for (int q = 0; q < 4; ++q) stage_phase(0, 0, q);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int buf = 0;
  for (int kb = 0; kb < nk; ++kb) {
    const bool more = kb + 1 < nk;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (more) stage_phase(buf ^ 1, kb + 1, q);
      // B-frags: wave's 32-col strip (W rows wid*32 + ni*16 + lc)
      u16x8 bfrag[2][2];  // [ni][kk]
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int row = wid * 32 + ni * 16 + lc;
          bfrag[ni][kk] = *reinterpret_cast<const u16x8*>(
              reinterpret_cast<char*>(Wl[buf]) +
              sg_swz(row, (kk * 32 + lg * 8) * 2));
        }
      // A-frags: C rows [64q, 64q+64)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int arow = q * 64 + mi * 16 + lc;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const u16x8 a = *reinterpret_cast<const u16x8*>(
              reinterpret_cast<char*>(Xl[buf]) +
              sg_swz(arow, (kk * 32 + lg * 8) * 2));
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[q * 4 + mi][ni] = sg_mfma(a, bfrag[ni][kk], acc[q * 4 + mi][ni]);
          __builtin_amdgcn_s_setprio(0);
        }
      }
      if (more) {
        if (q == 1) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else if (q == 3) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      } else if (q == 3) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __syncthreads();
    }
    buf ^= 1;
  }

  // epilogue
#pragma unroll
  for (int mi = 0; mi < 16; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = m0 + mi * 16 + lg * 4 + r;
        const int gcol = n0 + wid * 32 + ni * 16 + lc;
        if (grow >= M) continue;
        if (WRITE_PARTIAL) {
          reinterpret_cast<float*>(out)[((long)split * M + grow) * N + gcol] =
              acc[mi][ni][r];
        } else {
          reinterpret_cast<unsigned short*>(out)[(long)grow * N + gcol] =
              f2bf(acc[mi][ni][r]);
        }
      }
    }
  }
}

}  // namespace

void skinny_gemm_v2_launch(void* out, const void* x, const void* w,
                           void* workspace, int M, int N, int K, int splitk,
                           hipStream_t s) {
  const int mt = (M + V2_BM - 1) / V2_BM;
  const int nt = N / V2_BN;
  dim3 grid(mt * nt * splitk);
  dim3 block(SG_THREADS);
  if (splitk > 1) {
    hipLaunchKernelGGL((skinny_gemm_v2_kernel<true>), grid, block, 0, s,
                       workspace, (const unsigned short*)x,
                       (const unsigned short*)w, M, N, K, splitk);
    long MN = (long)M * N;
    int rgrid = (int)((MN + 255) / 256);
    if (rgrid > 2048) rgrid = 2048;
    if (rgrid < 1) rgrid = 1;
    hipLaunchKernelGGL(sg_reduce_kernel, dim3(rgrid), dim3(256), 0, s,
                       (unsigned short*)out, (const float*)workspace, MN,
                       splitk);
  } else {
    hipLaunchKernelGGL((skinny_gemm_v2_kernel<false>), grid, block, 0, s, out,
                       (const unsigned short*)x, (const unsigned short*)w, M,
                       N, K, splitk);
  }
}

void skinny_gemm_launch(void* out, const void* x, const void* w,
                        void* workspace, int M, int N, int K, int splitk,
                        hipStream_t s) {
  const int mt = (M + BM - 1) / BM;
  const int nt = N / BN;
  dim3 grid(mt * nt * splitk);
  dim3 block(SG_THREADS);
  if (splitk > 1) {
    hipLaunchKernelGGL((skinny_gemm_kernel<true>), grid, block, 0, s,
                       workspace, (const unsigned short*)x,
                       (const unsigned short*)w, M, N, K, splitk);
    long MN = (long)M * N;
    int rgrid = (int)((MN + 255) / 256);
    if (rgrid > 2048) rgrid = 2048;
    if (rgrid < 1) rgrid = 1;
    hipLaunchKernelGGL(sg_reduce_kernel, dim3(rgrid), dim3(256), 0, s,
                       (unsigned short*)out, (const float*)workspace, MN,
                       splitk);
  } else {
    hipLaunchKernelGGL((skinny_gemm_kernel<false>), grid, block, 0, s, out,
                       (const unsigned short*)x, (const unsigned short*)w, M,
                       N, K, splitk);
  }
}
