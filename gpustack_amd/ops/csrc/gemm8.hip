// 8-phase pipelined bf16 GEMM for MI355X (gfx950) — C[M,N] = X[M,K]·W[N,K]^T.
//
// The counted-vmcnt 256x256 schedule from the CDNA4 guide (verified
// template: 1563 TF @4k), re-derived for this codebase; see
// profiles/r03_dot2_decode_profile.md "Round-2 notes" for the full
// derivation that fixes the staging choreography:
//
//  - tile 256x256, BK=64 split into two 32-wide K-halves; LDS half-tile =
//    [256 rows][32 k] bf16 (16 KiB), double-buffered per tile parity:
//    2 buf x 2 half x (A,B) = 128 KiB.
//  - 8 waves (2M x 4N), 512 threads; wave output 128x64; each phase is one
//    (k-half, m-half) quadrant = 16 x mfma_f32_16x16x32_bf16.
//  - per phase: {ds_read frags; stage ONE half-tile via global_load_lds;
//    barrier; lgkmcnt(0); setprio(1); 16 MFMA; setprio(0); [vmcnt(6) at
//    phases 4/8]; barrier}. Staging runs 7 half-tiles ahead (prologue
//    4+3), so vmcnt(6) (= 3 halves x 2 loads/wave in flight) lands the
//    next tile exactly when consumed.
//  - LDS swizzle: byte ^= ((byte>>9)&1)<<5 (row stride 64 B -> rows 8..15
//    of each 16-row subtile swap 32 B k-halves), applied via pre-swizzled
//    global source at store and swizzled ds_read address at load.
//  - XCD-aware bijective workgroup remap so consecutive tiles land on one
//    XCD's L2 (8 XCDs).
//
// Constraints (dispatcher falls back to hipBLASLt otherwise):
//   N % 256 == 0, K % 128 == 0 (even tile count), any M (loads clamp,
//   stores guard). Targets decode-shape GEMMs that fill the chip without
//   split-K (lm_head N=128256 -> 1002 WGs, gate_up N=28672 -> 224).
#include "common.h"

namespace {

constexpr int G8_BM = 256;
constexpr int G8_BN = 256;
constexpr int G8_BK = 64;
constexpr int G8_THREADS = 512;

typedef __attribute__((ext_vector_type(8))) short g8_s16x8;

DEVICE_INLINE f32x4 g8_mfma(u16x8 a, u16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(g8_s16x8, a), __builtin_bit_cast(g8_s16x8, b), c, 0, 0, 0);
}

DEVICE_INLINE int g8_swz(int byte) { return byte ^ (((byte >> 9) & 1) << 5); }

template <bool SAFE>  // SAFE: drain vmcnt(0) + barrier after every stage
__global__ __launch_bounds__(G8_THREADS) void gemm8_kernel(
    unsigned short* __restrict__ out,      // [M, N] bf16
    const unsigned short* __restrict__ x,  // [M, K] bf16
    const unsigned short* __restrict__ w,  // [N, K] bf16
    int M, int N, int K) {
  // 2 buf x 2 khalf x 16 KiB per operand
  __shared__ unsigned short Al[2][2][G8_BM * 32];
  __shared__ unsigned short Bl[2][2][G8_BN * 32];

  const int NT = K / G8_BK;
  const int mtiles = (M + G8_BM - 1) / G8_BM;
  const int ntiles = N / G8_BN;
  const int nwg = mtiles * ntiles;
  // bijective XCD swizzle (8 XCDs)
  int wg = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg % mtiles) * G8_BM;
  const int n0 = (wg / mtiles) * G8_BN;

  const int t = threadIdx.x;
  const int lane = t & (WAVE_SIZE - 1);
  const int wid = t / WAVE_SIZE;
  const int wm = wid >> 2;        // 0..1: row half (128 rows)
  const int wn = wid & 3;         // 0..3: col quarter (64 cols)
  const int lc = lane & 15;
  const int lg = lane >> 4;

  // ---- staging: half h of the stream; tile = h>>2, id = h&3
  //      (0 = A k0, 1 = B k0, 2 = A k1, 3 = B k1)
  auto stage = [&](int h) {
    const int tile = h >> 2;
    const int id = h & 3;
    const int kh = id >> 1;
    const bool isA = (id & 1) == 0;
    unsigned short* dst = isA ? Al[tile & 1][kh] : Bl[tile & 1][kh];
    const long kbase = (long)tile * G8_BK + kh * 32;
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      const int L = (t + r * G8_THREADS) * 16;  // linear byte in half-tile
      const int Ls = g8_swz(L);                 // fetch what belongs here
      const int row = Ls / 64;
      const int kb = Ls % 64;
      long grow;
      if (isA) {
        grow = m0 + row;
        if (grow >= M) grow = M - 1;            // clamp; stores are guarded
      } else {
        grow = n0 + row;
      }
      const unsigned short* src =
          (isA ? x : w) + grow * K + kbase + kb / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (reinterpret_cast<char*>(dst) + L),
          16, 0, 0);
    }
    if (SAFE) {
      asm volatile("s_waitcnt vmcnt(0)");
      __syncthreads();
    }
  };

  f32x4 acc[2][4][4];  // [mhalf][rowtile][coltile]
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int rt = 0; rt < 4; ++rt)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) acc[mh][rt][ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  // frag LDS byte offsets (within a half-tile, swizzled)
  auto a_off = [&](int mh, int rt) {
    const int row = wm * 128 + mh * 64 + rt * 16 + lc;
    return g8_swz(row * 64 + lg * 16);
  };
  auto b_off = [&](int ct) {
    const int col = wn * 64 + ct * 16 + lc;
    return g8_swz(col * 64 + lg * 16);
  };

  u16x8 afr[4], bfr[4];
  // one phase: quadrant (kh, mh) of tile `tile`; stages stream half `sh`
  auto phase = [&](int tile, int kh, int mh, int sh, bool wait6) {
    const int buf = tile & 1;
    if (mh == 0) {
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        bfr[ct] = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<const char*>(Bl[buf][kh]) + b_off(ct));
    }
#pragma unroll
    for (int rt = 0; rt < 4; ++rt)
      afr[rt] = *reinterpret_cast<const u16x8*>(
          reinterpret_cast<const char*>(Al[buf][kh]) + a_off(mh, rt));
    if (sh >= 0) stage(sh);
    // ONE barrier per phase: paces the waves (LDS-slot safety additionally
    // rides the >=600ns global-load latency, see profiles notes); the
    // compiler inserts fine-grained lgkmcnt waits per MFMA operand, so
    // early MFMAs overlap the tail of the ds_read burst.
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int rt = 0; rt < 4; ++rt)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        acc[mh][rt][ct] = g8_mfma(afr[rt], bfr[ct], acc[mh][rt][ct]);
    __builtin_amdgcn_s_setprio(0);
    if (wait6 && !SAFE) asm volatile("s_waitcnt vmcnt(6)");
  };

  // ---- prologue: stage tile0 fully + 3 halves of tile1 (7 ahead)
  const int total_halves = 4 * NT;
  int pos = 0;
  for (; pos < 4 && pos < total_halves; ++pos) stage(pos);
  if (!SAFE) asm volatile("s_waitcnt vmcnt(4)");
  for (; pos < 7 && pos < total_halves; ++pos) stage(pos);
  if (NT == 2 && pos < total_halves) stage(pos++);
  if (!SAFE) asm volatile("s_waitcnt vmcnt(6)");
  if (NT == 2) asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  // ---- main loop: tile pairs up to NT-4 (staging stays in range)
  int T = 0;
  for (; T + 2 < NT; T += 2) {
#pragma unroll
    for (int p = 0; p < 8; ++p) {
      const int tile = T + (p >> 2);
      const int kh = (p >> 1) & 1;
      const int mh = p & 1;
      const int sh = (pos < total_halves) ? pos++ : -1;
      phase(tile, kh, mh, sh, p == 3 || p == 7);
    }
  }
  // ---- epilogue: stage the stream tail (the 7-ahead offset leaves the
  // final half un-staged by the main loop), then drain
  while (pos < total_halves) stage(pos++);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();
  for (; T < NT; ++T) {
#pragma unroll
    for (int p = 0; p < 4; ++p)
      phase(T, (p >> 1) & 1, p & 1, -1, false);
  }

  // ---- writeback: lane holds D[row=lg*4+r][col=lc] per 16x16 tile
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int rt = 0; rt < 4; ++rt) {
      const int row_base = m0 + wm * 128 + mh * 64 + rt * 16 + lg * 4;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const int col = n0 + wn * 64 + ct * 16 + lc;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = row_base + r;
          if (row < M) out[(long)row * N + col] = f2bf(acc[mh][rt][ct][r]);
        }
      }
    }
}

}  // namespace

void gemm8_launch(void* out, const void* x, const void* w, int M, int N,
                  int K, int safe, int* err_unsupported, hipStream_t s) {
  *err_unsupported = 0;
  if (N % G8_BN != 0 || K % (2 * G8_BK) != 0 || K < 2 * G8_BK) {
    *err_unsupported = 1;
    return;
  }
  const int mtiles = (M + G8_BM - 1) / G8_BM;
  dim3 grid(mtiles * (N / G8_BN));
  dim3 block(G8_THREADS);
  if (safe) {
    hipLaunchKernelGGL((gemm8_kernel<true>), grid, block, 0, s,
                       (unsigned short*)out, (const unsigned short*)x,
                       (const unsigned short*)w, M, N, K);
  } else {
    hipLaunchKernelGGL((gemm8_kernel<false>), grid, block, 0, s,
                       (unsigned short*)out, (const unsigned short*)x,
                       (const unsigned short*)w, M, N, K);
  }
}
