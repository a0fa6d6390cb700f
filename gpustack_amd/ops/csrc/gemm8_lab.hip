// GEMM schedule lab for MI355X (gfx950) — structural variants of the
// 8-phase 256x256 bf16 GEMM (gemm8.hip), benchmarked against hipBLASLt by
// scripts/bench_gemm_lab.py on decode shapes (lm_head M=512 N=128256,
// gate_up N=28672) where the r03 roofline shows hipBLASLt at 602-1196 TF
// vs the 1563 TF the guide's verified template reaches at prefill shapes.
//
// Variants (MODE):
//   0 = production gemm8 structure (1 barrier/phase, compiler lgkmcnt)
//   1 = template-exact: {ds_read; stage; barrier; lgkmcnt(0); setprio(1);
//       MFMA; setprio(0); [vmcnt(6) @ phase 4/8]; barrier}
//   2 = register-pipelined fragments: phase p ds_reads the frags for
//       phase p+1 while MFMAing phase p (double-buffered afr/bfr), one
//       barrier per phase; vmcnt moves one phase earlier (end of p2/p6,
//       count 4) because next-tile frag reads start at p3/p7.
//   3 = MODE 2 with a barrier only every second phase (halved barrier
//       count; LDS clobber-safety rides the HBM round-trip latency).
//
// All variants share the staging/LDS/swizzle geometry of gemm8.hip so an
// A/B isolates the schedule alone.
#include "common.h"

namespace {

constexpr int GL_BM = 256;
constexpr int GL_BN = 256;
constexpr int GL_BK = 64;
constexpr int GL_THREADS = 512;

typedef __attribute__((ext_vector_type(8))) short gl_s16x8;

DEVICE_INLINE f32x4 gl_mfma(u16x8 a, u16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(gl_s16x8, a), __builtin_bit_cast(gl_s16x8, b), c, 0, 0, 0);
}

DEVICE_INLINE int gl_swz(int byte) { return byte ^ (((byte >> 9) & 1) << 5); }

template <int MODE>
__global__ __launch_bounds__(GL_THREADS) void gemm_lab_kernel(
    unsigned short* __restrict__ out,      // [M, N] bf16
    const unsigned short* __restrict__ x,  // [M, K] bf16
    const unsigned short* __restrict__ w,  // [N, K] bf16
    int M, int N, int K) {
  __shared__ unsigned short Al[2][2][GL_BM * 32];
  __shared__ unsigned short Bl[2][2][GL_BN * 32];

  const int NT = K / GL_BK;
  const int mtiles = (M + GL_BM - 1) / GL_BM;
  const int ntiles = N / GL_BN;
  const int nwg = mtiles * ntiles;
  int wg = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg % mtiles) * GL_BM;
  const int n0 = (wg / mtiles) * GL_BN;

  const int t = threadIdx.x;
  const int lane = t & (WAVE_SIZE - 1);
  const int wid = t / WAVE_SIZE;
  const int wm = wid >> 2;
  const int wn = wid & 3;
  const int lc = lane & 15;
  const int lg = lane >> 4;

  // per-lane invariant staging addresses (element offsets into x/w and
  // byte offsets into the LDS half-tile) — the div/mod/swizzle chain runs
  // once instead of per stage call, cutting VALU work and live registers
  long a_src[2], b_src[2];
#pragma unroll
  for (int r = 0; r < 2; ++r) {
    const int L = (t + r * GL_THREADS) * 16;
    const int Ls = gl_swz(L);
    const int row = Ls / 64;
    const int kb = Ls % 64;
    long arow = m0 + row;
    if (arow >= M) arow = M - 1;  // clamp; stores are guarded
    a_src[r] = arow * (long)K + kb / 2;
    b_src[r] = (long)(n0 + row) * K + kb / 2;
  }
  auto stage = [&](int h) {
    const int tile = h >> 2;
    const int id = h & 3;
    const int kh = id >> 1;
    const bool isA = (id & 1) == 0;
    unsigned short* dst = isA ? Al[tile & 1][kh] : Bl[tile & 1][kh];
    const int kbase = tile * GL_BK + kh * 32;
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      const unsigned short* src =
          (isA ? x + a_src[r] : w + b_src[r]) + kbase;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              (reinterpret_cast<char*>(dst) + (t + r * GL_THREADS) * 16),
          16, 0, 0);
    }
  };

  f32x4 acc[2][4][4];
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int rt = 0; rt < 4; ++rt)
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) acc[mh][rt][ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  auto a_off = [&](int mh, int rt) {
    const int row = wm * 128 + mh * 64 + rt * 16 + lc;
    return gl_swz(row * 64 + lg * 16);
  };
  auto b_off = [&](int ct) {
    const int col = wn * 64 + ct * 16 + lc;
    return gl_swz(col * 64 + lg * 16);
  };

  const int total_halves = 4 * NT;

  if constexpr (MODE <= 1) {
    u16x8 afr[4], bfr[4];
    auto phase = [&](int tile, int kh, int mh, int sh, bool waitv) {
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
      __builtin_amdgcn_s_barrier();
      if constexpr (MODE == 1) asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int rt = 0; rt < 4; ++rt)
#pragma unroll
        for (int ct = 0; ct < 4; ++ct)
          acc[mh][rt][ct] = gl_mfma(afr[rt], bfr[ct], acc[mh][rt][ct]);
      __builtin_amdgcn_s_setprio(0);
      if (waitv) asm volatile("s_waitcnt vmcnt(6)");
      if constexpr (MODE == 1) __builtin_amdgcn_s_barrier();
    };

    int pos = 0;
    for (; pos < 4 && pos < total_halves; ++pos) stage(pos);
    asm volatile("s_waitcnt vmcnt(4)");
    for (; pos < 7 && pos < total_halves; ++pos) stage(pos);
    if (NT == 2 && pos < total_halves) stage(pos++);
    asm volatile("s_waitcnt vmcnt(6)");
    if (NT == 2) asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();

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
    while (pos < total_halves) stage(pos++);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    for (; T < NT; ++T) {
#pragma unroll
      for (int p = 0; p < 4; ++p)
        phase(T, (p >> 1) & 1, p & 1, -1, false);
    }
  } else {
    // MODE 2/3: A-fragments register-pipelined one phase ahead (A is the
    // per-phase recurring ds_read; prefetching it means the MFMA burst
    // only ever waits on the 4 B reads at the two quadrant-pair openers
    // per tile). Full A+B double-buffering spills 23 VGPRs past the
    // 256-per-wave cap at 2 waves/SIMD, so B stays in-phase and
    // single-buffered (WAR on bfr orders naturally: its last consumer
    // MFMA issued in the previous phase).
    // afr_pf: rt0/rt1 of the NEXT phase, prefetched (they head the MFMA
    // burst's critical path); afr_t: rt2/rt3 of the CURRENT phase, read
    // in-phase (their wait hides behind the rt0/rt1 MFMAs). Full A+B
    // double-buffering spills past the 256-VGPR cap; this costs +24 regs.
    u16x8 afr_pf[2][2], afr_t[2], bfr[4];
    auto load_b = [&](int tile, int kh) {
      const int buf = tile & 1;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
        bfr[ct] = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<const char*>(Bl[buf][kh]) + b_off(ct));
    };
    auto load_a_head = [&](int slot, int tile, int kh, int mh) {
      const int buf = tile & 1;
#pragma unroll
      for (int rt = 0; rt < 2; ++rt)
        afr_pf[slot][rt] = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<const char*>(Al[buf][kh]) + a_off(mh, rt));
    };
    auto load_a_tail = [&](int tile, int kh, int mh) {
      const int buf = tile & 1;
#pragma unroll
      for (int rt = 2; rt < 4; ++rt)
        afr_t[rt - 2] = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<const char*>(Al[buf][kh]) + a_off(mh, rt));
    };
    auto mfma_burst = [&](int slot, int mh) {
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int rt = 0; rt < 4; ++rt) {
        const u16x8 a = rt < 2 ? afr_pf[slot][rt] : afr_t[rt - 2];
#pragma unroll
        for (int ct = 0; ct < 4; ++ct)
          acc[mh][rt][ct] = gl_mfma(a, bfr[ct], acc[mh][rt][ct]);
      }
      __builtin_amdgcn_s_setprio(0);
    };

    int pos = 0;
    for (; pos < 4 && pos < total_halves; ++pos) stage(pos);
    asm volatile("s_waitcnt vmcnt(4)");
    for (; pos < 7 && pos < total_halves; ++pos) stage(pos);
    asm volatile("s_waitcnt vmcnt(6)");
    if (NT == 2) asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    load_a_head(0, 0, 0, 0);  // prologue: head A frags for phase 0

    int T = 0;
    for (; T + 2 < NT; T += 2) {
#pragma unroll
      for (int p = 0; p < 8; ++p) {
        const int tile = T + (p >> 2);
        const int kh = (p >> 1) & 1;
        if ((p & 1) == 0) load_b(tile, kh);  // quadrant-pair opener
        load_a_tail(tile, kh, p & 1);
        // prefetch next phase's head A frags (wraps into the next
        // 8-phase block at p==7: tile T+2, whose h0 the p==6 vmcnt
        // landed)
        const int pn = p + 1;
        load_a_head(pn & 1, T + (pn >> 2), (pn >> 1) & 1, pn & 1);
        const int sh = (pos < total_halves) ? pos++ : -1;
        if (sh >= 0) stage(sh);
        // next-tile frag reads start at p3/p7: its halves must have
        // landed one phase earlier than in MODE 0/1
        if (p == 2 || p == 6) asm volatile("s_waitcnt vmcnt(4)");
        mfma_burst(p & 1, p & 1);
        if constexpr (MODE == 2) {
          __builtin_amdgcn_s_barrier();
        } else if constexpr (MODE == 3) {
          if (p & 1) __builtin_amdgcn_s_barrier();
        } else {  // MODE 5: barriers at EVEN phases — they follow the
          // p2/p6 vmcnt(4), so each wave's "my tile-T+1 loads landed"
          // becomes a collective guarantee before any wave's p3/p7 reads
          // (the odd-phase cadence of MODE 3 lacked this propagation and
          // raced on the 1002-WG lm_head grid)
          if ((p & 1) == 0) __builtin_amdgcn_s_barrier();
        }
      }
    }
    while (pos < total_halves) stage(pos++);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    // epilogue: last 2 (or NT if tiny) tiles, read-then-compute
    {
      bool first = true;
      for (; T < NT; ++T) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          if ((p & 1) == 0) load_b(T, (p >> 1) & 1);
          load_a_tail(T, (p >> 1) & 1, p & 1);
          // phase 0's head frags were pre-loaded by the main loop's
          // last iteration — except when the main loop never ran
          if (!first || p > 0 || NT <= 2)
            load_a_head(p & 1, T, (p >> 1) & 1, p & 1);
          asm volatile("s_waitcnt lgkmcnt(0)");
          mfma_burst(p & 1, p & 1);
        }
        first = false;
      }
    }
  }

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

void gemm_lab_launch(void* out, const void* x, const void* w, int M, int N,
                     int K, int mode, int* err_unsupported, hipStream_t s) {
  *err_unsupported = 0;
  if (N % GL_BN != 0 || K % (2 * GL_BK) != 0 || K < 2 * GL_BK) {
    *err_unsupported = 1;
    return;
  }
  if (mode == 5) mode = 4;  // MODE index 4 template instantiation
  const int mtiles = (M + GL_BM - 1) / GL_BM;
  dim3 grid(mtiles * (N / GL_BN));
  dim3 block(GL_THREADS);
  switch (mode) {
    case 0:
      hipLaunchKernelGGL((gemm_lab_kernel<0>), grid, block, 0, s,
                         (unsigned short*)out, (const unsigned short*)x,
                         (const unsigned short*)w, M, N, K);
      break;
    case 1:
      hipLaunchKernelGGL((gemm_lab_kernel<1>), grid, block, 0, s,
                         (unsigned short*)out, (const unsigned short*)x,
                         (const unsigned short*)w, M, N, K);
      break;
    case 2:
      hipLaunchKernelGGL((gemm_lab_kernel<2>), grid, block, 0, s,
                         (unsigned short*)out, (const unsigned short*)x,
                         (const unsigned short*)w, M, N, K);
      break;
    case 3:
      hipLaunchKernelGGL((gemm_lab_kernel<3>), grid, block, 0, s,
                         (unsigned short*)out, (const unsigned short*)x,
                         (const unsigned short*)w, M, N, K);
      break;
    case 4:
      hipLaunchKernelGGL((gemm_lab_kernel<4>), grid, block, 0, s,
                         (unsigned short*)out, (const unsigned short*)x,
                         (const unsigned short*)w, M, N, K);
      break;
    default:
      *err_unsupported = 2;
  }
}
