// Varlen causal flash-attention prefill for MI355X (gfx950) on MFMA — v2.
//
// Replaces the prefill attention the reference gets from vLLM/SGLang
// (SURVEY.md §2.9 #1). CDNA4-first structure (not a CUDA port):
//
//  - mfma_f32_16x16x32_bf16 tiles; wavefront-64 fragment layouts:
//      A-frag: lane holds A[row = l&15][k = (l>>4)*8 + j]   (u16x8)
//      B-frag: lane holds B[k = (l>>4)*8 + j][col = l&15]   (u16x8)
//      C/D  : lane holds D[row = (l>>4)*4 + r][col = l&15]  (f32x4)
//  - swapped QK^T (S^T = K · Q^T) so BOTH operands are row-contiguous
//    u16x8 reads (K from LDS, Q from registers).
//  - K tile LDS-staged with the ((row&7)<<4) byte-XOR swizzle (guide §6 G4).
//  - V staged TRANSPOSED (VT[D][BK+pad]) so PV B-fragments are b128 reads.
//  - v2: BK = 64 kv per tile (twice the MFMA work per barrier pair) and
//    async-stage split (guide T14): each iteration ISSUES the next tile's
//    global loads into registers BEFORE the MFMA work, then writes them to
//    LDS after the barrier — HBM latency hides under compute.
//  - online softmax in f32 registers; per-q-row stats shared via shuffles.
//  - templated head_dim D in {64, 128} (GPT-OSS is D=64); optional
//    per-head attention-SINK logits (join the softmax denominator only)
//    and sliding WINDOW (kvpos in (qpos-window, qpos]), matching
//    torch_ref._sink_softmax / _window_mask exactly.
//
// Tiling: BQ = 64 q rows per workgroup (4 waves x 16 rows), BK = 64 kv.
#include "common.h"

namespace {

constexpr int BQ = 64;
constexpr int BK = 64;
constexpr int PF_WAVES = 4;
constexpr int PF_THREADS = PF_WAVES * WAVE_SIZE;
constexpr int VT_PAD = 4;  // elements; breaks the bank cycle on VT/P rows
constexpr int NST = BK / 16;       // S^T stiles per tile (4)
constexpr int KC2 = BK / 32;       // PV k-chunks per tile (2)

typedef __attribute__((ext_vector_type(8))) short s16x8;

DEVICE_INLINE f32x4 mfma16x16x32_bf16(u16x8 a, u16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(s16x8, a), __builtin_bit_cast(s16x8, b), c, 0, 0, 0);
}

// byte offset of (kv, dbyte) in the swizzled K tile
template <int D>
DEVICE_INLINE int kswz(int kv, int dbyte) {
  return kv * (D * 2) + (dbyte ^ ((kv & 7) << 4));
}

// EXT=false is the exact validated fast path (no sinks/window/softcap
// code); EXT=true carries the GPT-OSS/Gemma variants. DVK is the VALUE
// head dim (defaults to D; MLA expand-prefill runs D=192 qk over DVK=128
// values).
template <int D, bool EXT = false, int DVK = D>
__global__ __launch_bounds__(PF_THREADS) void flash_prefill_kernel(
    unsigned short* __restrict__ out,      // [T, Hq, DVK]
    const unsigned short* __restrict__ q,  // [T, Hq, D]
    const unsigned short* __restrict__ k,  // [T, Hkv, D]
    const unsigned short* __restrict__ v,  // [T, Hkv, DVK]
    const int* __restrict__ tile_start,    // [ntiles] seq start (global row)
    const int* __restrict__ tile_q0,       // [ntiles] q-tile offset in seq
    const int* __restrict__ tile_len,      // [ntiles] seq length
    int Hq, int Hkv, float scale,
    long qs, long ks, long vs,             // row strides (elements)
    const float* __restrict__ sinks,       // [Hq] or null
    int window,                            // 0 = full causal
    float softcap) {                       // 0 = off (Gemma-2 tanh cap)
  constexpr int KU = BK * (D / 8) / PF_THREADS;    // K u16x8 units/thread
  constexpr int VU = DVK * (BK / 8) / PF_THREADS;  // VT units/thread
  constexpr int QK = D / 32;                       // q k-chunks
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hkv);
  const int seq0 = tile_start[tile];
  const int q0 = tile_q0[tile];
  const int len = tile_len[tile];

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lc = lane & 15;        // column / row-id within 16
  const int lg = lane >> 4;        // 4-lane group id
  const int tid = threadIdx.x;

  __shared__ unsigned short Kl[BK * D];            // swizzled
  __shared__ unsigned short VTl[DVK][BK + VT_PAD]; // transposed V
  __shared__ unsigned short Pl[PF_WAVES][16][BK + VT_PAD];

  // Hoist this wave's 16 q rows into B-fragments (QK k-chunks of 32).
  const int qrow_local = q0 + wave * 16 + lc;
  const int qrow_clamped = (qrow_local < len) ? qrow_local : (len - 1);
  u16x8 qfrag[QK];
#pragma unroll
  for (int kk = 0; kk < QK; ++kk) {
    const unsigned short* qp = q + (long)(seq0 + qrow_clamped) * qs +
                               (long)qh * D + kk * 32 + lg * 8;
    qfrag[kk] = *reinterpret_cast<const u16x8*>(qp);
  }

  float mcol = -INFINITY;  // running max for q row `lc` (this wave)
  float lcol = 0.f;        // running denom for q row `lc`
  f32x4 o[DVK / 16];       // O[q=(lg*4+r)][d=lc+nt*16]
#pragma unroll
  for (int nt = 0; nt < DVK / 16; ++nt) o[nt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_hi = q0 + BQ - 1;
  const int kv_end = min(len, q_hi + 1);           // causal bound
  const int ntiles_kv = (kv_end + BK - 1) / BK;
  const int wave_q_hi = q0 + wave * 16 + 15;       // this wave's causal bound
  const int wave_q_lo = q0 + wave * 16;            // lowest window start

  // staging assignment (fixed per thread):
  //   K: unit u = tid + r*256 -> (kv = u/(D/8), d0 = (u%(D/8))*8)
  //   VT: unit u -> (d = u%D, kvc = (u/D)*8)
  int kst_kv[KU];
#pragma unroll
  for (int r = 0; r < KU; ++r) kst_kv[r] = (tid + r * PF_THREADS) / (D / 8);
  const int kst_d0 = (tid % (D / 8)) * 8;
  const int vst_d = tid % DVK;
  const int vst_kvc0 = (tid / DVK) * 8;  // + r*(8*PF_THREADS/DVK) per round
  constexpr int VST_STEP = 8 * PF_THREADS / DVK;

  u16x8 kstage[KU];
  unsigned short vstage[VU][8];

  auto issue_loads = [&](int kv0) {
#pragma unroll
    for (int r = 0; r < KU; ++r) {
      const int kv = kv0 + kst_kv[r];
      kstage[r] = (kv < len)
          ? *reinterpret_cast<const u16x8*>(
                k + (long)(seq0 + kv) * ks + (long)kvh * D + kst_d0)
          : u16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
#pragma unroll
    for (int r = 0; r < VU; ++r) {
      const int kvc = vst_kvc0 + r * VST_STEP;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int kv = kv0 + kvc + j;
        vstage[r][j] = (kv < len)
            ? v[(long)(seq0 + kv) * vs + (long)kvh * DVK + vst_d]
            : (unsigned short)0;
      }
    }
  };

  auto write_lds = [&]() {
#pragma unroll
    for (int r = 0; r < KU; ++r) {
      *reinterpret_cast<u16x8*>(reinterpret_cast<char*>(Kl) +
                                kswz<D>(kst_kv[r], kst_d0 * 2)) = kstage[r];
    }
#pragma unroll
    for (int r = 0; r < VU; ++r) {
      *reinterpret_cast<u16x8*>(&VTl[vst_d][vst_kvc0 + r * VST_STEP]) =
          *reinterpret_cast<u16x8*>(vstage[r]);
    }
  };

  issue_loads(0);
  for (int kt = 0; kt < ntiles_kv; ++kt) {
    const int kv0 = kt * BK;
    __syncthreads();   // previous tile's LDS reads are done
    write_lds();
    __syncthreads();   // tile ready
    if (kt + 1 < ntiles_kv) issue_loads(kv0 + BK);  // hide HBM under MFMA

    if (kv0 > wave_q_hi) continue;  // fully masked for this wave
    // sliding window (EXT): tile entirely below every row's window
    if (EXT && window > 0 && kv0 + BK <= wave_q_lo - window + 1) continue;

    // ---- S^T = K · Q^T  (NST stiles x QK k-chunks) ----
    f32x4 st[NST];
#pragma unroll
    for (int i = 0; i < NST; ++i) st[i] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int kk = 0; kk < QK; ++kk) {
        const u16x8 a = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<char*>(Kl) +
            kswz<D>(stile * 16 + lc, (kk * 32 + lg * 8) * 2));
        st[stile] = mfma16x16x32_bf16(a, qfrag[kk], st[stile]);
      }
    }

    // ---- mask + online softmax (stats per q row lc) ----
    const int qpos = q0 + wave * 16 + lc;
    float sv[NST * 4];
    float tmax = -INFINITY;
#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvpos = kv0 + stile * 16 + lg * 4 + r;
        float x = st[stile][r] * scale;
        bool ok = (kvpos <= qpos) && (kvpos < len) && (qpos < len);
        if constexpr (EXT) {
          if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
          if (window > 0) ok = ok && (kvpos > qpos - window);
        }
        x = ok ? x : -INFINITY;
        sv[stile * 4 + r] = x;
        tmax = fmaxf(tmax, x);
      }
    }
#pragma unroll
    for (int msk = 16; msk <= 32; msk <<= 1)
      tmax = fmaxf(tmax, __shfl_xor(tmax, msk, WAVE_SIZE));

    const float nm = fmaxf(mcol, tmax);
    float corr = 1.f, tsum = 0.f;
    float pv[NST * 4];
    if (nm != -INFINITY) {
      corr = __expf(mcol - nm);
#pragma unroll
      for (int i = 0; i < NST * 4; ++i) {
        pv[i] = (sv[i] == -INFINITY) ? 0.f : __expf(sv[i] - nm);
        tsum += pv[i];
      }
      mcol = nm;
    } else {
#pragma unroll
      for (int i = 0; i < NST * 4; ++i) pv[i] = 0.f;
    }
#pragma unroll
    for (int msk = 16; msk <= 32; msk <<= 1)
      tsum += __shfl_xor(tsum, msk, WAVE_SIZE);
    lcol = lcol * corr + tsum;

    // ---- stage P (bf16) into this wave's LDS buffer: P[q=lc][kv] ----
#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        Pl[wave][lc][stile * 16 + lg * 4 + r] = f2bf(pv[stile * 4 + r]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- rescale O by this tile's correction (row-matched via shuffle) ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = lg * 4 + r;
      const float c = __shfl(corr, orow, WAVE_SIZE);
#pragma unroll
      for (int nt = 0; nt < DVK / 16; ++nt) o[nt][r] *= c;
    }

    // ---- O += P · V  (A = P from LDS, B = V^T rows from LDS) ----
#pragma unroll
    for (int kk2 = 0; kk2 < KC2; ++kk2) {
      const u16x8 pa = *reinterpret_cast<const u16x8*>(
          &Pl[wave][lc][kk2 * 32 + lg * 8]);
#pragma unroll
      for (int nt = 0; nt < DVK / 16; ++nt) {
        const u16x8 b = *reinterpret_cast<const u16x8*>(
            &VTl[nt * 16 + lc][kk2 * 32 + lg * 8]);
        o[nt] = mfma16x16x32_bf16(pa, b, o[nt]);
      }
    }
  }

  // ---- epilogue: normalize rows and write; the SINK logit joins the
  // denominator only (GPT-OSS: its probability mass drops, no value) ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = lg * 4 + r;
    float denom = __shfl(lcol, orow, WAVE_SIZE);
    float onum = 1.f;
    if constexpr (EXT) {
      if (sinks != nullptr) {
        const float m_row = __shfl(mcol, orow, WAVE_SIZE);
        const float sk = sinks[qh];
        const float M2 = fmaxf(m_row, sk);
        onum = (m_row == -INFINITY) ? 0.f : __expf(m_row - M2);
        denom = denom * onum + __expf(sk - M2);
      }
    }
    const int qrow = q0 + wave * 16 + orow;
    if (qrow >= len || denom <= 0.f) continue;
    const float inv = onum / denom;
#pragma unroll
    for (int nt = 0; nt < DVK / 16; ++nt) {
      out[((long)(seq0 + qrow) * Hq + qh) * DVK + nt * 16 + lc] =
          f2bf(o[nt][r] * inv);
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Prefill-with-history: same MFMA flash structure, but the K/V stream is
// gathered block-wise from the PAGED pool ([nblocks, Hkv, 16, D] bf16)
// through the sequence's block table. Used for prefix-cache suffixes and
// chunked-prefill continuations, whose rows previously ran as paged-decode
// rows at ~3.5x the attention cost (profiles/r03 chunked A/B) — the
// measured round-1 penalty this kernel removes. The new tokens' K/V are
// already scattered into the pool (reshape_and_cache runs before
// attention), so ALL positions stream from the pool uniformly.
// ---------------------------------------------------------------------------
namespace {

constexpr int PP_BS = 16;  // pool block size (tokens per KV block)

template <int D, bool EXT = false>
__global__ __launch_bounds__(PF_THREADS) void flash_prefill_paged_kernel(
    unsigned short* __restrict__ out,      // [T, Hq, D] (suffix rows)
    const unsigned short* __restrict__ q,  // [T, Hq, D] (suffix rows)
    const unsigned short* __restrict__ kc, // [nblocks, Hkv, BS, D]
    const unsigned short* __restrict__ vc, // [nblocks, Hkv, BS, D]
    const int* __restrict__ block_tables,  // [nseq, maxb]
    const int* __restrict__ tile_qstart,   // [ntiles] seq's first row in q/out
    const int* __restrict__ tile_q0,       // [ntiles] q-tile offset in suffix
    const int* __restrict__ tile_hist,     // [ntiles] cached tokens (history)
    const int* __restrict__ tile_new,      // [ntiles] suffix length
    const int* __restrict__ tile_seq,      // [ntiles] row into block_tables
    int Hq, int Hkv, int maxb, float scale, long qs,
    const float* __restrict__ sinks,       // [Hq] or null
    int window,                            // 0 = full causal
    float softcap) {                       // 0 = off (Gemma-2 tanh cap)
  constexpr int KU = BK * (D / 8) / PF_THREADS;
  constexpr int VU = D * (BK / 8) / PF_THREADS;
  constexpr int QK = D / 32;
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hkv);
  const int qstart = tile_qstart[tile];
  const int q0 = tile_q0[tile];
  const int hist = tile_hist[tile];
  const int nnew = tile_new[tile];
  const int len = hist + nnew;  // total context length
  const int* bt = block_tables + (long)tile_seq[tile] * maxb;

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lc = lane & 15;
  const int lg = lane >> 4;
  const int tid = threadIdx.x;

  __shared__ unsigned short Kl[BK * D];
  __shared__ unsigned short VTl[D][BK + VT_PAD];
  __shared__ unsigned short Pl[PF_WAVES][16][BK + VT_PAD];

  const int qrow_local = q0 + wave * 16 + lc;           // row within suffix
  const int qrow_clamped = (qrow_local < nnew) ? qrow_local : (nnew - 1);
  u16x8 qfrag[QK];
#pragma unroll
  for (int kk = 0; kk < QK; ++kk) {
    const unsigned short* qp = q + (long)(qstart + qrow_clamped) * qs +
                               (long)qh * D + kk * 32 + lg * 8;
    qfrag[kk] = *reinterpret_cast<const u16x8*>(qp);
  }

  float mcol = -INFINITY;
  float lcol = 0.f;
  f32x4 o[D / 16];
#pragma unroll
  for (int nt = 0; nt < D / 16; ++nt) o[nt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_hi_abs = hist + q0 + BQ - 1;              // absolute position
  const int kv_end = min(len, q_hi_abs + 1);
  const int ntiles_kv = (kv_end + BK - 1) / BK;
  const int wave_q_hi_abs = hist + q0 + wave * 16 + 15;
  const int wave_q_lo_abs = hist + q0 + wave * 16;

  int kst_kv[KU];
#pragma unroll
  for (int r = 0; r < KU; ++r) kst_kv[r] = (tid + r * PF_THREADS) / (D / 8);
  const int kst_d0 = (tid % (D / 8)) * 8;
  const int vst_d = tid % D;
  const int vst_kvc0 = (tid / D) * 8;
  constexpr int VST_STEP = 8 * PF_THREADS / D;

  u16x8 kstage[KU];
  unsigned short vstage[VU][8];

  // pool row base (elements) for token position `kv`
  auto pool_row = [&](int kv) {
    const int blk = bt[kv / PP_BS];
    return (((long)blk * Hkv + kvh) * PP_BS + kv % PP_BS) * D;
  };

  auto issue_loads = [&](int kv0) {
    // clamp BEFORE the block-table lookup: a speculated address compute
    // for masked-out lanes must not index past the table row
#pragma unroll
    for (int r = 0; r < KU; ++r) {
      const int kv = kv0 + kst_kv[r];
      const bool ok = kv < len;
      const u16x8 val = *reinterpret_cast<const u16x8*>(
          kc + pool_row(ok ? kv : len - 1) + kst_d0);
      kstage[r] = ok ? val : u16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
#pragma unroll
    for (int r = 0; r < VU; ++r) {
      const int kvc = vst_kvc0 + r * VST_STEP;
      // 8 consecutive positions span at most 2 pool blocks; resolving the
      // row base per element keeps the gather exact at block boundaries
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int kv = kv0 + kvc + j;
        const bool ok = kv < len;
        const unsigned short val = vc[pool_row(ok ? kv : len - 1) + vst_d];
        vstage[r][j] = ok ? val : (unsigned short)0;
      }
    }
  };

  auto write_lds = [&]() {
#pragma unroll
    for (int r = 0; r < KU; ++r) {
      *reinterpret_cast<u16x8*>(reinterpret_cast<char*>(Kl) +
                                kswz<D>(kst_kv[r], kst_d0 * 2)) = kstage[r];
    }
#pragma unroll
    for (int r = 0; r < VU; ++r) {
      *reinterpret_cast<u16x8*>(&VTl[vst_d][vst_kvc0 + r * VST_STEP]) =
          *reinterpret_cast<u16x8*>(vstage[r]);
    }
  };

  issue_loads(0);
  for (int kt = 0; kt < ntiles_kv; ++kt) {
    const int kv0 = kt * BK;
    __syncthreads();
    write_lds();
    __syncthreads();
    if (kt + 1 < ntiles_kv) issue_loads(kv0 + BK);

    if (kv0 > wave_q_hi_abs) continue;
    if (EXT && window > 0 && kv0 + BK <= wave_q_lo_abs - window + 1)
      continue;

    f32x4 st[NST];
#pragma unroll
    for (int i = 0; i < NST; ++i) st[i] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int kk = 0; kk < QK; ++kk) {
        const u16x8 a = *reinterpret_cast<const u16x8*>(
            reinterpret_cast<char*>(Kl) +
            kswz<D>(stile * 16 + lc, (kk * 32 + lg * 8) * 2));
        st[stile] = mfma16x16x32_bf16(a, qfrag[kk], st[stile]);
      }
    }

    const int qpos = hist + q0 + wave * 16 + lc;  // absolute position
    const bool qvalid = (q0 + wave * 16 + lc) < nnew;
    float sv[NST * 4];
    float tmax = -INFINITY;
#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvpos = kv0 + stile * 16 + lg * 4 + r;
        float x = st[stile][r] * scale;
        bool ok = (kvpos <= qpos) && (kvpos < len) && qvalid;
        if constexpr (EXT) {
          if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
          if (window > 0) ok = ok && (kvpos > qpos - window);
        }
        x = ok ? x : -INFINITY;
        sv[stile * 4 + r] = x;
        tmax = fmaxf(tmax, x);
      }
    }
#pragma unroll
    for (int msk = 16; msk <= 32; msk <<= 1)
      tmax = fmaxf(tmax, __shfl_xor(tmax, msk, WAVE_SIZE));

    const float nm = fmaxf(mcol, tmax);
    float corr = 1.f, tsum = 0.f;
    float pv[NST * 4];
    if (nm != -INFINITY) {
      corr = __expf(mcol - nm);
#pragma unroll
      for (int i = 0; i < NST * 4; ++i) {
        pv[i] = (sv[i] == -INFINITY) ? 0.f : __expf(sv[i] - nm);
        tsum += pv[i];
      }
      mcol = nm;
    } else {
#pragma unroll
      for (int i = 0; i < NST * 4; ++i) pv[i] = 0.f;
    }
#pragma unroll
    for (int msk = 16; msk <= 32; msk <<= 1)
      tsum += __shfl_xor(tsum, msk, WAVE_SIZE);
    lcol = lcol * corr + tsum;

#pragma unroll
    for (int stile = 0; stile < NST; ++stile) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        Pl[wave][lc][stile * 16 + lg * 4 + r] = f2bf(pv[stile * 4 + r]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = lg * 4 + r;
      const float c = __shfl(corr, orow, WAVE_SIZE);
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) o[nt][r] *= c;
    }

#pragma unroll
    for (int kk2 = 0; kk2 < KC2; ++kk2) {
      const u16x8 pa = *reinterpret_cast<const u16x8*>(
          &Pl[wave][lc][kk2 * 32 + lg * 8]);
#pragma unroll
      for (int nt = 0; nt < D / 16; ++nt) {
        const u16x8 b = *reinterpret_cast<const u16x8*>(
            &VTl[nt * 16 + lc][kk2 * 32 + lg * 8]);
        o[nt] = mfma16x16x32_bf16(pa, b, o[nt]);
      }
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = lg * 4 + r;
    float denom = __shfl(lcol, orow, WAVE_SIZE);
    float onum = 1.f;
    if constexpr (EXT) {
      if (sinks != nullptr) {
        const float m_row = __shfl(mcol, orow, WAVE_SIZE);
        const float sk = sinks[qh];
        const float M2 = fmaxf(m_row, sk);
        onum = (m_row == -INFINITY) ? 0.f : __expf(m_row - M2);
        denom = denom * onum + __expf(sk - M2);
      }
    }
    const int qrow = q0 + wave * 16 + orow;
    if (qrow >= nnew || denom <= 0.f) continue;
    const float inv = onum / denom;
#pragma unroll
    for (int nt = 0; nt < D / 16; ++nt) {
      out[((long)(qstart + qrow) * Hq + qh) * D + nt * 16 + lc] =
          f2bf(o[nt][r] * inv);
    }
  }
}

}  // namespace

void flash_prefill_paged_launch(
    void* out, const void* q, const void* kc, const void* vc,
    const int* block_tables, const int* tile_qstart, const int* tile_q0,
    const int* tile_hist, const int* tile_new, const int* tile_seq,
    int ntiles, int Hq, int Hkv, int D, int maxb, float scale, long qs,
    const float* sinks, int window, float softcap, int* err_unsupported,
    hipStream_t s) {
  *err_unsupported = 0;
  if ((D != 128 && D != 64 && D != 256) || Hq % Hkv != 0) {
    *err_unsupported = 1;
    return;
  }
  dim3 grid(ntiles, Hq);
  const bool ext = (sinks != nullptr) || window > 0 || softcap > 0.f;
#define PPG_LAUNCH(DD, E)                                                     \
  hipLaunchKernelGGL((flash_prefill_paged_kernel<DD, E>), grid,               \
                     dim3(PF_THREADS), 0, s, (unsigned short*)out,            \
                     (const unsigned short*)q, (const unsigned short*)kc,     \
                     (const unsigned short*)vc, block_tables, tile_qstart,    \
                     tile_q0, tile_hist, tile_new, tile_seq, Hq, Hkv, maxb,   \
                     scale, qs, sinks, window, softcap)
  if (D == 128) {
    if (ext) PPG_LAUNCH(128, true);
    else PPG_LAUNCH(128, false);
  } else if (D == 64) {
    PPG_LAUNCH(64, true);
  } else {
    PPG_LAUNCH(256, true);
  }
#undef PPG_LAUNCH
}

void flash_prefill_launch(void* out, const void* q, const void* k,
                          const void* v, const int* tile_start,
                          const int* tile_q0, const int* tile_len, int ntiles,
                          int Hq, int Hkv, int D, int DV, float scale,
                          long qs, long ks, long vs, const float* sinks,
                          int window, float softcap, int* err_unsupported,
                          hipStream_t s) {
  *err_unsupported = 0;
  const bool mla_dims = (D == 192 && DV == 128);
  if (((D != 128 && D != 64 && D != 256) && !mla_dims)
      || (!mla_dims && DV != D) || Hq % Hkv != 0) {
    *err_unsupported = 1;
    return;
  }
  dim3 grid(ntiles, Hq);
  const bool ext = (sinks != nullptr) || window > 0 || softcap > 0.f;
#define PF_LAUNCH(DD, E, DVV)                                                \
  hipLaunchKernelGGL((flash_prefill_kernel<DD, E, DVV>), grid,               \
                     dim3(PF_THREADS), 0, s, (unsigned short*)out,           \
                     (const unsigned short*)q, (const unsigned short*)k,     \
                     (const unsigned short*)v, tile_start, tile_q0,          \
                     tile_len, Hq, Hkv, scale, qs, ks, vs, sinks, window,    \
                     softcap)
  if (mla_dims) {
    // MLA expand-prefill (DeepSeek): 192-dim qk over 128-dim values
    PF_LAUNCH(192, false, 128);
  } else if (D == 128) {
    if (ext) PF_LAUNCH(128, true, 128);
    else PF_LAUNCH(128, false, 128);
  } else if (D == 64) {
    PF_LAUNCH(64, true, 64);
  } else {
    PF_LAUNCH(256, true, 256);
  }
#undef PF_LAUNCH
}

// ---------------------------------------------------------------------------
// MFMA layout probe: D = A[16x32] * B[32x16] with the fragment layouts this
// file assumes. The GPU unit test checks it against torch.matmul with random
// asymmetric inputs (guide G9: transpose-detecting correctness checks).
// ---------------------------------------------------------------------------
namespace {
__global__ void mfma_probe_kernel(float* __restrict__ d,
                                  const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int lc = lane & 15, lg = lane >> 4;
  u16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[lc * 32 + lg * 8 + j];   // A[row=lc][k=lg*8+j]
    bf[j] = b[(lg * 8 + j) * 16 + lc]; // B[k=lg*8+j][col=lc]
  }
  f32x4 c{0.f, 0.f, 0.f, 0.f};
  c = mfma16x16x32_bf16(af, bf, c);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(lg * 4 + r) * 16 + lc] = c[r];
}
}  // namespace

void mfma_probe_launch(float* d, const void* a, const void* b, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s, d,
                     (const unsigned short*)a, (const unsigned short*)b);
}
