// Paged-attention decode for MI355X (gfx950) — the TPOT-dominant kernel.
//
// Single new token per sequence attends over the paged KV cache.
// Replaces what the reference delegates to vLLM's paged_attention kernels
// (SURVEY.md §2.9 #1); designed for CDNA4 rather than ported:
//
//  - one workgroup per (sequence, kv_head); GQA q-heads (GQ = Hq/Hkv) share
//    the K/V stream so each KV byte is read exactly once per step.
//  - 4 waves * 4 lane-groups of 16 = 16 independent online-softmax streams;
//    a 16-lane group owns one token per iteration, lane = 8 head dims
//    (16 B = u16x8 loads -> 1 KiB per wave per iteration, fully coalesced
//    against the [block, kv_head, block_size, D] pool layout).
//  - streams merge via 64-wide shuffles (cross-group) then LDS (cross-wave);
//    accumulation entirely in f32.
//
// KV pool layout: [num_blocks, Hkv, BS, D] bf16, BS = 16 tokens.
#include "common.h"

#include <type_traits>

namespace {

constexpr int BS = 16;       // tokens per KV block (page)
constexpr int NWAVES = 4;    // waves per workgroup
constexpr int THREADS = NWAVES * WAVE_SIZE;

// EXT=false compiles the exact validated fast path (no sinks / window /
// softcap code at all — bit-identical to the r1/r2-measured kernel);
// EXT=true carries the GPT-OSS/Gemma feature variants and only ever
// launches when one of those features is requested.
template <int D, int GQ, bool FP8, bool EXT = false>
__global__ __launch_bounds__(THREADS) void paged_attn_decode_kernel(
    unsigned short* __restrict__ out,        // [N, Hq, D]
    const unsigned short* __restrict__ q,    // [N, Hq, D]
    const void* __restrict__ kc,             // [B, Hkv, BS, D] bf16|fp8
    const void* __restrict__ vc,             // [B, Hkv, BS, D]
    const int* __restrict__ block_tables,    // [N, max_blocks]
    const int* __restrict__ seq_lens,        // [N]
    int Hkv, int max_blocks, float scale, long q_stride,
    const float* __restrict__ sinks,         // [Hq] or null (GPT-OSS)
    int window,                              // 0 = full attention
    float softcap) {                         // 0 = off (Gemma-2 tanh cap)
  constexpr int LPG = 16;           // lanes per token-group
  constexpr int DV = D / LPG;       // dims per lane (4/8/16 for D 64/128/256)
  static_assert(DV == 4 || DV == 8 || DV == 16,
                "decode kernel assumes D in {64, 128, 256}");
  constexpr int NV = (DV >= 8) ? DV / 8 : 1;  // u16x8 vectors per lane slice
  // tokens batched per softmax update: trade VALU savings against VGPR
  // pressure (GQ>=4 would spill at TB=4)
  constexpr int TB = (GQ <= 2) ? (BS / 4) : ((GQ <= 5) ? 2 : 1);

  const int seq = blockIdx.x;
  const int h = blockIdx.y;        // kv head
  const int Hq = Hkv * GQ;
  const int len = seq_lens[seq];
  const int npages = (len + BS - 1) / BS;

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int g = lane >> 4;         // token-group within wave [0,4)
  const int sub = lane & 15;       // dim-slice within group [0,16)

  // Load q rows for this kv head's GQ query heads. bf16 path: keep q PACKED
  // (u16x8 = 4 VGPRs/head vs 8 for f32) and dot with the native
  // v_dot2c_f32_bf16 instruction — halves both q/k register pressure and
  // dot-product instruction count (occupancy was the bottleneck: 2-3
  // waves/SIMD at 168-200 VGPRs, profiles/r02_optimization_log.md).
  // fp8 path: f32 q pre-scaled as before (k dequants through f32 anyway).
  // lane slice = DV dims: one u16x4 (DV=4) or NV u16x8 vectors
  using U16V = std::conditional_t<DV == 4, u16x4, u16x8>;
  using U8V = std::conditional_t<DV == 4, u8x4, u8x8>;
  constexpr int VW = (DV == 4) ? 4 : 8;     // elements per vector
  float qr[FP8 ? GQ : 1][DV];
  U16V qb[FP8 ? 1 : GQ][NV];
#pragma unroll
  for (int gq = 0; gq < GQ; ++gq) {
    const unsigned short* qp =
        q + (long)seq * q_stride + ((long)h * GQ + gq) * D + sub * DV;
#pragma unroll
    for (int vv = 0; vv < NV; ++vv) {
      U16V u = *reinterpret_cast<const U16V*>(qp + vv * VW);
      if constexpr (FP8) {
#pragma unroll
        for (int j = 0; j < VW; ++j)
          qr[gq][vv * VW + j] = bf2f(u[j]) * scale;
      } else {
        qb[gq][vv] = u;
      }
    }
  }

  float m[GQ], s[GQ], acc[GQ][DV];
#pragma unroll
  for (int gq = 0; gq < GQ; ++gq) {
    m[gq] = -INFINITY;
    s[gq] = 0.f;
#pragma unroll
    for (int j = 0; j < DV; ++j) acc[gq][j] = 0.f;
  }

  using KVT = std::conditional_t<FP8, unsigned char, unsigned short>;
  // sliding window (EXT): the single decode query sits at position len-1
  // and attends [max(0, len-window), len)
  const int wstart =
      (EXT && window > 0 && len > window) ? len - window : 0;
  const int page0 = EXT ? wstart / BS : 0;
  const int* bt = block_tables + (long)seq * max_blocks;
  for (int page = page0 + wave; page < npages; page += NWAVES) {
    const long blk = bt[page];
    const KVT* kbase = (const KVT*)kc + ((blk * Hkv + h) * BS) * D;
    const KVT* vbase = (const KVT*)vc + ((blk * Hkv + h) * BS) * D;
    // Whole page per wave-iteration: the group's 4 tokens are processed
    // together — 4 independent dots (ILP across the shuffle-reduce
    // chains), then ONE softmax update per gq for all 4 (cuts the
    // exp/rescale VALU work ~2.5x vs per-token online updates).
#pragma unroll
    for (int tb = 0; tb < BS / 4; tb += TB) {
      const int base_tok = page * BS + tb * 4 + g;
      using VecT = std::conditional_t<FP8, U8V, U16V>;
      VecT vu[TB][NV];
      VecT ku[TB][NV];          // bf16: K stays packed (dot2 consumes it)
      float kf[FP8 ? TB : 1][DV];  // fp8: K dequanted once, reused per gq
#pragma unroll
      for (int it = 0; it < TB; ++it) {
        const int tok = (tb + it) * 4 + g;
#pragma unroll
        for (int vv = 0; vv < NV; ++vv) {
          ku[it][vv] = *reinterpret_cast<const VecT*>(
              kbase + tok * D + sub * DV + vv * VW);
          vu[it][vv] = *reinterpret_cast<const VecT*>(
              vbase + tok * D + sub * DV + vv * VW);
          if constexpr (FP8) {
#pragma unroll
            for (int j = 0; j < VW; ++j)
              kf[it][vv * VW + j] = fp8_to_f32(ku[it][vv][j]);
          }
        }
      }
#pragma unroll
      for (int gq = 0; gq < GQ; ++gq) {
        float dot[TB];
#pragma unroll
        for (int it = 0; it < TB; ++it) {
          float d = 0.f;
          if constexpr (FP8) {
#pragma unroll
            for (int j = 0; j < DV; ++j) d += qr[gq][j] * kf[it][j];
          } else {
            const bf16x2* qa = reinterpret_cast<const bf16x2*>(&qb[gq][0]);
            const bf16x2* ka = reinterpret_cast<const bf16x2*>(&ku[it][0]);
#pragma unroll
            for (int j = 0; j < DV / 2; ++j)
              d = __builtin_amdgcn_fdot2_f32_bf16(qa[j], ka[j], d, false);
            d *= scale;  // q not pre-scaled on this path (kept packed)
          }
          dot[it] = d;
        }
#pragma unroll
        for (int it = 0; it < TB; ++it)
          dot[it] = group16_reduce_sum(dot[it]);  // all 16 lanes get the dot
        float pmax = -INFINITY;
#pragma unroll
        for (int it = 0; it < TB; ++it) {
          if constexpr (EXT) {
            const int tpos = base_tok + it * 4;
            const bool valid = tpos < len && tpos >= wstart;
            float dv = dot[it];
            if (softcap > 0.f) dv = tanhf(dv / softcap) * softcap;
            dot[it] = valid ? dv : -INFINITY;
          } else {
            const bool valid = base_tok + it * 4 < len;
            dot[it] = valid ? dot[it] : -INFINITY;
          }
          pmax = fmaxf(pmax, dot[it]);
        }
        if (pmax == -INFINITY) continue;
        const float nm = fmaxf(m[gq], pmax);
        const float corr = __expf(m[gq] - nm);  // exp(-inf - finite) = 0
        float p[TB];
        float psum = 0.f;
#pragma unroll
        for (int it = 0; it < TB; ++it) {
          p[it] = (dot[it] == -INFINITY) ? 0.f : __expf(dot[it] - nm);
          psum += p[it];
        }
        s[gq] = s[gq] * corr + psum;
#pragma unroll
        for (int j = 0; j < DV; ++j) {
          float a = acc[gq][j] * corr;
#pragma unroll
          for (int it = 0; it < TB; ++it)
            a += p[it] * (FP8 ? fp8_to_f32(vu[it][j / VW][j % VW])
                              : bf2f(vu[it][j / VW][j % VW]));
          acc[gq][j] = a;
        }
        m[gq] = nm;
      }
    }
  }

  // Merge the 4 token-group streams within each wave (butterfly over
  // lane masks 16 and 32). Lanes with equal `sub` end up identical.
#pragma unroll
  for (int mask = 16; mask <= 32; mask <<= 1) {
#pragma unroll
    for (int gq = 0; gq < GQ; ++gq) {
      const float om = __shfl_xor(m[gq], mask, WAVE_SIZE);
      const float os = __shfl_xor(s[gq], mask, WAVE_SIZE);
      float oacc[DV];
#pragma unroll
      for (int j = 0; j < DV; ++j)
        oacc[j] = __shfl_xor(acc[gq][j], mask, WAVE_SIZE);
      const float nm = fmaxf(m[gq], om);
      if (os > 0.f || s[gq] > 0.f) {
        const float w1 = (s[gq] > 0.f) ? __expf(m[gq] - nm) : 0.f;
        const float w2 = (os > 0.f) ? __expf(om - nm) : 0.f;
        s[gq] = s[gq] * w1 + os * w2;
#pragma unroll
        for (int j = 0; j < DV; ++j)
          acc[gq][j] = acc[gq][j] * w1 + oacc[j] * w2;
        m[gq] = nm;
      }
    }
  }

  // Cross-wave merge through LDS.
  __shared__ float lds_acc[NWAVES][GQ][D];
  __shared__ float lds_m[NWAVES][GQ];
  __shared__ float lds_s[NWAVES][GQ];
  if (g == 0) {  // one representative lane-group per wave
#pragma unroll
    for (int gq = 0; gq < GQ; ++gq) {
#pragma unroll
      for (int j = 0; j < DV; ++j) lds_acc[wave][gq][sub * DV + j] = acc[gq][j];
      if (sub == 0) {
        lds_m[wave][gq] = m[gq];
        lds_s[wave][gq] = s[gq];
      }
    }
  }
  __syncthreads();

  // 256 threads cover (gq, d) output elements.
  for (int u = threadIdx.x; u < GQ * D; u += THREADS) {
    const int gq = u / D;
    const int d = u % D;
    float M = -INFINITY;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w)
      if (lds_s[w][gq] > 0.f) M = fmaxf(M, lds_m[w][gq]);
    // GPT-OSS attention sink (EXT): a per-head learned logit joins the
    // softmax DENOMINATOR only (no value contribution)
    float sden = 0.f;
    if constexpr (EXT) {
      if (sinks != nullptr) {
        const float sk = sinks[h * GQ + gq];
        M = fmaxf(M, sk);
        sden = __expf(sk - M);
      }
    }
    float num = 0.f, den = sden;
#pragma unroll
    for (int w = 0; w < NWAVES; ++w) {
      if (lds_s[w][gq] > 0.f) {
        const float wt = __expf(lds_m[w][gq] - M);
        num += wt * lds_acc[w][gq][d];
        den += wt * lds_s[w][gq];
      }
    }
    out[((long)seq * Hq + (long)h * GQ + gq) * D + d] =
        f2bf(den > 0.f ? num / den : 0.f);
  }
}

}  // namespace

void paged_attn_decode_launch(void* out, const void* q, const void* kc,
                              const void* vc, const int* block_tables,
                              const int* seq_lens, int N, int Hq, int Hkv,
                              int D, int max_blocks, float scale, long q_stride,
                              int fp8, const float* sinks, int window,
                              float softcap, int* err_unsupported,
                              hipStream_t s) {
  const int GQ = Hq / Hkv;
  dim3 grid(N, Hkv);
  dim3 block(THREADS);
  *err_unsupported = 0;
  if (D != 128 && D != 64 && D != 256) { *err_unsupported = 1; return; }
  if (D == 256 && GQ > 4) { *err_unsupported = 1; return; }  // VGPR budget
  const bool ext = (sinks != nullptr) || window > 0 || softcap > 0.f;
#define LAUNCH_D(DD, G, F, E)                                                  \
  hipLaunchKernelGGL((paged_attn_decode_kernel<DD, G, F, E>), grid, block, 0,  \
                     s, (unsigned short*)out, (const unsigned short*)q, kc,    \
                     vc, block_tables, seq_lens, Hkv, max_blocks, scale,       \
                     q_stride, sinks, window, softcap)
#define LAUNCH_GQ2(G, F)                                                       \
  do {                                                                         \
    if (D == 128) {                                                            \
      if (ext) LAUNCH_D(128, G, F, true);                                      \
      else LAUNCH_D(128, G, F, false);                                         \
    } else if (D == 64) {                                                      \
      LAUNCH_D(64, G, F, true);                                                \
    } else if ((G) <= 4) {                                                     \
      LAUNCH_D(256, (G) <= 4 ? (G) : 1, F, true);                              \
    }                                                                          \
  } while (0)
#define LAUNCH_GQ(G)                                                          \
  do {                                                                        \
    if (fp8) LAUNCH_GQ2(G, true);                                             \
    else LAUNCH_GQ2(G, false);                                                \
  } while (0)
  switch (GQ) {
    case 1: LAUNCH_GQ(1); break;
    case 2: LAUNCH_GQ(2); break;
    case 4: LAUNCH_GQ(4); break;
    case 5: LAUNCH_GQ(5); break;
    case 6: LAUNCH_GQ(6); break;
    case 7: LAUNCH_GQ(7); break;
    case 8: LAUNCH_GQ(8); break;
    default: *err_unsupported = 1; return;
  }
#undef LAUNCH_GQ
#undef LAUNCH_GQ2
#undef LAUNCH_D
}
