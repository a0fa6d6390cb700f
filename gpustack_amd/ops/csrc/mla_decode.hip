// MLA (DeepSeek) absorbed-attention decode over the paged LATENT cache —
// MI355X (gfx950).
//
// The paged pool stores one compressed row per token:
//   lat[l] = [ c_kv (R=512, rmsnormed) ; k_rope (DR=64, rotated) ]
// and the absorbed formulation makes decode a single-dot score:
//   score(h, l) = q_cat[h] · lat[l]        (q_cat = [q_nope W_uk ; q_rope])
//   ctx(h)      = sum_l softmax(l) * lat[l][0:R]
// The per-head W_uk / W_uv contractions are plain batched GEMMs and run
// in hipBLASLt before/after this kernel (models/llama.py MLAAttention);
// the kernel owns what a library cannot fuse: the paged gather + online
// softmax + context accumulation, with the latent stream read ONCE per
// 16-head block (vs once per head naively — DeepSeek V3 has 128 heads).
//
// Layout: one workgroup per (sequence, 16-head block); 4 waves x 4
// sixteen-lane groups = 16 online-softmax streams (one head each). Pages
// iterate sequentially, staged through LDS ([BS=16, LD=576] bf16, 18 KB)
// so all 16 heads consume each token row from LDS, not HBM. Per (head,
// token): 16 lanes split the 576-dot (36 dims each, v_dot2c bf16) and
// butterfly-reduce; the 512-dim context accumulator lives striped across
// the group's lanes (32 f32 each).
//
// Correctness-first r2 groundwork (compile-checked; GPU-validated and
// perf-tuned in r3) — gated behind GPUSTACK_AMD_MLA_KERNEL=1.
#include "common.h"

namespace {

constexpr int ML_BS = 16;        // tokens per KV block (page)
constexpr int ML_R = 512;        // kv_lora_rank
constexpr int ML_DR = 64;        // qk_rope_head_dim
constexpr int ML_LD = ML_R + ML_DR;          // latent row width (576)
constexpr int ML_WAVES = 4;
constexpr int ML_THREADS = ML_WAVES * WAVE_SIZE;
constexpr int ML_HB = 16;        // heads per workgroup (4 waves x 4 groups)
constexpr int ML_DPL = ML_LD / 16;           // score dims per lane (36)
constexpr int ML_CPL = ML_R / 16;            // ctx dims per lane (32)

__global__ __launch_bounds__(ML_THREADS) void mla_decode_kernel(
    float* __restrict__ ctx_out,             // [N, H, R] f32
    const unsigned short* __restrict__ q,    // [N, H, LD] bf16 (q_cat)
    const unsigned short* __restrict__ lat,  // [B, 1, BS, LD] bf16 pool
    const int* __restrict__ block_tables,    // [N, max_blocks]
    const int* __restrict__ seq_lens,        // [N]
    int H, int max_blocks, float scale) {
  const int seq = blockIdx.x;
  const int h0 = blockIdx.y * ML_HB;         // first head of this block
  const int len = seq_lens[seq];
  const int npages = (len + ML_BS - 1) / ML_BS;

  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int g = lane >> 4;                   // group in wave [0,4)
  const int sub = lane & 15;                 // lane in group [0,16)
  const int head = h0 + wave * 4 + g;
  const int tid = threadIdx.x;

  __shared__ unsigned short Ll[ML_BS * ML_LD];   // one page of latent rows

  // this lane's slice of the head's q_cat (36 bf16, 4-byte aligned)
  const unsigned short* qp =
      q + ((long)seq * H + head) * ML_LD + sub * ML_DPL;
  bf16x2 qv[ML_DPL / 2];
#pragma unroll
  for (int j = 0; j < ML_DPL / 2; ++j)
    qv[j] = *reinterpret_cast<const bf16x2*>(qp + 2 * j);

  float m = -INFINITY, s = 0.f;
  float acc[ML_CPL];
#pragma unroll
  for (int j = 0; j < ML_CPL; ++j) acc[j] = 0.f;

  const int* bt = block_tables + (long)seq * max_blocks;
  for (int page = 0; page < npages; ++page) {
    const long blk = bt[page];
    const unsigned short* src = lat + blk * (ML_BS * ML_LD);
    __syncthreads();  // previous page's LDS reads complete
    // cooperative stage: 16*576 u16 = 2304 u16x4 units / 256 threads = 9
    for (int u = tid; u < ML_BS * ML_LD / 4; u += ML_THREADS) {
      *reinterpret_cast<u16x4*>(Ll + 4 * u) =
          *reinterpret_cast<const u16x4*>(src + 4 * u);
    }
    __syncthreads();

    const int ntok = min(ML_BS, len - page * ML_BS);
    for (int t = 0; t < ntok; ++t) {
      const unsigned short* row = Ll + t * ML_LD;
      // 576-dot split 36 dims/lane, v_dot2c bf16 pairs
      float d = 0.f;
      const bf16x2* ra =
          reinterpret_cast<const bf16x2*>(row + sub * ML_DPL);
#pragma unroll
      for (int j = 0; j < ML_DPL / 2; ++j)
        d = __builtin_amdgcn_fdot2_f32_bf16(qv[j], ra[j], d, false);
      d = group16_reduce_sum(d) * scale;     // all 16 lanes hold the score
      const float nm = fmaxf(m, d);
      const float corr = __expf(m - nm);     // exp(-inf - x) = 0
      const float p = __expf(d - nm);
      s = s * corr + p;
      // ctx += p * c_kv[t]; lane owns dims [sub*32, sub*32+32) of R
      const unsigned short* cv = row + sub * ML_CPL;
#pragma unroll
      for (int j = 0; j < ML_CPL; ++j)
        acc[j] = acc[j] * corr + p * bf2f(cv[j]);
      m = nm;
    }
  }

  // normalize and write this lane's slice of ctx[seq, head]
  const float inv = (s > 0.f) ? 1.f / s : 0.f;
  float* dst = ctx_out + ((long)seq * H + head) * ML_R + sub * ML_CPL;
#pragma unroll
  for (int j = 0; j < ML_CPL; ++j) dst[j] = acc[j] * inv;
}

}  // namespace

void mla_decode_launch(float* ctx_out, const void* q, const void* lat,
                       const int* block_tables, const int* seq_lens, int N,
                       int H, int R, int DR, int BS, int max_blocks,
                       float scale, int* err_unsupported, hipStream_t s) {
  *err_unsupported = 0;
  if (R != ML_R || DR != ML_DR || BS != ML_BS || H % ML_HB != 0) {
    *err_unsupported = 1;
    return;
  }
  dim3 grid(N, H / ML_HB);
  hipLaunchKernelGGL(mla_decode_kernel, grid, dim3(ML_THREADS), 0, s,
                     ctx_out, (const unsigned short*)q,
                     (const unsigned short*)lat, block_tables, seq_lens, H,
                     max_blocks, scale);
}
