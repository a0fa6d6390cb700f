// Pointwise / memory-bound serving kernels for MI355X (gfx950).
//
// Replaces the elementwise+norm ops the reference's external engines supply
// (survey: SURVEY.md §2.9 #1 — engine kernels: RMSNorm, RoPE, activation,
// KV-cache scatter, sampling). All kernels are HBM-bound: bf16 IO is
// vectorized as u16x8 (16 B/lane), accumulation in f32, grid-stride loops
// capped so the scheduler keeps ~8 blocks/CU in flight.
#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: out[t, :] = x[t, :] / rms(x[t, :]) * w   (row-parallel, 1 block/row)
// H <= 8192 keeps the row in registers (256 thr * 4 * u16x8); larger H
// re-reads through L2.
// ---------------------------------------------------------------------------
namespace {

constexpr int NORM_THREADS = 256;
constexpr int NORM_MAX_VEC = 4; // u16x8 per thread held in registers

__global__ void rms_norm_kernel(unsigned short* __restrict__ out,
                                const unsigned short* __restrict__ in,
                                const unsigned short* __restrict__ w,
                                float eps, int T, int H) {
  const int nvec = H / 8; // H % 8 == 0 enforced on host
  const bool cached = nvec <= NORM_THREADS * NORM_MAX_VEC;
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    const unsigned short* rp = in + (long)row * H;
    float vals[NORM_MAX_VEC][8];
    float ss = 0.f;
    for (int v = threadIdx.x, slot = 0; v < nvec; v += NORM_THREADS, ++slot) {
      u16x8 u = *reinterpret_cast<const u16x8*>(rp + v * 8);
      float f[8];
      bf8_to_f32(u, f);
#pragma unroll
      for (int i = 0; i < 8; ++i) ss += f[i] * f[i];
      if (cached && slot < NORM_MAX_VEC) {
#pragma unroll
        for (int i = 0; i < 8; ++i) vals[slot][i] = f[i];
      }
    }
    // block reduce: wave reduce + LDS
    __shared__ float red[NORM_THREADS / WAVE_SIZE];
    ss = wave_reduce_sum(ss);
    if ((threadIdx.x & (WAVE_SIZE - 1)) == 0) red[threadIdx.x / WAVE_SIZE] = ss;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < NORM_THREADS / WAVE_SIZE; ++i) tot += red[i];
    const float inv = rsqrtf(tot / H + eps);
    unsigned short* op = out + (long)row * H;
    for (int v = threadIdx.x, slot = 0; v < nvec; v += NORM_THREADS, ++slot) {
      float f[8];
      if (cached && slot < NORM_MAX_VEC) {
#pragma unroll
        for (int i = 0; i < 8; ++i) f[i] = vals[slot][i];
      } else {
        u16x8 u = *reinterpret_cast<const u16x8*>(rp + v * 8);
        bf8_to_f32(u, f);
      }
      u16x8 wv = *reinterpret_cast<const u16x8*>(w + v * 8);
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = f[i] * inv * bf2f(wv[i]);
      *reinterpret_cast<u16x8*>(op + v * 8) = f32_to_bf8(o);
    }
    __syncthreads();
  }
}

// residual += x ; x = rmsnorm(residual) * w      (both updated in place)
__global__ void fused_add_rms_norm_kernel(unsigned short* __restrict__ x,
                                          unsigned short* __restrict__ residual,
                                          const unsigned short* __restrict__ w,
                                          float eps, int T, int H) {
  const int nvec = H / 8;
  const bool cached = nvec <= NORM_THREADS * NORM_MAX_VEC;
  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    unsigned short* xp = x + (long)row * H;
    unsigned short* rp = residual + (long)row * H;
    float vals[NORM_MAX_VEC][8];
    float ss = 0.f;
    for (int v = threadIdx.x, slot = 0; v < nvec; v += NORM_THREADS, ++slot) {
      u16x8 ux = *reinterpret_cast<const u16x8*>(xp + v * 8);
      u16x8 ur = *reinterpret_cast<const u16x8*>(rp + v * 8);
      float f[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        f[i] = bf2f(ux[i]) + bf2f(ur[i]);
        ss += f[i] * f[i];
      }
      *reinterpret_cast<u16x8*>(rp + v * 8) = f32_to_bf8(f);
      if (cached && slot < NORM_MAX_VEC) {
#pragma unroll
        for (int i = 0; i < 8; ++i) vals[slot][i] = f[i];
      }
    }
    __shared__ float red[NORM_THREADS / WAVE_SIZE];
    ss = wave_reduce_sum(ss);
    if ((threadIdx.x & (WAVE_SIZE - 1)) == 0) red[threadIdx.x / WAVE_SIZE] = ss;
    __syncthreads();
    float tot = 0.f;
#pragma unroll
    for (int i = 0; i < NORM_THREADS / WAVE_SIZE; ++i) tot += red[i];
    const float inv = rsqrtf(tot / H + eps);
    for (int v = threadIdx.x, slot = 0; v < nvec; v += NORM_THREADS, ++slot) {
      float f[8];
      if (cached && slot < NORM_MAX_VEC) {
#pragma unroll
        for (int i = 0; i < 8; ++i) f[i] = vals[slot][i];
      } else {
        // residual now holds the sum (bf16-rounded); re-read it.
        u16x8 u = *reinterpret_cast<const u16x8*>(rp + v * 8);
        bf8_to_f32(u, f);
      }
      u16x8 wv = *reinterpret_cast<const u16x8*>(w + v * 8);
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = f[i] * inv * bf2f(wv[i]);
      *reinterpret_cast<u16x8*>(xp + v * 8) = f32_to_bf8(o);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Rotary embedding (GPT-NeoX style halves), in-place on q and k.
// cos_sin: [max_pos, rot_dim] f32, first rot_dim/2 cos then rot_dim/2 sin,
// precomputed on host (Appendix B: no on-device trig).
// Work unit: one (token, head, 8-dim chunk of the first half).
// ---------------------------------------------------------------------------
__global__ void rope_neox_kernel(const long* __restrict__ positions,
                                 unsigned short* __restrict__ q,
                                 unsigned short* __restrict__ k,
                                 const float* __restrict__ cos_sin,
                                 int T, int Hq, int Hk, int D, int R,
                                 long qs, long ks) {  // row strides (elements)
  const int half = R / 2;
  const int chunks_per_head = half / 8; // R/2 % 8 == 0 enforced on host
  const long total = (long)T * (Hq + Hk) * chunks_per_head;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c = idx % chunks_per_head;
    const int h = (idx / chunks_per_head) % (Hq + Hk);
    const int t = idx / ((long)chunks_per_head * (Hq + Hk));
    const long pos = positions[t];
    const float* cs = cos_sin + pos * R + c * 8;
    unsigned short* base =
        (h < Hq) ? q + (long)t * qs + (long)h * D
                 : k + (long)t * ks + (long)(h - Hq) * D;
    u16x8 u1 = *reinterpret_cast<const u16x8*>(base + c * 8);
    u16x8 u2 = *reinterpret_cast<const u16x8*>(base + half + c * 8);
    float o1[8], o2[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float x1 = bf2f(u1[i]), x2 = bf2f(u2[i]);
      const float co = cs[i], si = cs[half + i];
      o1[i] = x1 * co - x2 * si;
      o2[i] = x2 * co + x1 * si;
    }
    *reinterpret_cast<u16x8*>(base + c * 8) = f32_to_bf8(o1);
    *reinterpret_cast<u16x8*>(base + half + c * 8) = f32_to_bf8(o2);
  }
}

// ---------------------------------------------------------------------------
// SiLU-and-mul: out[t, i] = silu(x[t, i]) * x[t, I + i]
// ---------------------------------------------------------------------------
__global__ void silu_and_mul_kernel(unsigned short* __restrict__ out,
                                    const unsigned short* __restrict__ x,
                                    long T, int I) {
  const long total = T * (I / 8);
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long t = idx / (I / 8);
    const int c = idx % (I / 8);
    const unsigned short* g = x + t * 2 * I + c * 8;
    const unsigned short* u = x + t * 2 * I + I + c * 8;
    u16x8 gv = *reinterpret_cast<const u16x8*>(g);
    u16x8 uv = *reinterpret_cast<const u16x8*>(u);
    float o[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float a = bf2f(gv[i]);
      o[i] = a / (1.f + __expf(-a)) * bf2f(uv[i]);
    }
    *reinterpret_cast<u16x8*>(out + t * I + c * 8) = f32_to_bf8(o);
  }
}

// ---------------------------------------------------------------------------
// reshape_and_cache: scatter fresh per-token K/V into the paged pool.
// k/v: [T, Hkv, D] bf16 ; caches: [num_blocks, Hkv, BS, D] ; slot[t] = global
// slot index (block*BS + offset), -1 = skip.
// ---------------------------------------------------------------------------
__global__ void reshape_and_cache_fp8_kernel(const unsigned short* __restrict__ k,
                                             const unsigned short* __restrict__ v,
                                             unsigned char* __restrict__ kc,
                                             unsigned char* __restrict__ vc,
                                             const long* __restrict__ slots,
                                             int T, int Hkv, int D, int BS,
                                             long ks, long vs) {
  const int chunks = D / 8;
  const long total = (long)T * Hkv * chunks;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c = idx % chunks;
    const int h = (idx / chunks) % Hkv;
    const int t = idx / ((long)chunks * Hkv);
    const long slot = slots[t];
    if (slot < 0) continue;
    const long blk = slot / BS, off = slot % BS;
    const long dst = ((blk * Hkv + h) * BS + off) * D + c * 8;
    const long hoff = (long)h * D + c * 8;
    u16x8 kv8 = *reinterpret_cast<const u16x8*>(k + (long)t * ks + hoff);
    u16x8 vv8 = *reinterpret_cast<const u16x8*>(v + (long)t * vs + hoff);
    u8x8 ko, vo;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      ko[i] = f32_to_fp8(bf2f(kv8[i]));
      vo[i] = f32_to_fp8(bf2f(vv8[i]));
    }
    *reinterpret_cast<u8x8*>(kc + dst) = ko;
    *reinterpret_cast<u8x8*>(vc + dst) = vo;
  }
}

__global__ void reshape_and_cache_kernel(const unsigned short* __restrict__ k,
                                         const unsigned short* __restrict__ v,
                                         unsigned short* __restrict__ kc,
                                         unsigned short* __restrict__ vc,
                                         const long* __restrict__ slots,
                                         int T, int Hkv, int D, int BS,
                                         long ks, long vs) {
  const int chunks = D / 8;
  const long total = (long)T * Hkv * chunks;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int c = idx % chunks;
    const int h = (idx / chunks) % Hkv;
    const int t = idx / ((long)chunks * Hkv);
    const long slot = slots[t];
    if (slot < 0) continue;
    const long blk = slot / BS, off = slot % BS;
    const long dst = ((blk * Hkv + h) * BS + off) * D + c * 8;
    const long hoff = (long)h * D + c * 8;
    *reinterpret_cast<u16x8*>(kc + dst) =
        *reinterpret_cast<const u16x8*>(k + (long)t * ks + hoff);
    *reinterpret_cast<u16x8*>(vc + dst) =
        *reinterpret_cast<const u16x8*>(v + (long)t * vs + hoff);
  }
}

// ---------------------------------------------------------------------------
// Greedy sampling: argmax over vocab per row (ties -> lowest index).
// One block per row; strided scan with (val, idx) carried per thread.
// ---------------------------------------------------------------------------
__global__ void greedy_sample_kernel(long* __restrict__ out,
                                     const unsigned short* __restrict__ logits,
                                     int N, int V) {
  const int row = blockIdx.x;
  if (row >= N) return;
  const unsigned short* rp = logits + (long)row * V;
  float best = -INFINITY;
  int besti = 0x7fffffff;
  const int nvec = V / 8;
  for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
    u16x8 u = *reinterpret_cast<const u16x8*>(rp + v * 8);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float f = bf2f(u[i]);
      const int gi = v * 8 + i;
      if (f > best || (f == best && gi < besti)) { best = f; besti = gi; }
    }
  }
  for (int gi = nvec * 8 + threadIdx.x; gi < V; gi += blockDim.x) {
    const float f = bf2f(rp[gi]);
    if (f > best || (f == best && gi < besti)) { best = f; besti = gi; }
  }
  // wave reduce, then LDS reduce
#pragma unroll
  for (int m = 32; m >= 1; m >>= 1) {
    const float ov = __shfl_xor(best, m, WAVE_SIZE);
    const int oi = __shfl_xor(besti, m, WAVE_SIZE);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float sv[8];
  __shared__ int si[8];
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & (WAVE_SIZE - 1)) == 0) { sv[wid] = best; si[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int i = 1; i < (int)(blockDim.x / WAVE_SIZE); ++i) {
      if (sv[i] > best || (sv[i] == best && si[i] < besti)) {
        best = sv[i]; besti = si[i];
      }
    }
    out[row] = besti;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Launchers (called from bindings.cpp)
// ---------------------------------------------------------------------------
static inline int pw_grid(long total_threads_needed, int block) {
  long blocks = (total_threads_needed + block - 1) / block;
  if (blocks > 2048) blocks = 2048; // ~8 blocks/CU, grid-stride the rest
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

void rms_norm_launch(void* out, const void* in, const void* w, float eps,
                     int T, int H, hipStream_t s) {
  int grid = T < 8192 ? (T < 1 ? 1 : T) : 8192;
  hipLaunchKernelGGL(rms_norm_kernel, dim3(grid), dim3(NORM_THREADS), 0, s,
                     (unsigned short*)out, (const unsigned short*)in,
                     (const unsigned short*)w, eps, T, H);
}

void fused_add_rms_norm_launch(void* x, void* residual, const void* w,
                               float eps, int T, int H, hipStream_t s) {
  int grid = T < 8192 ? (T < 1 ? 1 : T) : 8192;
  hipLaunchKernelGGL(fused_add_rms_norm_kernel, dim3(grid), dim3(NORM_THREADS),
                     0, s, (unsigned short*)x, (unsigned short*)residual,
                     (const unsigned short*)w, eps, T, H);
}

void rope_neox_launch(const long* positions, void* q, void* k,
                      const float* cos_sin, int T, int Hq, int Hk, int D,
                      int R, long qs, long ks, hipStream_t s) {
  long total = (long)T * (Hq + Hk) * (R / 2 / 8);
  hipLaunchKernelGGL(rope_neox_kernel, dim3(pw_grid(total, 256)), dim3(256), 0,
                     s, positions, (unsigned short*)q, (unsigned short*)k,
                     cos_sin, T, Hq, Hk, D, R, qs, ks);
}

void silu_and_mul_launch(void* out, const void* x, long T, int I,
                         hipStream_t s) {
  long total = T * (I / 8);
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(pw_grid(total, 256)), dim3(256),
                     0, s, (unsigned short*)out, (const unsigned short*)x, T, I);
}

void reshape_and_cache_launch(const void* k, const void* v, void* kc, void* vc,
                              const long* slots, int T, int Hkv, int D, int BS,
                              long ks, long vs, int fp8, hipStream_t s) {
  long total = (long)T * Hkv * (D / 8);
  if (fp8) {
    hipLaunchKernelGGL(reshape_and_cache_fp8_kernel, dim3(pw_grid(total, 256)),
                       dim3(256), 0, s, (const unsigned short*)k,
                       (const unsigned short*)v, (unsigned char*)kc,
                       (unsigned char*)vc, slots, T, Hkv, D, BS, ks, vs);
  } else {
    hipLaunchKernelGGL(reshape_and_cache_kernel, dim3(pw_grid(total, 256)),
                       dim3(256), 0, s, (const unsigned short*)k,
                       (const unsigned short*)v, (unsigned short*)kc,
                       (unsigned short*)vc, slots, T, Hkv, D, BS, ks, vs);
  }
}

void greedy_sample_launch(long* out, const void* logits, int N, int V,
                          hipStream_t s) {
  hipLaunchKernelGGL(greedy_sample_kernel, dim3(N), dim3(256), 0, s, out,
                     (const unsigned short*)logits, N, V);
}
