// W4A16 GEMM for MI355X (gfx950): out[M,N] = x[M,K] @ W^T with W kept
// PACKED int4 in HBM and dequantized in-register ahead of the MFMAs.
//
// Round-2 item: GPTQ/AWQ/Q4 checkpoints previously
// dequantized to bf16 at load (4x HBM inflation); this kernel halves the
// weight traffic of decode-shaped GEMMs (M<=1k is weight-stream bound:
// AI = M flop/byte) and keeps 70B-class models resident packed.
//
// Packed layout (models/quantized.py pack_w4_runtime): qw u8 [N, K/2] in
// FRAGMENT ORDER — within each 128-k block, the 16 bytes lane lg loads
// contain exactly the 32 nibbles (4 MFMA k-chunks x 8 values) that form
// its B-fragments, so no cross-lane shuffles are needed:
//   nibble i (0..31) of lane lg's 16B = q[col][blk*128 + (i/8)*32 + lg*8 + i%8]
// Scales/zero-scales: bf16 [N, K/128], one group per 128-k block
// (checkpoint group sizes that divide into 128-blocks are expanded at
// load). Dequant: w = q * s - zs   (zs = zero * s, precomputed).
//
// Tiling: BM=256 rows (16 m-frags) x BN=64 cols (4 waves x 16), K step
// 128. Weights stream ONCE per 256-row M pass; X tiles stage through LDS
// and re-reads across N-workgroups hit L2 (XCD-aware remap).
#include "common.h"

namespace {

constexpr int W4_BM = 256;
constexpr int W4_BN = 64;
constexpr int W4_KC = 128;
constexpr int W4_THREADS = 256;  // 4 waves
constexpr int W4_PAD = 8;        // elements; breaks the LDS bank cycle

typedef __attribute__((ext_vector_type(8))) short w4_s16x8;

DEVICE_INLINE f32x4 w4_mfma(u16x8 a, u16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      __builtin_bit_cast(w4_s16x8, a), __builtin_bit_cast(w4_s16x8, b), c, 0, 0, 0);
}

__global__ __launch_bounds__(W4_THREADS) void w4_gemm_kernel(
    unsigned short* __restrict__ out,     // [M, N] bf16
    const unsigned short* __restrict__ x, // [M, K] bf16
    const unsigned char* __restrict__ qw, // [N, K/2] frag-ordered nibbles
    const unsigned short* __restrict__ sc,  // [N, K/128] bf16 scale
    const unsigned short* __restrict__ zs,  // [N, K/128] bf16 zero*scale
    int M, int N, int K) {
  const int mtiles = (M + W4_BM - 1) / W4_BM;
  const int nwg = mtiles * (N / W4_BN);
  int wg = blockIdx.x;
  {  // bijective XCD remap (8 XCDs share the X panel via their L2s)
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg % mtiles) * W4_BM;
  const int n0 = (wg / mtiles) * W4_BN;

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wave = tid / WAVE_SIZE;
  const int lc = lane & 15;
  const int lg = lane >> 4;
  const int col = n0 + wave * 16 + lc;  // this lane's output column

  f32x4 acc[W4_BM / 16];
#pragma unroll
  for (int mf = 0; mf < W4_BM / 16; ++mf) acc[mf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nblk = K / W4_KC;
  const long qrow = (long)col * (K / 2);
  const long srow = (long)col * nblk;

  // A-fragments read straight from L2 (decode-shaped X is a few MB and
  // every N-workgroup re-reads the same panel): no LDS staging, no
  // barriers, occupancy limited only by registers. Row base offsets are
  // per-mf invariant; rows past M clamp to M-1 (stores are guarded).
  long row_off[W4_BM / 16];
#pragma unroll
  for (int mf = 0; mf < W4_BM / 16; ++mf) {
    int row = m0 + mf * 16 + lc;
    if (row >= M) row = M - 1;
    row_off[mf] = (long)row * K;
  }

  for (int kb = 0; kb < nblk; ++kb) {
    // lane's packed weights for this block: 16B = 4 dwords = 32 nibbles
    const uint4 w = *reinterpret_cast<const uint4*>(
        qw + qrow + (long)kb * 64 + lg * 16);
    const float s = bf2f(sc[srow + kb]);
    const float z = bf2f(zs[srow + kb]);
    u16x8 bfr[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const unsigned int lo = (&w.x)[c];  // dword c = nibbles c*8..c*8+7
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float q4 = (float)((lo >> (4 * j)) & 0xF);
        bfr[c][j] = f2bf(q4 * s - z);
      }
    }
    const int k0 = kb * W4_KC;
#pragma unroll
    for (int mf = 0; mf < W4_BM / 16; ++mf) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const u16x8 a = *reinterpret_cast<const u16x8*>(
            x + row_off[mf] + k0 + c * 32 + lg * 8);
        acc[mf] = w4_mfma(a, bfr[c], acc[mf]);
      }
    }
  }

  // D[row = lg*4 + r][col = lc] per 16x16 tile
#pragma unroll
  for (int mf = 0; mf < W4_BM / 16; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + mf * 16 + lg * 4 + r;
      if (row < M) out[(long)row * N + col] = f2bf(acc[mf][r]);
    }
  }
}

// Transient dequant for prefill-shaped GEMMs (bandwidth-bound): packed
// frag-order nibbles -> bf16 [N, K] in canonical k order, so hipBLASLt
// consumes the result directly. Each thread expands one 16-byte packed
// chunk (lane-lg slice of a 128-k block) into its four 8-value octets at
// k = blk*128 + c*32 + lg*8.
__global__ __launch_bounds__(256) void w4_dequant_kernel(
    unsigned short* __restrict__ out,      // [N, K] bf16
    const unsigned char* __restrict__ qw,  // [N, K/2] frag-ordered
    const unsigned short* __restrict__ sc, // [N, K/128]
    const unsigned short* __restrict__ zs, // [N, K/128]
    long N, int K) {
  const long chunks = N * (K / 32);        // 16B packed chunks
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < chunks;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / (K / 32);
    const int ck = idx % (K / 32);         // chunk within row
    const int blk = ck / 4;
    const int lg = ck % 4;
    const uint4 w = *reinterpret_cast<const uint4*>(
        qw + row * (K / 2) + (long)blk * 64 + lg * 16);
    const float s = bf2f(sc[row * (K / 128) + blk]);
    const float z = bf2f(zs[row * (K / 128) + blk]);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const unsigned int d = (&w.x)[c];
      unsigned short o[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf((float)((d >> (4 * j)) & 0xF) * s - z);
      *reinterpret_cast<u16x8*>(out + row * K + blk * 128 + c * 32 + lg * 8) =
          *reinterpret_cast<u16x8*>(o);
    }
  }
}

}  // namespace

void w4_dequant_launch(void* out, const void* qw, const void* sc,
                       const void* zs, long N, int K, int* err_unsupported,
                       hipStream_t s) {
  *err_unsupported = 0;
  if (K % W4_KC != 0) { *err_unsupported = 1; return; }
  const long chunks = N * (K / 32);
  long grid = (chunks + 255) / 256;
  if (grid > 16384) grid = 16384;
  hipLaunchKernelGGL(w4_dequant_kernel, dim3((unsigned)grid), dim3(256), 0, s,
                     (unsigned short*)out, (const unsigned char*)qw,
                     (const unsigned short*)sc, (const unsigned short*)zs, N,
                     K);
}

void w4_gemm_launch(void* out, const void* x, const void* qw, const void* sc,
                    const void* zs, int M, int N, int K,
                    int* err_unsupported, hipStream_t s) {
  *err_unsupported = 0;
  if (N % W4_BN != 0 || K % W4_KC != 0) { *err_unsupported = 1; return; }
  const int mtiles = (M + W4_BM - 1) / W4_BM;
  dim3 grid(mtiles * (N / W4_BN));
  hipLaunchKernelGGL(w4_gemm_kernel, grid, dim3(W4_THREADS), 0, s,
                     (unsigned short*)out, (const unsigned short*)x,
                     (const unsigned char*)qw, (const unsigned short*)sc,
                     (const unsigned short*)zs, M, N, K);
}
