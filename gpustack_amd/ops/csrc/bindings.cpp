// Torch bindings for the gpustack_amd HIP ops (native ROCm: no hipify,
// streams come straight from c10::hip).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

// launchers (defined in the .hip translation units)
void rms_norm_launch(void*, const void*, const void*, float, int, int, hipStream_t);
void fused_add_rms_norm_launch(void*, void*, const void*, float, int, int, hipStream_t);
void rope_neox_launch(const long*, void*, void*, const float*, int, int, int, int, int, long, long, hipStream_t);
void silu_and_mul_launch(void*, const void*, long, int, hipStream_t);
void reshape_and_cache_launch(const void*, const void*, void*, void*, const long*, int, int, int, int, long, long, int, hipStream_t);
void greedy_sample_launch(long*, const void*, int, int, hipStream_t);
void mla_decode_launch(float*, const void*, const void*, const int*, const int*, int, int, int, int, int, int, float, int*, hipStream_t);
void paged_attn_decode_launch(void*, const void*, const void*, const void*, const int*, const int*, int, int, int, int, int, float, long, int, const float*, int, float, int*, hipStream_t);
void flash_prefill_launch(void*, const void*, const void*, const void*, const int*, const int*, const int*, int, int, int, int, int, float, long, long, long, const float*, int, float, int*, hipStream_t);
void flash_prefill_paged_launch(void*, const void*, const void*, const void*, const int*, const int*, const int*, const int*, const int*, const int*, int, int, int, int, int, float, long, const float*, int, float, int*, hipStream_t);
void mfma_probe_launch(float*, const void*, const void*, hipStream_t);
void skinny_gemm_launch(void*, const void*, const void*, void*, int, int, int, int, hipStream_t);
void gemm8_launch(void*, const void*, const void*, int, int, int, int, int*, hipStream_t);
void skinny_gemm_v2_launch(void*, const void*, const void*, void*, int, int, int, int, hipStream_t);
void gemm_lab_launch(void*, const void*, const void*, int, int, int, int, int*, hipStream_t);
void moe_gate_up_silu_launch(void*, const void*, const void*, const int*, const int*, const int*, const void*, int, int, int, int, int*, hipStream_t);
void moe_down_scale_launch(void*, const void*, const void*, const int*, const int*, const int*, const float*, const void*, int, int, int, int*, hipStream_t);
void w4_gemm_launch(void*, const void*, const void*, const void*, const void*, int, int, int, int*, hipStream_t);
void w4_dequant_launch(void*, const void*, const void*, const void*, long, int, int*, hipStream_t);

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e_ = hipGetLastError();                                       \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",              \
                hipGetErrorString(e_));                                      \
  } while (0)

namespace {

hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.device().index()).stream();
}

void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

bool is_fp8_cache(const at::Tensor& t) {
  return t.scalar_type() == at::kFloat8_e4m3fn || t.scalar_type() == at::kByte;
}

void check_cache(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16 || is_fp8_cache(t),
              name, " must be bf16 or fp8-e4m3");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// [T, H, D] tensor allowed a non-contiguous row stride (view into a fused
// qkv buffer); heads and dims must stay contiguous.
long row_stride_3d(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 3, name, " must be [T, H, D]");
  TORCH_CHECK(t.stride(2) == 1 && t.stride(1) == t.size(2),
              name, " heads/dims must be contiguous");
  return t.stride(0);
}

void rms_norm(at::Tensor out, at::Tensor input, at::Tensor weight, double eps) {
  check_bf16(out, "out"); check_bf16(input, "input"); check_bf16(weight, "weight");
  const int H = input.size(-1);
  const long T = input.numel() / H;
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  rms_norm_launch(out.data_ptr(), input.data_ptr(), weight.data_ptr(),
                  (float)eps, (int)T, H, cur_stream(input));
  HIP_CHECK_LAST();
}

void fused_add_rms_norm(at::Tensor x, at::Tensor residual, at::Tensor weight,
                        double eps) {
  check_bf16(x, "x"); check_bf16(residual, "residual"); check_bf16(weight, "weight");
  const int H = x.size(-1);
  const long T = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  fused_add_rms_norm_launch(x.data_ptr(), residual.data_ptr(),
                            weight.data_ptr(), (float)eps, (int)T, H,
                            cur_stream(x));
  HIP_CHECK_LAST();
}

void rotary_embedding(at::Tensor positions, at::Tensor q, at::Tensor k,
                      at::Tensor cos_sin, long head_dim, long rot_dim) {
  const long qs = row_stride_3d(q, "q");
  const long ks = row_stride_3d(k, "k");
  TORCH_CHECK(positions.scalar_type() == at::kLong && positions.is_contiguous());
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat && cos_sin.is_contiguous());
  const int T = positions.size(0);
  const int D = (int)head_dim, R = (int)rot_dim;
  const int Hq = q.size(1);
  const int Hk = k.size(1);
  TORCH_CHECK((R / 2) % 8 == 0, "rot_dim/2 must be a multiple of 8");
  rope_neox_launch(positions.data_ptr<long>(), q.data_ptr(), k.data_ptr(),
                   cos_sin.data_ptr<float>(), T, Hq, Hk, D, R, qs, ks,
                   cur_stream(q));
  HIP_CHECK_LAST();
}

void silu_and_mul(at::Tensor out, at::Tensor x) {
  check_bf16(out, "out"); check_bf16(x, "x");
  const int I = out.size(-1);
  const long T = out.numel() / I;
  TORCH_CHECK(x.size(-1) == 2 * I && I % 8 == 0);
  silu_and_mul_launch(out.data_ptr(), x.data_ptr(), T, I, cur_stream(x));
  HIP_CHECK_LAST();
}

void reshape_and_cache(at::Tensor k, at::Tensor v, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slots) {
  const long ks = row_stride_3d(k, "k");
  const long vs = row_stride_3d(v, "v");
  check_cache(k_cache, "k_cache"); check_cache(v_cache, "v_cache");
  TORCH_CHECK(slots.scalar_type() == at::kLong && slots.is_contiguous());
  const int T = k.size(0), Hkv = k.size(1), D = k.size(2);
  const int BS = k_cache.size(2);
  TORCH_CHECK(k_cache.size(1) == Hkv && k_cache.size(3) == D && D % 8 == 0);
  reshape_and_cache_launch(k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
                           v_cache.data_ptr(), slots.data_ptr<long>(), T, Hkv,
                           D, BS, ks, vs, is_fp8_cache(k_cache) ? 1 : 0,
                           cur_stream(k));
  HIP_CHECK_LAST();
}

void greedy_sample(at::Tensor out, at::Tensor logits) {
  check_bf16(logits, "logits");
  TORCH_CHECK(out.scalar_type() == at::kLong && out.is_contiguous());
  const int N = logits.size(0), V = logits.size(1);
  greedy_sample_launch(out.data_ptr<long>(), logits.data_ptr(), N, V,
                       cur_stream(logits));
  HIP_CHECK_LAST();
}

void mla_decode(at::Tensor ctx_out, at::Tensor q, at::Tensor lat,
                at::Tensor block_tables, at::Tensor seq_lens, double scale) {
  TORCH_CHECK(ctx_out.scalar_type() == at::kFloat && ctx_out.is_contiguous(),
              "ctx_out must be f32 contiguous");
  check_bf16(q, "q"); check_bf16(lat, "lat");
  TORCH_CHECK(q.is_contiguous() && lat.is_contiguous());
  TORCH_CHECK(block_tables.scalar_type() == at::kInt && block_tables.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int N = q.size(0), H = q.size(1), LD = q.size(2);
  const int BS = lat.size(2);
  const int R = ctx_out.size(2);
  TORCH_CHECK(lat.size(3) == LD && ctx_out.size(1) == H);
  const int max_blocks = block_tables.size(1);
  int err = 0;
  mla_decode_launch(ctx_out.data_ptr<float>(), q.data_ptr(), lat.data_ptr(),
                    block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
                    N, H, R, LD - R, BS, max_blocks, (float)scale, &err,
                    cur_stream(q));
  TORCH_CHECK(!err, "mla_decode: unsupported config R=", R, " LD=", LD,
              " BS=", BS, " H=", H);
  HIP_CHECK_LAST();
}

static const float* sink_ptr_checked(const c10::optional<at::Tensor>& sinks,
                                     int Hq) {
  if (!sinks.has_value()) return nullptr;
  TORCH_CHECK(sinks->scalar_type() == at::kFloat && sinks->is_contiguous()
                  && sinks->numel() == Hq,
              "sinks must be float32 [Hq]");
  return sinks->data_ptr<float>();
}

void paged_attn_decode(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor block_tables,
                       at::Tensor seq_lens, double scale,
                       c10::optional<at::Tensor> sinks, long window,
                       double softcap) {
  check_bf16(out, "out");
  const long qstride = row_stride_3d(q, "q");
  check_cache(k_cache, "k_cache"); check_cache(v_cache, "v_cache");
  TORCH_CHECK(block_tables.scalar_type() == at::kInt && block_tables.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int N = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int BS = k_cache.size(2);
  TORCH_CHECK(BS == 16, "decode kernel assumes block_size 16");
  const int max_blocks = block_tables.size(1);
  const float* sink_ptr = sink_ptr_checked(sinks, Hq);
  int err = 0;
  paged_attn_decode_launch(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                           v_cache.data_ptr(), block_tables.data_ptr<int>(),
                           seq_lens.data_ptr<int>(), N, Hq, Hkv, D, max_blocks,
                           (float)scale, qstride,
                           is_fp8_cache(k_cache) ? 1 : 0, sink_ptr,
                           (int)window, (float)softcap, &err, cur_stream(q));
  TORCH_CHECK(!err, "paged_attn_decode: unsupported head_dim/GQ combination: D=",
              D, " Hq=", Hq, " Hkv=", Hkv);
  HIP_CHECK_LAST();
}

void flash_prefill(at::Tensor out, at::Tensor q, at::Tensor k, at::Tensor v,
                   at::Tensor tile_start, at::Tensor tile_q0,
                   at::Tensor tile_len, double scale,
                   c10::optional<at::Tensor> sinks, long window,
                   double softcap) {
  check_bf16(out, "out");
  const long qs = row_stride_3d(q, "q");
  const long ks = row_stride_3d(k, "k");
  const long vs = row_stride_3d(v, "v");
  TORCH_CHECK(tile_start.scalar_type() == at::kInt && tile_start.is_contiguous());
  const int ntiles = tile_start.size(0);
  const int Hq = q.size(1), D = q.size(2), Hkv = k.size(1);
  const int DV = out.size(2);  // value dim (== D except MLA 192/128)
  TORCH_CHECK(v.size(2) == DV, "v/out value dims must match");
  int err = 0;
  flash_prefill_launch(out.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                       tile_start.data_ptr<int>(), tile_q0.data_ptr<int>(),
                       tile_len.data_ptr<int>(), ntiles, Hq, Hkv, D, DV,
                       (float)scale, qs, ks, vs, sink_ptr_checked(sinks, Hq),
                       (int)window, (float)softcap, &err, cur_stream(q));
  TORCH_CHECK(!err, "flash_prefill: unsupported config D=", D, " DV=", DV);
  HIP_CHECK_LAST();
}

void flash_prefill_paged(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                         at::Tensor v_cache, at::Tensor block_tables,
                         at::Tensor tile_qstart, at::Tensor tile_q0,
                         at::Tensor tile_hist, at::Tensor tile_new,
                         at::Tensor tile_seq, double scale,
                         c10::optional<at::Tensor> sinks, long window,
                         double softcap) {
  check_bf16(out, "out");
  const long qs = row_stride_3d(q, "q");
  check_bf16(k_cache, "k_cache"); check_bf16(v_cache, "v_cache");
  for (auto* t : {&block_tables, &tile_qstart, &tile_q0, &tile_hist,
                  &tile_new, &tile_seq}) {
    TORCH_CHECK(t->scalar_type() == at::kInt && t->is_contiguous());
  }
  const int ntiles = tile_qstart.size(0);
  const int Hq = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "paged prefill assumes block_size 16");
  const int maxb = block_tables.size(1);
  int err = 0;
  flash_prefill_paged_launch(
      out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
      block_tables.data_ptr<int>(), tile_qstart.data_ptr<int>(),
      tile_q0.data_ptr<int>(), tile_hist.data_ptr<int>(),
      tile_new.data_ptr<int>(), tile_seq.data_ptr<int>(), ntiles, Hq, Hkv, D,
      maxb, (float)scale, qs, sink_ptr_checked(sinks, Hq), (int)window,
      (float)softcap, &err, cur_stream(q));
  TORCH_CHECK(!err, "flash_prefill_paged: unsupported config D=", D);
  HIP_CHECK_LAST();
}

void gemm8(at::Tensor out, at::Tensor x, at::Tensor w, long safe) {
  check_bf16(out, "out"); check_bf16(x, "x"); check_bf16(w, "w");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  int err = 0;
  gemm8_launch(out.data_ptr(), x.data_ptr(), w.data_ptr(), M, N, K,
               (int)safe, &err, cur_stream(x));
  TORCH_CHECK(!err, "gemm8: unsupported shape M=", M, " N=", N, " K=", K,
              " (need N%256==0, K%128==0)");
  HIP_CHECK_LAST();
}

void skinny_gemm(at::Tensor out, at::Tensor x, at::Tensor w,
                 c10::optional<at::Tensor> workspace, long splitk,
                 long version) {
  check_bf16(out, "out"); check_bf16(x, "x"); check_bf16(w, "w");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(N % 128 == 0, "N must be a multiple of 128");
  TORCH_CHECK(K % (64 * splitk) == 0, "K must be a multiple of 64*splitk");
  void* ws = nullptr;
  if (splitk > 1) {
    TORCH_CHECK(workspace.has_value(), "splitk>1 needs an f32 workspace");
    TORCH_CHECK(workspace->scalar_type() == at::kFloat &&
                workspace->numel() >= (long)splitk * M * N);
    ws = workspace->data_ptr();
  }
  if (version == 2) {
    TORCH_CHECK(N % 256 == 0, "v2 needs N % 256 == 0");
    skinny_gemm_v2_launch(out.data_ptr(), x.data_ptr(), w.data_ptr(), ws, M,
                          N, K, (int)splitk, cur_stream(x));
  } else {
    skinny_gemm_launch(out.data_ptr(), x.data_ptr(), w.data_ptr(), ws, M, N,
                       K, (int)splitk, cur_stream(x));
  }
  HIP_CHECK_LAST();
}

void moe_gate_up_silu(at::Tensor act, at::Tensor x, at::Tensor w,
                      at::Tensor s_tok, at::Tensor offs, at::Tensor counts,
                      c10::optional<at::Tensor> bias, long act_mode) {
  check_bf16(act, "act"); check_bf16(x, "x"); check_bf16(w, "w");
  for (auto* t : {&s_tok, &offs, &counts}) {
    TORCH_CHECK(t->scalar_type() == at::kInt && t->is_contiguous());
  }
  const int E = w.size(0), H = x.size(1);
  const long I2 = w.size(1);
  const int I = (int)(I2 / 2);
  TORCH_CHECK(w.size(2) == H && act.size(1) == I);
  TORCH_CHECK(act.size(0) == s_tok.size(0));
  const void* bias_ptr = nullptr;
  if (bias.has_value()) {
    check_bf16(*bias, "bias");
    TORCH_CHECK(bias->numel() == (long)E * 2 * I, "gate_up bias must be [E, 2I]");
    bias_ptr = bias->data_ptr();
  }
  int err = 0;
  moe_gate_up_silu_launch(act.data_ptr(), x.data_ptr(), w.data_ptr(),
                          s_tok.data_ptr<int>(), offs.data_ptr<int>(),
                          counts.data_ptr<int>(), bias_ptr, (int)act_mode,
                          E, H, I, &err, cur_stream(x));
  TORCH_CHECK(!err, "moe_gate_up_silu: unsupported dims H=", H, " I=", I);
  HIP_CHECK_LAST();
}

void moe_down_scale(at::Tensor contrib, at::Tensor act, at::Tensor w,
                    at::Tensor offs, at::Tensor counts, at::Tensor order,
                    at::Tensor flat_w, c10::optional<at::Tensor> bias) {
  check_bf16(contrib, "contrib"); check_bf16(act, "act"); check_bf16(w, "w");
  for (auto* t : {&offs, &counts, &order}) {
    TORCH_CHECK(t->scalar_type() == at::kInt && t->is_contiguous());
  }
  TORCH_CHECK(flat_w.scalar_type() == at::kFloat && flat_w.is_contiguous());
  const int E = w.size(0), H = w.size(1), I = w.size(2);
  TORCH_CHECK(act.size(1) == I && contrib.size(1) == H);
  TORCH_CHECK(contrib.size(0) == order.size(0));
  const void* bias_ptr = nullptr;
  if (bias.has_value()) {
    check_bf16(*bias, "bias");
    TORCH_CHECK(bias->numel() == (long)E * H, "down bias must be [E, H]");
    bias_ptr = bias->data_ptr();
  }
  int err = 0;
  moe_down_scale_launch(contrib.data_ptr(), act.data_ptr(), w.data_ptr(),
                        offs.data_ptr<int>(), counts.data_ptr<int>(),
                        order.data_ptr<int>(), flat_w.data_ptr<float>(),
                        bias_ptr, E, H, I, &err, cur_stream(act));
  TORCH_CHECK(!err, "moe_down_scale: unsupported dims H=", H, " I=", I);
  HIP_CHECK_LAST();
}

void w4_gemm(at::Tensor out, at::Tensor x, at::Tensor qw, at::Tensor sc,
             at::Tensor zs) {
  check_bf16(out, "out"); check_bf16(x, "x");
  check_bf16(sc, "sc"); check_bf16(zs, "zs");
  TORCH_CHECK(qw.scalar_type() == at::kByte && qw.is_contiguous());
  const int M = x.size(0), K = x.size(1), N = qw.size(0);
  TORCH_CHECK(qw.size(1) == K / 2 && out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(sc.size(0) == N && sc.size(1) == K / 128);
  int err = 0;
  w4_gemm_launch(out.data_ptr(), x.data_ptr(), qw.data_ptr(), sc.data_ptr(),
                 zs.data_ptr(), M, N, K, &err, cur_stream(x));
  TORCH_CHECK(!err, "w4_gemm: unsupported shape M=", M, " N=", N, " K=", K,
              " (need N%64==0, K%128==0)");
  HIP_CHECK_LAST();
}

void w4_dequant(at::Tensor out, at::Tensor qw, at::Tensor sc, at::Tensor zs) {
  check_bf16(out, "out"); check_bf16(sc, "sc"); check_bf16(zs, "zs");
  TORCH_CHECK(qw.scalar_type() == at::kByte && qw.is_contiguous());
  const long N = qw.size(0);
  const int K = (int)(qw.size(1) * 2);
  TORCH_CHECK(out.size(0) == N && out.size(1) == K);
  int err = 0;
  w4_dequant_launch(out.data_ptr(), qw.data_ptr(), sc.data_ptr(),
                    zs.data_ptr(), N, K, &err, cur_stream(out));
  TORCH_CHECK(!err, "w4_dequant: K must be a multiple of 128");
  HIP_CHECK_LAST();
}

void gemm_lab(at::Tensor out, at::Tensor x, at::Tensor w, long mode) {
  check_bf16(out, "out"); check_bf16(x, "x"); check_bf16(w, "w");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  int err = 0;
  gemm_lab_launch(out.data_ptr(), x.data_ptr(), w.data_ptr(), M, N, K,
                  (int)mode, &err, cur_stream(x));
  TORCH_CHECK(!err, "gemm_lab: unsupported shape/mode M=", M, " N=", N,
              " K=", K, " mode=", mode);
  HIP_CHECK_LAST();
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  auto d = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  mfma_probe_launch(d.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                    cur_stream(a));
  HIP_CHECK_LAST();
  return d;
}

}  // namespace

#undef HIP_CHECK_LAST
PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm, "residual+=x; x=norm(residual)*w");
  m.def("rotary_embedding", &rotary_embedding, "in-place neox RoPE on q,k");
  m.def("silu_and_mul", &silu_and_mul, "silu(x[:d])*x[d:]");
  m.def("reshape_and_cache", &reshape_and_cache, "scatter K/V into paged pool");
  m.def("greedy_sample", &greedy_sample, "argmax over vocab");
  m.def("mla_decode", &mla_decode,
        "MLA absorbed decode over the paged latent cache (DeepSeek)");
  m.def("paged_attn_decode", &paged_attn_decode, "paged GQA decode attention",
        py::arg("out"), py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_tables"), py::arg("seq_lens"), py::arg("scale"),
        py::arg("sinks") = py::none(), py::arg("window") = 0,
        py::arg("softcap") = 0.0);
  m.def("flash_prefill", &flash_prefill, "varlen causal MFMA prefill attention",
        py::arg("out"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("tile_start"), py::arg("tile_q0"), py::arg("tile_len"),
        py::arg("scale"), py::arg("sinks") = py::none(),
        py::arg("window") = 0, py::arg("softcap") = 0.0);
  m.def("flash_prefill_paged", &flash_prefill_paged,
        py::arg("out"), py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_tables"), py::arg("tile_qstart"), py::arg("tile_q0"),
        py::arg("tile_hist"), py::arg("tile_new"), py::arg("tile_seq"),
        py::arg("scale"), py::arg("sinks") = py::none(),
        py::arg("window") = 0, py::arg("softcap") = 0.0,
        "MFMA prefill attention with paged-KV history (suffix/chunk rows)");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("skinny_gemm", &skinny_gemm, "split-K skinny GEMM (bf16, f32 accum)");
  m.def("gemm8", &gemm8, "8-phase pipelined 256x256 GEMM (bf16, f32 accum)");
  m.def("gemm_lab", &gemm_lab, "GEMM schedule lab variants (A/B vs hipBLASLt)");
  m.def("moe_gate_up_silu", &moe_gate_up_silu,
        py::arg("act"), py::arg("x"), py::arg("w"), py::arg("s_tok"),
        py::arg("offs"), py::arg("counts"), py::arg("bias") = py::none(),
        py::arg("act_mode") = 0,
        "grouped MoE gate/up GEMM + SiLU (sorted assignments, sync-free)");
  m.def("moe_down_scale", &moe_down_scale,
        py::arg("contrib"), py::arg("act"), py::arg("w"), py::arg("offs"),
        py::arg("counts"), py::arg("order"), py::arg("flat_w"),
        py::arg("bias") = py::none(),
        "grouped MoE down GEMM + routing-weight scale/scatter");
  m.def("w4_gemm", &w4_gemm,
        "W4A16 GEMM: packed-int4 weights dequantized in-register");
  m.def("w4_dequant", &w4_dequant,
        "packed-int4 -> bf16 weight expansion (prefill transient)");
}
