"""First-party GGUF metadata reader (scheduler estimation parity with the
reference's gguf-parser Go binary — SURVEY.md §2 row 2)."""
import pytest

from gpustack_amd.utils.gguf import (GGUFTensorInfo, read_gguf, spec_from_gguf,
                                     write_gguf)


def _fixture(tmp_path, arch="llama"):
    path = tmp_path / "m.gguf"
    meta = {
        "general.architecture": arch,
        f"{arch}.block_count": 4,
        f"{arch}.embedding_length": 64,
        f"{arch}.feed_forward_length": 128,
        f"{arch}.attention.head_count": 8,
        f"{arch}.attention.head_count_kv": 2,
        f"{arch}.context_length": 2048,
        f"{arch}.rope.freq_base": 500000.0,
        f"{arch}.attention.layer_norm_rms_epsilon": 1e-5,
        f"{arch}.vocab_size": 256,
        "general.name": "tiny-test",
        "general.quantized": True,
        "tokenizer.ggml.tokens": ["<s>", "</s>", "a"],
    }
    tensors = [
        ("token_embd.weight", (64, 256), 0),       # F32
        ("blk.0.attn_q.weight", (64, 64), 12),     # Q4_K
        ("blk.0.ffn_down.weight", (128, 64), 14),  # Q6_K
        ("output.weight", (64, 256), 1),           # F16
    ]
    write_gguf(path, meta, tensors)
    return path


def test_roundtrip(tmp_path):
    info = read_gguf(_fixture(tmp_path))
    assert info.version == 3
    assert info.architecture == "llama"
    assert info.metadata["llama.block_count"] == 4
    assert info.metadata["general.quantized"] is True
    assert info.metadata["general.name"] == "tiny-test"
    assert info.metadata["tokenizer.ggml.tokens"] == ["<s>", "</s>", "a"]
    assert len(info.tensors) == 4
    t = {x.name: x for x in info.tensors}
    assert t["token_embd.weight"].nbytes == 64 * 256 * 4
    assert t["output.weight"].nbytes == 64 * 256 * 2
    # Q4_K: 256-element blocks of 144 B
    assert t["blk.0.attn_q.weight"].nbytes == (64 * 64 // 256) * 144
    assert info.n_params == 2 * 64 * 256 + 64 * 64 + 128 * 64


def test_spec_from_gguf(tmp_path):
    spec = spec_from_gguf(_fixture(tmp_path))
    assert spec.num_layers == 4
    assert spec.hidden_size == 64
    assert spec.num_kv_heads == 2
    assert spec.vocab_size == 256
    assert spec.rope_theta == 500000.0
    assert spec.max_position_embeddings == 2048
    assert spec.architecture == "LlamaForCausalLM"


def test_spec_qwen3(tmp_path):
    spec = spec_from_gguf(_fixture(tmp_path, arch="qwen3"))
    assert spec.architecture == "Qwen3ForCausalLM"
    assert spec.qk_norm


def test_scheduler_uses_gguf(tmp_path):
    from gpustack_amd.scheduler.policies import estimate_vram_claim, model_spec_for

    path = _fixture(tmp_path)
    model = {"source": "local_path", "model_ref": str(path),
             "gpu_memory_utilization": 0.9}
    spec = model_spec_for(model)
    assert spec is not None and spec.num_layers == 4
    assert estimate_vram_claim(model, spec, tp=1) > 0


def test_bad_magic(tmp_path):
    p = tmp_path / "x.gguf"
    p.write_bytes(b"NOPE" + b"\x00" * 64)
    with pytest.raises(ValueError):
        read_gguf(p)


def test_unknown_type_fallback():
    t = GGUFTensorInfo("x", (100,), 99, 0)
    assert t.nbytes == 100  # 1 byte/element fallback


# ---- execution: dequantization + load_gguf ---------------------------------

def test_q8_0_roundtrip():
    import numpy as np

    from gpustack_amd.utils.gguf import GGML_Q8_0, dequantize, quantize_q8_0

    rng = np.random.default_rng(0)
    x = rng.standard_normal(32 * 40).astype(np.float32)
    y = dequantize(quantize_q8_0(x), GGML_Q8_0, x.size)
    step = np.abs(x).reshape(-1, 32).max(axis=1) / 127.0
    assert np.abs(y - x).reshape(-1, 32).max(axis=1).max() <= step.max() * 0.51


def test_q4_0_roundtrip():
    import numpy as np

    from gpustack_amd.utils.gguf import GGML_Q4_0, dequantize, quantize_q4_0

    rng = np.random.default_rng(1)
    x = rng.standard_normal(32 * 40).astype(np.float32)
    y = dequantize(quantize_q4_0(x), GGML_Q4_0, x.size)
    step = np.abs(x).reshape(-1, 32).max(axis=1) / 8.0
    err = np.abs(y - x).reshape(-1, 32)
    assert (err <= step[:, None] * 1.01).all()


def _scalar_q4_k(block):
    """Direct translation of ggml dequantize_row_q4_K (independent ref)."""
    import numpy as np

    d = float(np.frombuffer(block[0:2], np.float16)[0])
    dmin = float(np.frombuffer(block[2:4], np.float16)[0])
    scales = block[4:16]
    qs = block[16:144]

    def get_scale_min(j):
        if j < 4:
            return scales[j] & 63, scales[j + 4] & 63
        return ((scales[j + 4] & 0xF) | ((scales[j - 4] >> 6) << 4),
                (scales[j + 4] >> 4) | ((scales[j] >> 6) << 4))

    y = [0.0] * 256
    idx, is_ = 0, 0
    for j in range(0, 256, 64):
        sc1, m1 = get_scale_min(is_)
        sc2, m2 = get_scale_min(is_ + 1)
        for l in range(32):
            y[j + l] = d * sc1 * (qs[idx + l] & 0xF) - dmin * m1
            y[j + 32 + l] = d * sc2 * (qs[idx + l] >> 4) - dmin * m2
        idx += 32
        is_ += 2
    return y


def _scalar_q6_k(block):
    """Direct translation of ggml dequantize_row_q6_K."""
    import numpy as np

    ql = block[0:128]
    qh = block[128:192]
    sc = np.frombuffer(block[192:208], np.int8)
    d = float(np.frombuffer(block[208:210], np.float16)[0])
    y = [0.0] * 256
    for half in range(2):
        yb, qlb, qhb, scb = 128 * half, 64 * half, 32 * half, 8 * half
        for l in range(32):
            is_ = l // 16
            q1 = ((ql[qlb + l] & 0xF) | (((qh[qhb + l] >> 0) & 3) << 4)) - 32
            q2 = ((ql[qlb + l + 32] & 0xF) | (((qh[qhb + l] >> 2) & 3) << 4)) - 32
            q3 = ((ql[qlb + l] >> 4) | (((qh[qhb + l] >> 4) & 3) << 4)) - 32
            q4 = ((ql[qlb + l + 32] >> 4) | (((qh[qhb + l] >> 6) & 3) << 4)) - 32
            y[yb + l] = d * int(sc[scb + is_ + 0]) * q1
            y[yb + l + 32] = d * int(sc[scb + is_ + 2]) * q2
            y[yb + l + 64] = d * int(sc[scb + is_ + 4]) * q3
            y[yb + l + 96] = d * int(sc[scb + is_ + 6]) * q4
    return y


def test_q4_k_matches_scalar_reference():
    import numpy as np

    from gpustack_amd.utils.gguf import GGML_Q4_K, dequantize

    rng = np.random.default_rng(2)
    nb = 5
    blocks = bytearray()
    for _ in range(nb):
        blocks += np.float16(rng.uniform(0.01, 0.1)).tobytes()
        blocks += np.float16(rng.uniform(0.01, 0.1)).tobytes()
        blocks += rng.integers(0, 256, 12, dtype=np.uint8).tobytes()
        blocks += rng.integers(0, 256, 128, dtype=np.uint8).tobytes()
    got = dequantize(bytes(blocks), GGML_Q4_K, nb * 256)
    want = np.array([v for i in range(nb)
                     for v in _scalar_q4_k(bytes(blocks[144 * i:144 * (i + 1)]))],
                    dtype=np.float32)
    assert np.allclose(got, want, atol=1e-5)


def test_q6_k_matches_scalar_reference():
    import numpy as np

    from gpustack_amd.utils.gguf import GGML_Q6_K, dequantize

    rng = np.random.default_rng(3)
    nb = 5
    blocks = bytearray()
    for _ in range(nb):
        blocks += rng.integers(0, 256, 128, dtype=np.uint8).tobytes()  # ql
        blocks += rng.integers(0, 256, 64, dtype=np.uint8).tobytes()   # qh
        blocks += rng.integers(-60, 60, 16, dtype=np.int8).tobytes()   # scales
        blocks += np.float16(rng.uniform(0.01, 0.1)).tobytes()         # d
    got = dequantize(bytes(blocks), GGML_Q6_K, nb * 256)
    want = np.array([v for i in range(nb)
                     for v in _scalar_q6_k(bytes(blocks[210 * i:210 * (i + 1)]))],
                    dtype=np.float32)
    assert np.allclose(got, want, atol=1e-5)


def _export_tiny_gguf(path, eng, quant=None):
    """Export a tiny engine's weights as a llama-arch GGUF, applying the
    convert_hf_to_gguf q/k permute (the loader must undo it)."""
    import numpy as np

    from gpustack_amd.utils.gguf import (
        GGML_F32, GGML_Q8_0, quantize_q8_0, write_gguf_with_data,
    )

    spec = eng.cfg.spec
    d = spec.head_dim

    def permute(w, nh):
        return (w.reshape(nh, 2, d // 2, w.shape[-1])
                 .transpose(1, 2).reshape(nh * d, w.shape[-1]))

    tensors = []

    def add(name, w, q=False):
        a = w.float().cpu().numpy()
        shape = tuple(reversed(a.shape))
        if q:
            tensors.append((name, shape, GGML_Q8_0, quantize_q8_0(a.reshape(-1))))
        else:
            tensors.append((name, shape, GGML_F32,
                            np.ascontiguousarray(a).tobytes()))

    m = eng.runner.model
    add("token_embd.weight", m.embed)
    add("output_norm.weight", m.final_norm)
    add("output.weight", m.lm_head)
    nq, nk = spec.num_heads * d, spec.num_kv_heads * d
    for li, layer in enumerate(m.layers):
        p = f"blk.{li}."
        qkv = layer.attn.qkv_w.data
        add(p + "attn_q.weight", permute(qkv[:nq], spec.num_heads), quant)
        add(p + "attn_k.weight", permute(qkv[nq:nq + nk], spec.num_kv_heads), quant)
        add(p + "attn_v.weight", qkv[nq + nk:], quant)
        add(p + "attn_output.weight", layer.attn.o_w, quant)
        i = spec.intermediate_size
        add(p + "ffn_gate.weight", layer.mlp.gate_up_w[:i], quant)
        add(p + "ffn_up.weight", layer.mlp.gate_up_w[i:], quant)
        add(p + "ffn_down.weight", layer.mlp.down_w, quant)
        add(p + "attn_norm.weight", layer.input_norm)
        add(p + "ffn_norm.weight", layer.post_attn_norm)
    meta = {
        "general.architecture": "llama",
        "llama.attention.head_count": spec.num_heads,
        "llama.attention.head_count_kv": spec.num_kv_heads,
        "llama.attention.key_length": spec.head_dim,
        "llama.embedding_length": spec.hidden_size,
        "llama.feed_forward_length": spec.intermediate_size,
        "llama.block_count": spec.num_layers,
        "llama.vocab_size": spec.vocab_size,
        "llama.context_length": 512,
        "llama.rope.freq_base": float(spec.rope_theta),
        "llama.attention.layer_norm_rms_epsilon": float(spec.rms_norm_eps),
    }
    write_gguf_with_data(path, meta, tensors)


def test_load_gguf_f32_exact(tmp_path):
    """F32 GGUF of the tiny model's weights reproduces its outputs exactly
    (incl. the llama q/k un-permute)."""
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    ref = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64))
    path = tmp_path / "tiny-f32.gguf"
    _export_tiny_gguf(path, ref)
    eng = LLMEngine(EngineConfig(model=str(path), device="cpu",
                                 kv_cache_blocks=64,
                                 enforce_random_weights=False))
    assert eng.cfg.gguf_path == str(path)
    assert eng.cfg.spec.num_layers == ref.cfg.spec.num_layers
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    assert eng.generate([[1, 2, 3, 4, 5]], p) == ref.generate([[1, 2, 3, 4, 5]], p)


def test_load_gguf_q8_0_close(tmp_path):
    """Q8_0-quantized projections: weights land within quantization error
    and the engine decodes."""
    import torch

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    ref = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64))
    path = tmp_path / "tiny-q8.gguf"
    _export_tiny_gguf(path, ref, quant=True)
    eng = LLMEngine(EngineConfig(model=str(path), device="cpu",
                                 kv_cache_blocks=64,
                                 enforce_random_weights=False))
    w_ref = ref.runner.model.layers[0].mlp.down_w.float()
    w_got = eng.runner.model.layers[0].mlp.down_w.float()
    scale = w_ref.abs().max()
    assert (w_got - w_ref).abs().max() <= scale / 127 + 1e-2
    assert torch.equal(ref.runner.model.embed, eng.runner.model.embed)
    out = eng.generate([[1, 2, 3, 4, 5]], SamplingParams(max_tokens=8,
                                                         ignore_eos=True))[0]
    assert len(out) == 8


def _gguf_tp_rank_main(rank: int, port: int, out_path: str, gguf_path: str):
    import json
    import os

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_tp

    comm = init_tp(2, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model=gguf_path, device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, tp_size=2, tp_rank=rank,
                       enforce_random_weights=False)
    eng = LLMEngine(cfg, comm)
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request([1, 2, 3, 4, 5],
                                SamplingParams(max_tokens=6, ignore_eos=True))]
        results = {r: [] for r in rids}
    while eng.tp_active():
        for o in eng.step():
            if rank == 0:
                results[o.request_id].append(o.token_id)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump([results[r] for r in rids], f)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gguf_tp2_matches_tp1(tmp_path):
    """GGUF loader under TP=2: per-rank row shards of the dequantized
    (and llama-unpermuted) tensors compose to the TP=1 weights."""
    import json
    import multiprocessing as mp
    import socket
    import tempfile

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    ref = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, max_model_len=128,
                                 seed=0))
    path = tmp_path / "tiny-tp.gguf"
    _export_tiny_gguf(path, ref)
    want = LLMEngine(EngineConfig(model=str(path), device="cpu",
                                  kv_cache_blocks=64, max_model_len=128,
                                  enforce_random_weights=False)).generate(
        [[1, 2, 3, 4, 5]], SamplingParams(max_tokens=6, ignore_eos=True))

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_gguf_tp_rank_main,
                         args=(r, port, out_path, str(path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path) as f:
        assert json.load(f) == want
