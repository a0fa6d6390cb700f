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
