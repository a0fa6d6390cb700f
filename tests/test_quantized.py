"""GPTQ / AWQ W4A16 checkpoint loading (reference: vLLM --quantization
gptq/awq via gpustack backend_parameters): dequant-on-load to bf16."""
import json
from pathlib import Path

import pytest
import torch
from safetensors.torch import save_file

import gpustack_amd.engine  # noqa: F401  (resolves the models<->engine import order)
from gpustack_amd.models.quantized import (
    AWQ_ORDER, dequant_awq, dequant_gptq, maybe_dequant, quant_config,
)


def _pack_gptq_rows(q: torch.Tensor) -> torch.Tensor:
    """q [in, out] uint4 -> int32 [in/8, out] (sequential nibbles)."""
    n_in, n_out = q.shape
    q = q.reshape(n_in // 8, 8, n_out).to(torch.int64)
    w = torch.zeros(n_in // 8, n_out, dtype=torch.int64)
    for j in range(8):
        w |= q[:, j] << (4 * j)
    return w.to(torch.int32)


def _pack_awq_cols(q: torch.Tensor) -> torch.Tensor:
    """q [rows, out] uint4 -> int32 [rows, out/8] (AWQ interleaved order)."""
    rows, n_out = q.shape
    q = q.reshape(rows, n_out // 8, 8).to(torch.int64)
    w = torch.zeros(rows, n_out // 8, dtype=torch.int64)
    for j, col in enumerate(AWQ_ORDER):
        w |= q[:, :, col] << (4 * j)
    return w.to(torch.int32)


def _rand_gptq(n_in, n_out, group, seed=0):
    g = torch.Generator().manual_seed(seed)
    q = torch.randint(0, 16, (n_in, n_out), generator=g)
    z = torch.randint(0, 15, (n_in // group, n_out), generator=g)
    s = torch.rand(n_in // group, n_out, generator=g) * 0.05 + 0.01
    gi = torch.arange(n_in) // group
    expected = ((q.float() - (z[gi].float() + 1)) * s[gi]).t().contiguous()
    packed = {
        "qweight": _pack_gptq_rows(q),
        "qzeros": _pack_awq_cols(z)
        if False else _pack_gptq_cols_seq(z),
        "scales": s.to(torch.float16),
    }
    return packed, expected


def _pack_gptq_cols_seq(z: torch.Tensor) -> torch.Tensor:
    """z [groups, out] -> int32 [groups, out/8] sequential nibbles."""
    gqs, n_out = z.shape
    z = z.reshape(gqs, n_out // 8, 8).to(torch.int64)
    w = torch.zeros(gqs, n_out // 8, dtype=torch.int64)
    for j in range(8):
        w |= z[:, :, j] << (4 * j)
    return w.to(torch.int32)


def test_dequant_gptq_matches_reference():
    packed, expected = _rand_gptq(128, 64, group=32)
    got = dequant_gptq(packed["qweight"], packed["qzeros"],
                       packed["scales"].float(), None)
    assert torch.allclose(got, expected, atol=1e-3)


def test_dequant_gptq_desc_act():
    """g_idx permutation (desc_act): groups assigned per input channel."""
    packed, _ = _rand_gptq(128, 64, group=32, seed=3)
    gi = torch.randperm(128) % 4  # arbitrary group map
    got = dequant_gptq(packed["qweight"], packed["qzeros"],
                       packed["scales"].float(), gi)
    # recompute expected with the same permuted group map
    q = _unpack_rows(packed["qweight"])
    z = _unpack_cols_seq(packed["qzeros"])
    s = packed["scales"].float()
    want = ((q.float() - (z[gi].float() + 1)) * s[gi]).t()
    assert torch.allclose(got, want, atol=1e-3)


def _unpack_rows(w):
    out = []
    for j in range(8):
        out.append((w.to(torch.int64) >> (4 * j)) & 0xF)
    return torch.stack(out, dim=1).reshape(w.shape[0] * 8, w.shape[1])


def _unpack_cols_seq(w):
    out = []
    for j in range(8):
        out.append((w.to(torch.int64) >> (4 * j)) & 0xF)
    return torch.stack(out, dim=-1).reshape(w.shape[0], w.shape[1] * 8)


def test_dequant_awq_matches_reference():
    g = torch.Generator().manual_seed(1)
    n_in, n_out, group = 64, 128, 32
    q = torch.randint(0, 16, (n_in, n_out), generator=g)
    z = torch.randint(0, 16, (n_in // group, n_out), generator=g)
    s = torch.rand(n_in // group, n_out, generator=g) * 0.05 + 0.01
    gi = torch.arange(n_in) // group
    expected = ((q.float() - z[gi].float()) * s[gi]).t().contiguous()
    got = dequant_awq(_pack_awq_cols(q), _pack_awq_cols(z), s.float())
    assert torch.allclose(got, expected, atol=1e-3)


def _write_quant_checkpoint(tmp: Path, method: str) -> dict:
    """Tiny llama checkpoint with quantized projections; returns the
    expected dequantized fp weights by name."""
    h, i, v, nl = 128, 256, 512, 2
    nh, nkv, d = 4, 2, 32
    cfg = {
        "architectures": ["LlamaForCausalLM"],
        "hidden_size": h, "intermediate_size": i, "vocab_size": v,
        "num_hidden_layers": nl, "num_attention_heads": nh,
        "num_key_value_heads": nkv, "head_dim": d,
        "max_position_embeddings": 512, "rope_theta": 10000.0,
        "rms_norm_eps": 1e-5, "eos_token_id": 1,
        "quantization_config": {"quant_method": method, "bits": 4,
                                "group_size": 32},
    }
    (tmp / "config.json").write_text(json.dumps(cfg))
    g = torch.Generator().manual_seed(5)
    tensors = {
        "model.embed_tokens.weight": torch.randn(v, h, generator=g) * 0.02,
        "model.norm.weight": torch.ones(h),
        "lm_head.weight": torch.randn(v, h, generator=g) * 0.02,
    }
    expected: dict = {}

    def add_quant(base: str, n_in: int, n_out: int):
        group = 32
        q = torch.randint(0, 16, (n_in, n_out), generator=g)
        z = torch.randint(0, 15, (n_in // group, n_out), generator=g)
        s = torch.rand(n_in // group, n_out, generator=g) * 0.02 + 0.005
        gi = torch.arange(n_in) // group
        s16 = s.to(torch.float16).float()  # the file stores fp16 scales
        if method == "gptq":
            expected[base] = ((q.float() - (z[gi].float() + 1)) * s16[gi]).t()
            tensors[base + ".qweight"] = _pack_gptq_rows(q)
            tensors[base + ".qzeros"] = _pack_gptq_cols_seq(z)
        else:
            expected[base] = ((q.float() - z[gi].float()) * s16[gi]).t()
            tensors[base + ".qweight"] = _pack_awq_cols(q)
            tensors[base + ".qzeros"] = _pack_awq_cols(z)
        tensors[base + ".scales"] = s.to(torch.float16)

    for li in range(nl):
        p = f"model.layers.{li}."
        add_quant(p + "self_attn.q_proj", h, nh * d)
        add_quant(p + "self_attn.k_proj", h, nkv * d)
        add_quant(p + "self_attn.v_proj", h, nkv * d)
        add_quant(p + "self_attn.o_proj", nh * d, h)
        add_quant(p + "mlp.gate_proj", h, i)
        add_quant(p + "mlp.up_proj", h, i)
        add_quant(p + "mlp.down_proj", i, h)
        tensors[p + "input_layernorm.weight"] = torch.ones(h)
        tensors[p + "post_attention_layernorm.weight"] = torch.ones(h)
    save_file({k: v.contiguous() for k, v in tensors.items()},
              str(tmp / "model.safetensors"))
    return expected


@pytest.mark.parametrize("method", ["gptq", "awq"])
def test_engine_loads_quantized_checkpoint(tmp_path, method):
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    expected = _write_quant_checkpoint(tmp_path, method)
    eng = LLMEngine(EngineConfig(model=str(tmp_path), device="cpu",
                                 kv_cache_blocks=64,
                                 enforce_random_weights=False))
    # dequantized projections landed in the fused layout
    want_q = expected["model.layers.0.self_attn.q_proj"].to(torch.bfloat16)
    got_q = eng.runner.model.layers[0].attn.qkv_w[: want_q.shape[0]]
    assert torch.equal(got_q.float(), want_q.float())
    want_d = expected["model.layers.1.mlp.down_proj"].to(torch.bfloat16)
    assert torch.equal(eng.runner.model.layers[1].mlp.down_w.float(),
                       want_d.float())
    out = eng.generate([[1, 2, 3, 4]], SamplingParams(max_tokens=6,
                                                      ignore_eos=True))[0]
    assert len(out) == 6


def test_quant_config_detection(tmp_path):
    (tmp_path / "config.json").write_text(json.dumps(
        {"quantization_config": {"quant_method": "gptq", "bits": 4}}))
    assert quant_config(tmp_path) == {"method": "gptq", "bits": 4}
    (tmp_path / "config.json").write_text(json.dumps({"hidden_size": 1}))
    assert quant_config(tmp_path) is None
    (tmp_path / "config.json").write_text(json.dumps(
        {"quantization_config": {"quant_method": "fp8"}}))
    with pytest.raises(NotImplementedError):
        quant_config(tmp_path)


def test_maybe_dequant_missing_raises():
    with pytest.raises(KeyError):
        maybe_dequant({}, "x.weight", {"method": "gptq", "bits": 4})


def _quant_tp_rank_main(rank: int, port: int, out_path: str, model_dir: str):
    import json
    import os

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.parallel import init_tp

    comm = init_tp(2, rank, master_port=port, backend="gloo")
    cfg = EngineConfig(model=model_dir, device="cpu", kv_cache_blocks=64,
                       max_model_len=128, seed=0, tp_size=2, tp_rank=rank,
                       enforce_random_weights=False)
    eng = LLMEngine(cfg, comm)
    results, rids = {}, []
    if rank == 0:
        rids = [eng.add_request([1, 2, 3, 4],
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
def test_gptq_tp2_matches_tp1(tmp_path):
    """Dequant-on-load + TP row sharding compose (also exercises the
    checkpoint loader's TP path end to end)."""
    import json as _json
    import multiprocessing as mp
    import socket
    import tempfile

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    _write_quant_checkpoint(tmp_path, "gptq")
    want = LLMEngine(EngineConfig(model=str(tmp_path), device="cpu",
                                  kv_cache_blocks=64, max_model_len=128,
                                  enforce_random_weights=False)).generate(
        [[1, 2, 3, 4]], SamplingParams(max_tokens=6, ignore_eos=True))

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    out_path = tempfile.mktemp(suffix=".json")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_quant_tp_rank_main,
                         args=(r, port, out_path, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path) as f:
        assert _json.load(f) == want
