"""LoRA adapter merge-at-load (Model.lora_list parity)."""
import json
import tempfile
from pathlib import Path

import torch
from safetensors.torch import save_file

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _make_adapter(tmp: Path, spec, r=4, alpha=8):
    torch.manual_seed(7)
    tensors = {}
    d = spec.head_dim
    for li in range(spec.num_layers):
        pre = f"base_model.model.model.layers.{li}.self_attn.q_proj"
        tensors[f"{pre}.lora_A.weight"] = torch.randn(r, spec.hidden_size) * 0.05
        tensors[f"{pre}.lora_B.weight"] = torch.randn(spec.num_heads * d, r) * 0.05
        pre = f"base_model.model.model.layers.{li}.mlp.down_proj"
        tensors[f"{pre}.lora_A.weight"] = torch.randn(r, spec.intermediate_size) * 0.05
        tensors[f"{pre}.lora_B.weight"] = torch.randn(spec.hidden_size, r) * 0.05
    save_file(tensors, str(tmp / "adapter_model.safetensors"))
    (tmp / "adapter_config.json").write_text(json.dumps({"r": r, "lora_alpha": alpha}))
    return tensors


def test_lora_merge_changes_weights_correctly():
    tmp = Path(tempfile.mkdtemp())
    base = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64))
    tensors = _make_adapter(tmp, base.runner.cfg.spec)
    lora = LLMEngine(EngineConfig(model="tiny", device="cpu", kv_cache_blocks=64,
                                  lora_dirs=[str(tmp)]))
    spec = base.runner.cfg.spec
    d = spec.head_dim
    nq = spec.num_heads * d
    for li in range(spec.num_layers):
        A = tensors[f"base_model.model.model.layers.{li}.self_attn.q_proj.lora_A.weight"]
        B = tensors[f"base_model.model.model.layers.{li}.self_attn.q_proj.lora_B.weight"]
        delta = (B @ A) * (8 / 4)
        got = (lora.runner.model.layers[li].attn.qkv_w.data[:nq]
               - base.runner.model.layers[li].attn.qkv_w.data[:nq]).float()
        assert torch.allclose(got, delta.to(torch.bfloat16).float(), atol=1e-2, rtol=5e-2)
        # k/v rows untouched
        assert torch.equal(lora.runner.model.layers[li].attn.qkv_w.data[nq:],
                           base.runner.model.layers[li].attn.qkv_w.data[nq:])
    # generation actually changes
    p = SamplingParams(max_tokens=8, ignore_eos=True)
    assert base.generate([[1, 2, 3, 4]], p) != lora.generate([[1, 2, 3, 4]], p)
