"""Dynamic multi-LoRA serving: per-request adapters applied unmerged.

Reference parity: vLLM `--enable-lora` + `/v1/load_lora_adapter` dynamic
adapters, which gpustack surfaces as per-LoRA child model routes
(gpustack/server/lora_model_routes.py) and `Model.lora_list`
(gpustack/schemas/models.py:468). Merge-at-load (models/weights.py
merge_lora) remains the zero-overhead path for a single adapter; this
module serves MANY adapters concurrently: each batch row carries an
adapter slot, and every target projection adds its low-rank delta
  out[rows] += (x[rows] @ A^T) @ B^T * (alpha / r)
for the rows of each active adapter. Deltas ride hipBLASLt GEMMs shaped
[t, r] x [r, out] — skinny but tiny next to the base GEMM at realistic
rank (r <= 64), so decode stays HBM-bound on the base weights.

TP sharding matches the base layout: column-parallel targets (q/k/v,
gate/up) shard B's output rows per rank; row-parallel targets (o, down)
shard A's input columns per rank, and the existing post-projection
all-reduce sums the partial low-rank contributions.
"""
from __future__ import annotations

import json
import logging
from pathlib import Path

import torch

logger = logging.getLogger(__name__)

# projection key -> (parent module, fused buffer, slice resolver)
_ATTN_KEYS = ("q_proj", "k_proj", "v_proj", "o_proj")
_MLP_KEYS = ("gate_proj", "up_proj", "down_proj")


class LoraAdapter:
    """One loaded PEFT adapter, sharded for this TP rank.

    mods[(layer_idx, proj)] = (A [r, in_loc], B [out_loc, r]) with the
    alpha/r scaling folded into B.
    """

    def __init__(self, name: str, mods: dict, rank: int):
        self.name = name
        self.mods = mods
        self.rank = rank

    @classmethod
    def load(cls, name: str, adapter_dir: str | Path, spec, tp: int,
             tp_rank: int, device, dtype) -> "LoraAdapter":
        from safetensors import safe_open

        adapter_dir = Path(adapter_dir)
        alpha, r = 16.0, 8.0
        cfg_path = adapter_dir / "adapter_config.json"
        if cfg_path.exists():
            with open(cfg_path) as f:
                ac = json.load(f)
            alpha = float(ac.get("lora_alpha", alpha))
            r = float(ac.get("r", r))
        scaling = alpha / r

        files = sorted(adapter_dir.glob("*.safetensors"))
        if not files:
            raise FileNotFoundError(f"no adapter safetensors under {adapter_dir}")
        tensors: dict[str, torch.Tensor] = {}
        for f in files:
            with safe_open(str(f), framework="pt") as sf:
                for key in sf.keys():
                    tensors[key] = sf.get_tensor(key)

        d = spec.head_dim
        hq, hkv = spec.num_heads // tp, max(1, spec.num_kv_heads // tp)
        i_loc = spec.intermediate_size // tp
        h = spec.hidden_size

        def find(li: int, proj: str, which: str) -> torch.Tensor | None:
            for key, t in tensors.items():
                if (f"layers.{li}." in key and f"{proj}." in key
                        and f"lora_{which}" in key):
                    return t.float()
            return None

        # (proj -> (B row-shard offset, rows)) for column-parallel targets
        col_shards = {
            "q_proj": (tp_rank * hq * d, hq * d),
            "k_proj": (tp_rank * hkv * d, hkv * d),
            "v_proj": (tp_rank * hkv * d, hkv * d),
            "gate_proj": (tp_rank * i_loc, i_loc),
            "up_proj": (tp_rank * i_loc, i_loc),
        }
        row_shards = {  # A column-shard (input dim) for row-parallel targets
            "o_proj": (tp_rank * hq * d, hq * d),
            "down_proj": (tp_rank * i_loc, i_loc),
        }
        mods: dict = {}
        rank_seen = int(r)
        for li in range(spec.num_layers):
            for proj in _ATTN_KEYS + _MLP_KEYS:
                A, B = find(li, proj, "A"), find(li, proj, "B")
                if A is None or B is None:
                    continue
                rank_seen = A.shape[0]
                if proj in col_shards:
                    off, rows = col_shards[proj]
                    B = B[off:off + rows]
                else:
                    off, cols = row_shards[proj]
                    A = A[:, off:off + cols]
                mods[(li, proj)] = (
                    A.to(dtype).to(device).contiguous(),
                    (B * scaling).to(dtype).to(device).contiguous(),
                )
        if spec.num_experts > 0 and any(k[1] in _MLP_KEYS for k in mods):
            raise ValueError("LoRA on MoE expert MLPs is not supported; "
                             "restrict target_modules to attention projections")
        if not mods:
            raise ValueError(f"adapter {adapter_dir} targets no supported "
                             f"projections (q/k/v/o/gate/up/down)")
        logger.info("loaded LoRA adapter %s: %d tensors, r=%d",
                    name, len(mods), rank_seen)
        return cls(name, mods, rank_seen)


class LoraBank:
    """Registry of live adapters; slot 0 is reserved for "no adapter"."""

    def __init__(self, spec, cfg):
        self.spec = spec
        self.cfg = cfg
        self.adapters: dict[int, LoraAdapter] = {}
        self.by_name: dict[str, int] = {}
        self._next_slot = 1

    def add(self, name: str, adapter_dir: str, device, dtype) -> int:
        if name in self.by_name:
            return self.by_name[name]
        ad = LoraAdapter.load(name, adapter_dir, self.spec, self.cfg.tp_size,
                              self.cfg.tp_rank, device, dtype)
        slot = self._next_slot
        self._next_slot += 1
        self.adapters[slot] = ad
        self.by_name[name] = slot
        return slot

    def remove(self, name: str) -> bool:
        slot = self.by_name.pop(name, None)
        if slot is None:
            return False
        self.adapters.pop(slot, None)
        return True

    def slot_of(self, name: str) -> int | None:
        return self.by_name.get(name)

    def names(self) -> list[str]:
        return list(self.by_name)


class BatchLora:
    """Per-batch LoRA context: adapter row groups, precomputed host-side so
    the per-layer apply is pure device work (no syncs on the hot path)."""

    def __init__(self, bank: LoraBank, groups: list[tuple[int, torch.Tensor]]):
        self.bank = bank
        self.groups = groups  # [(slot, row-index tensor on device)]

    @classmethod
    def from_rows(cls, bank: LoraBank, row_slots: list[int], device):
        groups: dict[int, list[int]] = {}
        for i, s in enumerate(row_slots):
            if s > 0:
                groups.setdefault(s, []).append(i)
        if not groups:
            return None
        return cls(bank, [
            (slot, torch.tensor(rows, dtype=torch.long, device=device))
            for slot, rows in sorted(groups.items())
        ])

    def _delta(self, A, B, x, rows):
        xe = x.index_select(0, rows)
        return torch.nn.functional.linear(
            torch.nn.functional.linear(xe, A), B)

    def apply(self, li: int, x: torch.Tensor, out: torch.Tensor,
              projs: list[tuple[str, int, int]]) -> None:
        """out[rows, off:off+n] += lora_delta for each (proj, off, n)."""
        for slot, rows in self.groups:
            ad = self.bank.adapters.get(slot)
            if ad is None:
                continue
            for proj, off, n in projs:
                mod = ad.mods.get((li, proj))
                if mod is None:
                    continue
                A, B = mod
                d = self._delta(A, B, x, rows)
                out[:, off:off + n].index_add_(0, rows, d)
