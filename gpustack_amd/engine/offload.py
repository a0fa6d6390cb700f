"""CPU weight offload: serve models whose weights exceed the GPU budget.

Reference semantics: vLLM --cpu-offload-gb / the GGUF partial-offload
placement path (/root/reference/gpustack/policies/candidate_selectors/
gguf_resource_fit_selector.py:129-300 computes offload_layers into the
claim; schemas/models.py:623-630 carries it). MI355X-native design: the
offloaded TAIL layers' big weights live PINNED in host DRAM and stream
to a double-buffered device staging area on a side HIP stream one layer
ahead of use — compute stays entirely on the MFMA kernels, and the
GPU-resident working set is bounded by two layer-sized buffers.

hipGraphs are disabled while offload is active (parameter re-binding per
step cannot be captured); the async decode pipeline is unaffected.
"""
from __future__ import annotations

import logging

import torch

logger = logging.getLogger(__name__)

# the per-layer parameters large enough to be worth streaming
_BIG = ("attn.qkv_w", "attn.o_w", "mlp.gate_up_w", "mlp.down_w")


def _get(layer, dotted: str):
    obj = layer
    for part in dotted.split("."):
        obj = getattr(obj, part, None)
        if obj is None:
            return None
    return obj


class CpuOffload:
    """Streams the trailing `num_layers - first` layers' weights from
    pinned host memory; bind(i) must be called before layer i runs."""

    def __init__(self, model, first: int):
        self.model = model
        self.first = first
        self.device = model.device
        self.is_cuda = self.device.type == "cuda"
        n = len(model.layers)
        self.host: dict[int, list[tuple[torch.nn.Parameter, torch.Tensor]]] = {}
        shapes: list[tuple[torch.Size, torch.dtype]] = []
        for li in range(first, n):
            layer = model.layers[li]
            entry = []
            for name in _BIG:
                p = _get(layer, name)
                if p is None or p.numel() == 0:
                    continue
                host = torch.empty_like(p.data, device="cpu",
                                        pin_memory=self.is_cuda)
                host.copy_(p.data)
                entry.append((p, host))
            self.host[li] = entry
            if not shapes:
                shapes = [(h.shape, h.dtype) for _, h in entry]
            for p, _h in entry:
                p.data = torch.empty(0, dtype=p.dtype, device=p.device)
        # two staging slots, each holding one layer's parameter set
        # (homogeneous decoder layers: every offloaded layer shares shapes)
        self.slots = [
            [torch.empty(s, dtype=d, device=self.device) for s, d in shapes]
            for _ in range(2)
        ]
        self.stream = torch.cuda.Stream() if self.is_cuda else None
        self.events = [torch.cuda.Event() if self.is_cuda else None
                       for _ in range(2)]
        self.slot_layer = [-1, -1]
        if self.is_cuda:
            torch.cuda.empty_cache()
        freed = sum(h.numel() * h.element_size()
                    for e in self.host.values() for _, h in e)
        logger.info("cpu offload: layers %d..%d streamed from host "
                    "(%.1f GiB freed on device)", first, n - 1, freed / 2**30)

    def _prefetch(self, li: int, slot: int) -> None:
        if li not in self.host:
            return
        bufs = self.slots[slot]
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                for (_, h), buf in zip(self.host[li], bufs):
                    buf.copy_(h, non_blocking=True)
            self.events[slot].record(self.stream)
        else:
            for (_, h), buf in zip(self.host[li], bufs):
                buf.copy_(h)
        self.slot_layer[slot] = li

    def begin(self) -> None:
        """Call at forward start: prefetch the first offloaded layer."""
        if self.slot_layer[self.first & 1] != self.first:
            self._prefetch(self.first, self.first & 1)

    def bind(self, li: int) -> None:
        """Make layer li's weights resident (blocks on its copy), rebind
        its parameters to the staging slot, and prefetch layer li+1."""
        slot = li & 1
        if self.slot_layer[slot] != li:
            self._prefetch(li, slot)
        if self.events[slot] is not None:
            torch.cuda.current_stream().wait_event(self.events[slot])
        for (p, _h), buf in zip(self.host[li], self.slots[slot]):
            p.data = buf
        nxt = li + 1
        if nxt in self.host and self.slot_layer[nxt & 1] != nxt:
            self._prefetch(nxt, nxt & 1)
        # this slot's buffers are re-used two layers later; the prefetch
        # stream must not overwrite them while the compute stream still
        # reads (the wait_stream in _prefetch orders against compute)
        self.slot_layer[slot] = li


def setup_cpu_offload(model, offload_gb: float) -> CpuOffload | None:
    """Choose the offloaded tail so ~offload_gb of layer weights move to
    host, then attach the streamer to the model."""
    if offload_gb <= 0:
        return None
    per_layer = []
    for layer in model.layers:
        b = sum(_get(layer, n).numel() * _get(layer, n).element_size()
                for n in _BIG if _get(layer, n) is not None
                and _get(layer, n).numel())
        per_layer.append(b)
    budget = int(offload_gb * 2**30)
    acc = 0
    first = len(model.layers)
    for li in range(len(model.layers) - 1, -1, -1):
        if acc >= budget or first <= 1:
            break
        acc += per_layer[li]
        first = li
    if first >= len(model.layers):
        return None
    off = CpuOffload(model, first)
    model.offload = off
    return off
