"""Model runner: turns a ScheduledBatch into device tensors, runs the
forward pass on the native ops, and samples next tokens."""
from __future__ import annotations

import os

import torch

from .. import ops
from ..models.llama import ForwardMeta, LlamaForCausalLM
from ..models.weights import load_safetensors, random_init
from ..parallel import Communicator
from .config import EngineConfig
from .kv_cache import KVCache
from .scheduler import ScheduledBatch
from .sequence import Sequence


class Sampler:
    def __init__(self, device, model_eos_token_id: int | None = None):
        self.device = device
        # the model's EOS id (cfg.spec.eos_token_id) — the one the engine's
        # finish check tests; params.eos_token_id is the guided-decoding
        # terminator and may differ
        self.model_eos_token_id = model_eos_token_id
        self.generator = None
        # guided-JSON decoding: per-token-id decoded strings (set by the
        # engine server; None on TP followers, whose picks rank 0 overwrites)
        self.token_table: list[str] | None = None

    def _guided_pick(self, row: torch.Tensor, seq: Sequence) -> int:
        """Grammar-constrained pick (engine/guided.py): sync the machine
        with the emitted text, then take the best token whose string keeps
        the output a valid JSON prefix (top-down candidate rejection)."""
        from .guided import GuidedJsonState

        p = seq.params
        table = self.token_table
        if table is None:
            return int(row.argmax())
        if seq.guided_sm is None:
            if p.guided_regex is not None:
                from .guided import GuidedRegexState

                seq.guided_sm = GuidedRegexState(p.guided_regex)
            elif p.guided_grammar is not None:
                from .guided import GuidedGrammarState

                seq.guided_sm = GuidedGrammarState(p.guided_grammar)
            else:
                schema = p.guided_json if isinstance(p.guided_json, dict) else None
                seq.guided_sm = GuidedJsonState(schema)
            seq.guided_consumed = 0
        sm = seq.guided_sm
        out = seq.output_token_ids
        for tok in out[seq.guided_consumed:]:
            m = sm.try_advance(table[tok] if tok < len(table) else "")
            if m is not None:
                sm.commit(m)
        seq.guided_consumed = len(out)
        # `complete` means the text so far is a finished match — but the
        # machine may still be extensible ('\d+' after one digit, a JSON
        # number mid-stream). Forcing EOS here would give shortest-match
        # semantics; instead EOS becomes a *candidate* competing on logits
        # with tokens that keep the output matchable (reference guided-
        # decoding behavior: generation continues while still matchable).
        complete = sm.complete

        def valid(tid: int) -> bool:
            if complete and tid == p.eos_token_id:
                return True
            s = table[tid] if tid < len(table) else ""
            return bool(s) and sm.try_advance(s) is not None

        if not p.greedy:
            k = min(256, row.shape[-1])
            vals, idx = torch.topk(row, k)
            ids = idx.tolist()
            valid_j = [j for j in range(k) if valid(ids[j])]
            if complete and p.eos_token_id not in ids:
                # make sure termination stays reachable under sampling
                ids.append(p.eos_token_id)
                vals = torch.cat([vals, row[p.eos_token_id].reshape(1)])
                valid_j.append(len(ids) - 1)
            if valid_j:
                probs = torch.softmax(
                    vals.float()[valid_j] / max(p.temperature, 1e-5), -1)
                if self.generator is None:
                    self.generator = torch.Generator(device=row.device)
                if p.seed is not None:
                    self.generator.manual_seed(p.seed + len(out))
                j = int(torch.multinomial(probs, 1, generator=self.generator))
                return ids[valid_j[j]]
        for tid in torch.argsort(row, descending=True).tolist():
            if valid(tid):
                return tid
        return p.eos_token_id  # vocab cannot extend the prefix: terminate

    def _process_logits(self, row: torch.Tensor, seq: Sequence) -> torch.Tensor:
        """OpenAI/HF-style logit processors (presence/frequency/repetition
        penalties + logit_bias), applied on the fp32 logits row."""
        p = seq.params
        seen = seq.output_token_ids
        if seen and (p.presence_penalty or p.frequency_penalty
                     or p.repetition_penalty != 1.0):
            ids = torch.tensor(seen, dtype=torch.long, device=row.device)
            uniq, counts = torch.unique(ids, return_counts=True)
            if p.repetition_penalty != 1.0:
                vals = row[uniq]
                row[uniq] = torch.where(vals > 0, vals / p.repetition_penalty,
                                        vals * p.repetition_penalty)
            if p.presence_penalty:
                row[uniq] -= p.presence_penalty
            if p.frequency_penalty:
                row[uniq] -= p.frequency_penalty * counts.to(row.dtype)
        if p.logit_bias:
            for tid, b in p.logit_bias.items():
                t = int(tid)
                if 0 <= t < row.shape[-1]:
                    row[t] += float(b)
        if p.min_tokens > 0 and not p.ignore_eos \
                and len(seen) < p.min_tokens:
            # vLLM min-tokens semantics: EOS/stop ids are masked out of the
            # distribution, not merely ignored by the finish check (an
            # unmasked greedy model would emit EOS repeatedly into the
            # user-visible output)
            eos = (self.model_eos_token_id
                   if self.model_eos_token_id is not None else p.eos_token_id)
            if 0 <= eos < row.shape[-1]:
                row[eos] = float("-inf")
            for t in p.stop_token_ids:
                if 0 <= t < row.shape[-1]:
                    row[t] = float("-inf")
        if p.guided_token_seqs:
            out = seq.output_token_ids
            allowed = set()
            done = False
            for choice in p.guided_token_seqs:
                c = list(choice)
                if len(out) < len(c) and c[:len(out)] == out:
                    allowed.add(c[len(out)])
                elif c == out:
                    done = True
            mask = torch.full_like(row, float("-inf"))
            if allowed and not done:
                ids = torch.tensor(sorted(allowed), dtype=torch.long,
                                   device=row.device)
                mask[ids] = 0.0
                row = row + mask
            else:  # complete (or dead-end): force EOS
                mask[p.eos_token_id] = 0.0
                row = row + mask
        return row

    def sample(self, logits: torch.Tensor, seqs: list[Sequence]) -> list[int]:
        if all(s.params.greedy and not s.params.needs_logit_processing
               for s in seqs):
            return ops.greedy_sample(logits).tolist()
        out: list[int] = [0] * len(seqs)
        guided = {i for i, s in enumerate(seqs)
                  if s.params.guided_json is not None
                  or s.params.guided_regex is not None
                  or s.params.guided_grammar is not None}
        greedy_idx = [i for i, s in enumerate(seqs)
                      if s.params.greedy and not s.params.needs_logit_processing]
        proc_greedy = [i for i, s in enumerate(seqs)
                       if i not in guided
                       and s.params.greedy and s.params.needs_logit_processing]
        rand_idx = [i for i, s in enumerate(seqs)
                    if i not in guided and not s.params.greedy]
        for i in guided:
            out[i] = self._guided_pick(logits[i].float(), seqs[i])
        if greedy_idx:
            ids = ops.greedy_sample(logits[greedy_idx])
            for j, i in enumerate(greedy_idx):
                out[i] = int(ids[j])
        for i in proc_greedy:
            row = self._process_logits(logits[i].float(), seqs[i])
            out[i] = int(row.argmax())
        if rand_idx:
            lg = logits[rand_idx].float()
            temps = torch.tensor(
                [seqs[i].params.temperature for i in rand_idx],
                device=logits.device,
            ).unsqueeze(1)
            lg = lg / temps
            probs = torch.softmax(lg, dim=-1)
            for j, i in enumerate(rand_idx):
                p = seqs[i].params
                if p.needs_logit_processing:
                    r = self._process_logits(logits[i].float(), seqs[i])
                    probs[j] = torch.softmax(r / max(p.temperature, 1e-5), dim=-1)
                row = probs[j]
                if p.min_p > 0.0:
                    row = torch.where(row >= p.min_p * row.max(), row,
                                      torch.zeros_like(row))
                if p.top_k > 0 and p.top_k < row.shape[-1]:
                    vals, idx = torch.topk(row, p.top_k)
                    row = torch.zeros_like(row).scatter_(0, idx, vals)
                if p.top_p < 1.0:
                    sorted_p, sorted_i = torch.sort(row, descending=True)
                    cum = torch.cumsum(sorted_p, dim=-1)
                    keep = cum - sorted_p <= p.top_p
                    keep[0] = True
                    row = torch.zeros_like(row).scatter_(0, sorted_i[keep], sorted_p[keep])
                row = row / row.sum()
                if self.generator is None:
                    self.generator = torch.Generator(device=logits.device)
                if p.seed is not None:
                    self.generator.manual_seed(p.seed + len(seqs[i].output_token_ids))
                out[i] = int(torch.multinomial(row, 1, generator=self.generator))
        return out


class ModelRunner:
    def __init__(self, cfg: EngineConfig, comm: Communicator | None = None):
        self.cfg = cfg
        self.comm = comm or Communicator()
        self.cp_prefills = 0
        self.cp_suffixes = 0
        if torch.device(cfg.device).type == "cuda" and (
                cfg.spec.attention_sinks or cfg.spec.sliding_window
                or cfg.spec.attn_logit_softcap
                or cfg.spec.norm_type != "rmsnorm"
                or cfg.spec.head_dim not in (128,)) \
                and not cfg.spec.kv_lora_rank \
                and os.environ.get("GPUSTACK_AMD_OSS_KERNELS") != "1":
            # GPT-OSS / Gemma-class specs need the gated kernel variants;
            # refuse at init with a clear message instead of crashing
            # inside hipGraph capture on the first forward
            raise NotImplementedError(
                f"{cfg.spec.architecture} GPU serving needs the gated "
                "CDNA4 kernel variants (sinks/window/softcap/layernorm/"
                f"head_dim {cfg.spec.head_dim}) — set "
                "GPUSTACK_AMD_OSS_KERNELS=1 "
                "after the r3 validation pass; CPU serving is available")
        if cfg.spec.kv_lora_rank:
            # MLA (DeepSeek): latent attention serves on the CPU oracle
            # path today; the CDNA4 absorbed-attention kernels are the r3
            # item — refuse loudly rather than crash mid-capture
            if torch.device(cfg.device).type == "cuda":
                raise NotImplementedError(
                    "MLA (DeepSeek) GPU serving requires the r3 CDNA4 "
                    "absorbed-attention kernels; CPU serving is available")
            if cfg.kv_cache_dtype == "fp8":
                raise ValueError("fp8 KV is not supported with MLA latent "
                                 "caches yet")
        self.device = torch.device(cfg.device)

        if self.device.type == "cuda" and os.environ.get("GPUSTACK_AMD_BLASLT", "1") == "1":
            try:  # hipBLASLt picks much better skinny-GEMM kernels than rocBLAS
                torch.backends.cuda.preferred_blas_library("cublaslt")
            except Exception:  # noqa: BLE001
                pass
        if self.device.type == "cuda":
            self._init_tunableop()
        self.graph_runner = None
        self.model = LlamaForCausalLM(cfg, self.comm, self.device)
        if cfg.gguf_path and not cfg.enforce_random_weights:
            from ..models.weights import load_gguf

            load_gguf(self.model, cfg, cfg.gguf_path)
        elif cfg.model_dir and not cfg.enforce_random_weights:
            load_safetensors(self.model, cfg, cfg.model_dir)
        else:
            random_init(self.model, cfg)
        if cfg.lora_dirs:
            from ..models.weights import merge_lora

            for d in cfg.lora_dirs:
                n = merge_lora(self.model, cfg, d)
                import logging

                logging.getLogger(__name__).info("merged LoRA %s (%d tensors)", d, n)
        if getattr(cfg, "quantize_runtime", None) == "w4":
            from ..models.weights import convert_to_w4_runtime

            n = convert_to_w4_runtime(self.model, cfg)
            import logging

            logging.getLogger(__name__).info(
                "W4 runtime: %d weight tensors packed int4", n)
        if getattr(cfg, "cpu_offload_gb", 0) > 0:
            from .offload import setup_cpu_offload

            setup_cpu_offload(self.model, cfg.cpu_offload_gb)
        self.sampler = Sampler(self.device, cfg.spec.eos_token_id)
        from ..models.lora import LoraBank

        self.lora_bank = LoraBank(cfg.spec, cfg)
        self.kv: KVCache | None = None
        # async decode pipeline buffers (pinned side is double-buffered:
        # step N+1's D2H must not overwrite step N before the host reads it)
        mb = cfg.max_num_seqs
        self._sampled_dev = torch.zeros(mb, dtype=torch.long, device=self.device)
        pin = self.device.type == "cuda"
        self._sampled_pin = [torch.zeros(mb, dtype=torch.long, pin_memory=pin)
                             for _ in range(2)]
        self._pin_idx = 0

    def _init_tunableop(self) -> None:
        """hipBLASLt algorithm selection via torch TunableOp.

        The shipped per-shape tuning table (ops/tunableop_gfx950.csv, produced
        by scripts/tune_gemms once on an MI355X) is loaded read-only; set
        GPUSTACK_AMD_TUNE=1 to re-tune and write a fresh table.
        """
        from pathlib import Path

        try:
            tunable = torch.cuda.tunable
        except AttributeError:
            return
        tuning = os.environ.get("GPUSTACK_AMD_TUNE", "0") == "1"
        shipped = Path(__file__).resolve().parent.parent / "ops" / "tunableop_gfx950.csv"
        if tuning:
            out = os.environ.get("GPUSTACK_AMD_TUNE_OUT", "gpurun_out/tunableop_gfx950.csv")
            Path(out).parent.mkdir(parents=True, exist_ok=True)
            tunable.set_filename(out)
            tunable.enable(True)
            tunable.tuning_enable(True)
        elif shipped.exists() and os.environ.get("GPUSTACK_AMD_TUNABLEOP", "1") == "1":
            tunable.set_filename(str(shipped))
            tunable.enable(True)
            tunable.tuning_enable(False)
            tunable.read_file(str(shipped))

    def init_kv_cache(self) -> KVCache:
        cfg = self.cfg
        local_layers = self.model.num_local_layers
        if cfg.kv_cache_blocks is not None:
            nblocks = cfg.kv_cache_blocks
        elif self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            nblocks = KVCache.compute_num_blocks(cfg, free, local_layers)
        else:
            nblocks = 512
        max_needed = cfg.max_num_seqs * (
            (cfg.max_model_len + cfg.block_size - 1) // cfg.block_size
        )
        nblocks = min(nblocks, max_needed)
        host_blocks = KVCache.compute_host_blocks(cfg, local_layers)
        if self.comm.world_size > 1:
            # scheduling is replicated on every TP/PP rank and gates
            # admission/preemption on allocator capacity; ranks measure
            # different free memory (and PP stages hold different layer
            # counts), so without consensus the lockstep schedulers diverge
            # under KV pressure -> mismatched batches -> collective hang.
            # Take the world-wide MIN of both pool sizes.
            import torch.distributed as dist

            t = torch.tensor([nblocks, host_blocks], dtype=torch.long,
                             device=self.device)
            dist.all_reduce(t, op=dist.ReduceOp.MIN)
            nblocks, host_blocks = int(t[0]), int(t[1])
        self.kv = KVCache(cfg, nblocks, self.device, num_layers=local_layers,
                          host_blocks=host_blocks)
        self.eagle = None
        spec = getattr(cfg, "speculative", None)
        if spec and spec.get("method") in ("eagle", "eagle3", "mtp"):
            if self.comm.pp_size > 1:
                raise ValueError("draft-model speculative decoding is not "
                                 "supported with pipeline parallelism")
            if (cfg.spec.kv_lora_rank or cfg.spec.sandwich_norms
                    or cfg.spec.norm_after or cfg.spec.num_experts):
                raise ValueError(
                    "draft-model speculative decoding currently supports "
                    "dense pre-norm decoder families (llama/qwen/mistral "
                    "class) — MLA / sandwich-norm / norm-after / MoE "
                    "targets are not wired to the EAGLE draft layer yet")
            from .eagle import EagleProposer

            k = int(spec.get("num_draft_tokens", 3))
            draft_blocks = min(nblocks, cfg.max_num_seqs * (
                (cfg.max_model_len + k + cfg.block_size) // cfg.block_size + 1))
            self.eagle = EagleProposer(cfg, self.model, self.comm,
                                       self.device, draft_blocks)
        if self.device.type == "cuda":
            from .graph_runner import DecodeGraphRunner, graphs_enabled

            # MoE decode is graph-capturable since r2: the fused grouped-
            # GEMM dispatch (ops/csrc/moe_gemm.hip) has no host syncs and
            # fixed launch grids. Only the torch fallback paths (unaligned
            # dims) still sync per layer -> eager for those.
            moe_graph_ok = (cfg.spec.num_experts == 0
                            or (cfg.spec.moe_intermediate_size
                                // max(1, cfg.tp_size) % 128 == 0
                                and cfg.spec.hidden_size % 128 == 0
                                and cfg.spec.moe_act == "silu"
                                and not cfg.spec.moe_bias
                                and os.environ.get("GPUSTACK_AMD_FUSED_MOE",
                                                   "1") == "1"
                                and ops.hip_available()))
            # EAGLE decode graphs (the verify forward captures like any
            # decode batch, additionally returning hidden states) are
            # OPT-IN for now: the targeted eagle/spec GPU tests pass with
            # them, but the full suite showed a crash in sequence with
            # other engines on one box — isolate before defaulting on.
            eagle_graphs = (self.eagle is not None
                            and os.environ.get("GPUSTACK_AMD_EAGLE_GRAPHS",
                                               "0") == "1")
            if graphs_enabled() and moe_graph_ok \
                    and (self.eagle is None or eagle_graphs) \
                    and self.comm.pp_size == 1 \
                    and self.model.offload is None:
                self.graph_runner = DecodeGraphRunner(
                    self, want_hidden=self.eagle is not None)
                self.graph_runner.capture()
        return self.kv

    def add_lora(self, name: str, adapter_dir: str) -> int:
        if self.cfg.spec.kv_lora_rank:
            raise ValueError("dynamic LoRA is not supported on MLA "
                             "(DeepSeek) models yet — the latent attention "
                             "projections have no adapter hook")
        return self.lora_bank.add(name, adapter_dir, self.device,
                                  getattr(torch, self.cfg.dtype))

    def remove_lora(self, name: str) -> bool:
        return self.lora_bank.remove(name)

    def _row_lora_slots(self, batch: ScheduledBatch) -> list[int] | None:
        """Per-row adapter slots, aligned with the batch's flat row order
        (prefill tokens first, then decode rows). None when no row uses an
        adapter — the common case pays one all() over the seq list."""
        if not self.lora_bank.adapters:
            return None
        if all(s.lora_slot == 0 for s in batch.seqs):
            return None
        slots: list[int] = []
        if batch.is_prefill:
            n_pre = batch.n_prefill_seqs or len(batch.seqs)
            for s, L in zip(batch.seqs[:n_pre], batch.seq_lens[:n_pre]):
                slots.extend([s.lora_slot] * L)
            slots.extend(s.lora_slot for s in batch.seqs[n_pre:])
        elif batch.is_suffix:
            for s, nr in zip(batch.seqs, batch.suffix_rows):
                slots.extend([s.lora_slot] * nr)
        else:
            for s in batch.seqs:
                slots.extend([s.lora_slot] * batch.rows_per_seq)
        return slots

    def batch_uses_lora(self, batch: ScheduledBatch) -> bool:
        return bool(self.lora_bank.adapters) and any(
            s.lora_slot != 0 for s in batch.seqs)

    def _cp_splittable(self, batch: ScheduledBatch) -> bool:
        """CP-split applies to pure-prefill batches and to suffix/chunk
        batches on the paged-prefill path; anything else (mixed decode
        rows, LoRA row groups, fp8-KV suffix fallback) falls back to
        replicated execution — correct by construction, just not
        accelerated."""
        if self.comm.cp_size == 1:
            return False
        if self.cfg.spec.kv_lora_rank:
            return False  # MLA latent gather is an r3 item; replicated CP
        if self.batch_uses_lora(batch):
            return False
        if batch.is_prefill:
            n_pre = batch.n_prefill_seqs or len(batch.seqs)
            return n_pre == len(batch.seqs)  # mixed: decode rows ride along
        if batch.is_suffix:
            # chunk continuations / prefix-cache suffixes: the CP path runs
            # the prefill-with-history attention, so it needs the same
            # conditions as the paged-prefill kernel dispatch
            if self.kv is None:
                return False
            if self.device.type == "cuda":
                return (self.kv.kv_dtype == torch.bfloat16
                        and self.cfg.spec.head_dim == 128)
            return self.kv.kv_dtype != torch.float8_e4m3fn
        return False

    def _meta_cp(self, batch: ScheduledBatch) -> tuple[torch.Tensor, ForwardMeta]:
        """Prefill context parallelism (parallel/cp.py): this rank embeds
        and runs only its contiguous position chunk of every sequence;
        attention sees the full K/V via the per-layer CP gather + cache
        write and runs the local rows through the prefill-with-history
        path (hist = chunk start). Only the tail rank (cp_size-1) produces
        logits rows; execute() broadcasts its sampled ids world-wide."""
        from ..parallel import CPMeta, build_cp_prefill

        self.cp_prefills += 1  # observability: tests assert CP engaged
        if batch.is_suffix:
            self.cp_suffixes += 1
        dev = self.device
        comm = self.comm
        if batch.is_suffix:
            # chunk-continuation / suffix rows: split each seq's NEW rows;
            # this rank's history = the seq's cached tokens + the chunk
            # rows owned by earlier CP ranks
            lens = list(batch.suffix_rows)
            base_hists = [
                batch.seq_lens[sum(batch.suffix_rows[:i])] - 1
                for i in range(len(batch.seqs))
            ]
        else:
            lens = list(batch.seq_lens)
            base_hists = [0] * len(batch.seqs)
        local_rows, hists, news, perm, pad_rows, _counts = build_cp_prefill(
            lens, comm.cp_size, comm.cp_rank)
        hists = [b + h for b, h in zip(base_hists, hists)]
        tok_l = [batch.token_ids[i] for i in local_rows]
        pos_l = [batch.positions[i] for i in local_rows]
        starts, off = [], 0
        for n in news:
            starts.append(off)
            off += n
        idx = []
        if comm.cp_rank == comm.cp_size - 1:
            # floor-bound partition: the tail chunk is never empty, so this
            # rank owns every sequence's last row
            idx = [st + n - 1 for st, n in zip(starts, news)]
        maxb = max(len(s.block_table) for s in batch.seqs)
        nseq = len(batch.seqs)
        dummy = not tok_l
        bt = torch.zeros(nseq + (1 if dummy else 0), maxb, dtype=torch.int32)
        for i, s in enumerate(batch.seqs):
            bt[i, : len(s.block_table)] = torch.tensor(s.block_table,
                                                       dtype=torch.int32)
        if dummy:
            # a rank with zero rows (every seq shorter than cp) still joins
            # the per-layer CP gathers: run one throwaway row that attends
            # seq 0's first cached position; perm never selects it and its
            # output produces no logits
            tok_l = [int(batch.token_ids[0])]
            pos_l = [0]
            starts, hists, news = [0], [0], [1]
            bt[nseq] = bt[0]
        tokens = torch.as_tensor(tok_l, dtype=torch.long).to(dev)
        tiles = ops.build_paged_prefill_tiles(starts, hists, news, dev)
        meta = ForwardMeta(
            is_prefill=False,
            positions=torch.as_tensor(pos_l, dtype=torch.long).to(dev),
            slot_mapping=torch.as_tensor(batch.slot_mapping,
                                         dtype=torch.long).to(dev),
            logits_indices=torch.tensor(idx, dtype=torch.long, device=dev),
            block_tables=bt.to(dev),
            suffix_meta=(tiles, starts, hists, news),
            cp=CPMeta(comm, pad_rows, perm.to(dev)),
        )
        return tokens, meta

    def _meta(self, batch: ScheduledBatch) -> tuple[torch.Tensor, ForwardMeta]:
        if self._cp_splittable(batch):
            return self._meta_cp(batch)
        dev = self.device
        tokens = torch.as_tensor(batch.token_ids, dtype=torch.long).to(dev)
        positions = torch.as_tensor(batch.positions, dtype=torch.long).to(dev)
        slots = torch.as_tensor(batch.slot_mapping, dtype=torch.long).to(dev)
        if batch.is_prefill:
            # last token of each prefill sequence produces its logits row
            n_pre = batch.n_prefill_seqs or len(batch.seqs)
            pre_lens = batch.seq_lens[:n_pre]
            idx, off = [], 0
            for L in pre_lens:
                idx.append(off + L - 1)
                off += L
            tp = batch.num_prefill_tokens or batch.num_tokens
            n_dec = len(batch.seqs) - n_pre
            # decode rows (mixed batch) produce logits directly
            idx.extend(range(tp, tp + n_dec))
            tiles = ops.build_prefill_tiles(pre_lens, dev)
            bt = None
            dec_lens = None
            if self.cfg.spec.kv_lora_rank and dev.type == "cuda":
                # MLA gated GPU path: the absorbed kernel reads the latent
                # through per-seq block tables even for prefill rows
                maxb_p = max(len(s.block_table) for s in batch.seqs[:n_pre])
                btp = torch.zeros(n_pre, maxb_p, dtype=torch.int32)
                for i, sq in enumerate(batch.seqs[:n_pre]):
                    btp[i, : len(sq.block_table)] = torch.tensor(
                        sq.block_table, dtype=torch.int32)
                bt = btp.to(dev)
            if n_dec:
                maxb = max(len(s.block_table) for s in batch.seqs[n_pre:])
                btc = torch.zeros(n_dec, maxb, dtype=torch.int32)
                for i, s in enumerate(batch.seqs[n_pre:]):
                    btc[i, : len(s.block_table)] = torch.tensor(s.block_table, dtype=torch.int32)
                bt = btc.to(dev)
                dec_lens = torch.tensor(batch.decode_seq_lens, dtype=torch.int32, device=dev)
            meta = ForwardMeta(
                is_prefill=True,
                positions=positions,
                slot_mapping=slots,
                logits_indices=torch.tensor(idx, dtype=torch.long, device=dev),
                seq_lens_list=pre_lens,
                tile_start=tiles[0], tile_q0=tiles[1], tile_len=tiles[2],
                block_tables=bt,
                seq_lens=dec_lens,
                num_prefill_tokens=tp,
            )
        elif batch.is_suffix:
            # prefix-cache suffix / chunk-continuation rows. Preferred path:
            # the paged prefill-with-history MFMA kernel (one flash pass per
            # seq whose K/V stream starts from cached blocks). Fallback to
            # per-row paged-decode only where the kernel doesn't apply
            # (fp8 KV cache, head_dim != 128 on GPU).
            hists = [L0 - 1 for L0 in
                     (batch.seq_lens[sum(batch.suffix_rows[:i])]
                      for i in range(len(batch.seqs)))]
            news = list(batch.suffix_rows)
            starts = []
            off = 0
            for nr in news:
                starts.append(off)
                off += nr
            idx = [st + nr - 1 for st, nr in zip(starts, news)]
            if dev.type == "cuda":
                use_pp = (self.kv is not None
                          and self.kv.kv_dtype == torch.bfloat16
                          and self.cfg.spec.head_dim == 128)
            else:  # torch_ref path handles any dtype/head_dim except fp8
                use_pp = (self.kv is not None
                          and self.kv.kv_dtype != torch.float8_e4m3fn)
            if use_pp:
                maxb = max(len(s.block_table) for s in batch.seqs)
                bt = torch.zeros(len(batch.seqs), maxb, dtype=torch.int32)
                for i, s in enumerate(batch.seqs):
                    bt[i, : len(s.block_table)] = torch.tensor(
                        s.block_table, dtype=torch.int32)
                tiles = ops.build_paged_prefill_tiles(starts, hists, news, dev)
                meta = ForwardMeta(
                    is_prefill=False,
                    positions=positions,
                    slot_mapping=slots,
                    logits_indices=torch.tensor(idx, dtype=torch.long,
                                                device=dev),
                    block_tables=bt.to(dev),
                    suffix_meta=(tiles, starts, hists, news),
                )
            else:
                nrows = sum(batch.suffix_rows)
                maxb = max(len(s.block_table) for s in batch.seqs)
                bt = torch.zeros(nrows, maxb, dtype=torch.int32)
                r = 0
                for s, nr in zip(batch.seqs, batch.suffix_rows):
                    row = torch.tensor(s.block_table, dtype=torch.int32)
                    for _ in range(nr):
                        bt[r, : len(s.block_table)] = row
                        r += 1
                meta = ForwardMeta(
                    is_prefill=False,
                    positions=positions,
                    slot_mapping=slots,
                    logits_indices=torch.tensor(idx, dtype=torch.long,
                                                device=dev),
                    block_tables=bt.to(dev),
                    seq_lens=torch.tensor(batch.seq_lens, dtype=torch.int32,
                                          device=dev),
                )
        else:
            rps = batch.rows_per_seq
            nrows = len(batch.seqs) * rps
            maxb = max(len(s.block_table) for s in batch.seqs)
            bt = torch.zeros(nrows, maxb, dtype=torch.int32)
            for i, s in enumerate(batch.seqs):
                row = torch.tensor(s.block_table, dtype=torch.int32)
                for j in range(rps):
                    bt[i * rps + j, : len(s.block_table)] = row
            meta = ForwardMeta(
                is_prefill=False,
                positions=positions,
                slot_mapping=slots,
                logits_indices=torch.arange(nrows, dtype=torch.long, device=dev),
                block_tables=bt.to(dev),
                seq_lens=torch.tensor(batch.seq_lens, dtype=torch.int32, device=dev),
            )
        slots = self._row_lora_slots(batch)
        if slots is not None:
            from ..models.lora import BatchLora

            meta.lora = BatchLora.from_rows(self.lora_bank, slots, dev)
        return tokens, meta

    @torch.inference_mode()
    def score_tokens(self, prompts: list[list[int]],
                     token_ids: list[int]) -> list[list[float]]:
        """Last-token logits restricted to `token_ids`, per prompt — the
        generative-reranker primitive (Qwen3-Reranker class models score
        relevance as P("yes") vs P("no") at the judgment position).
        KV-free prefill like embed()."""
        dev = self.device
        lens = [min(len(p), self.cfg.max_model_len) for p in prompts]
        flat: list[int] = []
        positions: list[int] = []
        for p, L in zip(prompts, lens):
            flat.extend(p[:L])
            positions.extend(range(L))
        tokens = torch.tensor(flat, dtype=torch.long, device=dev)
        pos = torch.tensor(positions, dtype=torch.long, device=dev)
        slots = torch.full((len(flat),), -1, dtype=torch.long, device=dev)
        tiles = ops.build_prefill_tiles(lens, dev)
        idx, off = [], 0
        for L in lens:
            idx.append(off + L - 1)
            off += L
        meta = ForwardMeta(
            is_prefill=True, positions=pos, slot_mapping=slots,
            logits_indices=torch.tensor(idx, dtype=torch.long, device=dev),
            seq_lens_list=lens,
            tile_start=tiles[0], tile_q0=tiles[1], tile_len=tiles[2],
        )
        logits = self.model(tokens, meta, self.kv)
        if self.comm.pp_size > 1:
            if logits is None:
                logits = torch.empty(len(prompts),
                                     self.cfg.spec.vocab_size,
                                     dtype=getattr(torch, self.cfg.dtype),
                                     device=dev)
            self.comm.broadcast_world(logits, src=self.comm.last_stage_rank)
        sel = logits.float()[:, token_ids]
        return sel.tolist()

    @torch.inference_mode()
    def embed(self, prompts: list[list[int]], pooling: str = "last") -> list[list[float]]:
        """Embedding forward (reference category "embedding",
        schemas/models.py:51): prefill with KV writes disabled
        (slot_mapping = -1), pooled + L2-normalized hidden states."""
        dev = self.device
        lens = [min(len(p), self.cfg.max_model_len) for p in prompts]
        flat: list[int] = []
        positions: list[int] = []
        for p, L in zip(prompts, lens):
            flat.extend(p[:L])
            positions.extend(range(L))
        tokens = torch.tensor(flat, dtype=torch.long, device=dev)
        pos = torch.tensor(positions, dtype=torch.long, device=dev)
        slots = torch.full((len(flat),), -1, dtype=torch.long, device=dev)
        tiles = ops.build_prefill_tiles(lens, dev)
        meta = ForwardMeta(
            is_prefill=True, positions=pos, slot_mapping=slots,
            logits_indices=torch.zeros(1, dtype=torch.long, device=dev),
            seq_lens_list=lens,
            tile_start=tiles[0], tile_q0=tiles[1], tile_len=tiles[2],
        )
        hidden = self.model(tokens, meta, self.kv, return_hidden=True)
        if self.comm.pp_size > 1:
            if hidden is None:  # non-final stage: receive the broadcast
                hidden = torch.empty(len(flat), self.cfg.spec.hidden_size,
                                     dtype=getattr(torch, self.cfg.dtype),
                                     device=dev)
            self.comm.broadcast_world(hidden, src=self.comm.last_stage_rank)
        hidden = hidden.float()
        out: list[list[float]] = []
        off = 0
        for L in lens:
            seg = hidden[off:off + L]
            vec = seg.mean(dim=0) if pooling == "mean" else seg[-1]
            vec = vec / (vec.norm() + 1e-12)
            out.append(vec.cpu().tolist())
            off += L
        return out

    @torch.inference_mode()
    def execute_async(self, batch: ScheduledBatch, reuse_tokens: bool):
        """Submit a greedy decode step without reading results back.

        Samples on-device into the persistent buffer and starts an async
        D2H copy; read_sampled() collects it one step later. With
        reuse_tokens, the previous step's sampled ids feed this step's
        token input device-side (no host round-trip at all)."""
        bs = len(batch.seqs)
        token_src = self._sampled_dev if reuse_tokens else None
        if (self.graph_runner is not None and self.graph_runner.can_run(batch)
                and not self.batch_uses_lora(batch)):
            logits = self.graph_runner.run(batch, token_src=token_src)
        else:
            tokens, meta = self._meta(batch)
            if token_src is not None:
                tokens = token_src[:bs].to(self.device)
            logits = self.model(tokens, meta, self.kv)
        from .. import ops

        ops.greedy_sample_into(self._sampled_dev[:bs], logits)
        slot = self._pin_idx
        self._pin_idx ^= 1
        self._sampled_pin[slot][:bs].copy_(self._sampled_dev[:bs], non_blocking=True)
        if self.device.type == "cuda":
            ev = torch.cuda.Event()
            ev.record()
            return (ev, slot)
        return (None, slot)

    def read_sampled(self, bs: int, handle) -> list[int]:
        ev, slot = handle
        if ev is not None:
            ev.synchronize()
        return self._sampled_pin[slot][:bs].tolist()

    @torch.inference_mode()
    def execute(self, batch: ScheduledBatch) -> list[int]:
        self.last_hidden = None
        if self.eagle is not None:
            # draft-model speculative: the verify step must also surface the
            # target hidden states that condition the next draft window.
            # Decode-shaped verify batches replay the captured hipGraph
            # (want_hidden=True); prefill seeding stays eager.
            if (not batch.is_prefill and self.graph_runner is not None
                    and self.graph_runner.can_run(batch)
                    and not self.batch_uses_lora(batch)):
                logits, self.last_hidden = self.graph_runner.run(batch)
            elif batch.is_prefill:
                tokens, meta = self._meta(batch)
                from ..models.llama import qlinear

                hidden_all = self.model(tokens, meta, self.kv, return_hidden=True)
                logits = qlinear(hidden_all[meta.logits_indices],
                                 self.model.lm_head, self.model.lm_head_pack)
                self.last_hidden = hidden_all
            else:
                tokens, meta = self._meta(batch)
                logits, self.last_hidden = self.model(tokens, meta, self.kv,
                                                      return_both=True)
        elif (self.graph_runner is not None and self.graph_runner.can_run(batch)
                and not self.batch_uses_lora(batch)):
            logits = self.graph_runner.run(batch)
        else:
            tokens, meta = self._meta(batch)
            logits = self.model(tokens, meta, self.kv)
        row_seqs = batch.seqs
        if not batch.is_prefill and batch.rows_per_seq > 1:
            row_seqs = [s for s in batch.seqs for _ in range(batch.rows_per_seq)]
        if self.comm.pp_size > 1:
            return self._pp_finish(batch, logits, row_seqs)
        if self.comm.cp_size > 1 and self._cp_splittable(batch):
            # CP prefill: only the tail rank holds logits rows; it samples
            # and broadcasts (the CP analog of the PP sampling stage)
            tail = self.comm.cp_rank == self.comm.cp_size - 1
            return self._pp_finish(batch, logits if tail else None, row_seqs,
                                   src=self.comm.cp_tail_rank)
        token_ids = self.sampler.sample(logits, row_seqs)
        self.last_logprobs = None
        self.last_top_logprobs = None
        if any(s.params.logprobs or s.params.top_logprobs for s in batch.seqs):
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=-1)
            chosen = lf.gather(
                1, torch.as_tensor(token_ids, dtype=torch.long,
                                   device=logits.device).unsqueeze(1)
            ).squeeze(1)
            self.last_logprobs = (chosen - lse).tolist()
            k = max((s.params.top_logprobs for s in batch.seqs), default=0)
            if k > 0:
                lsm = lf - lse.unsqueeze(1)
                vals, idx = torch.topk(lsm, min(k, lf.shape[-1]), dim=-1)
                self.last_top_logprobs = [
                    list(zip(idx[r].tolist(), vals[r].tolist()))
                    for r in range(lf.shape[0])
                ]
        if self.comm.world_size > 1:
            # ranks must agree on sampled tokens; rank 0 decides (TP
            # followers and CP replicated-decode ranks receive)
            t = torch.tensor(token_ids, dtype=torch.long, device=self.device)
            self.comm.broadcast_world(t, src=0)
            token_ids = t.tolist()
        return token_ids

    def _pp_finish(self, batch: ScheduledBatch, logits, row_seqs,
                   src: int | None = None) -> list[int]:
        """Sampling-owner epilogue (PP last stage, or the CP tail rank):
        the owning rank samples (its tp rank 0 decides) and every rank
        receives the token ids so the replicated schedulers stay in
        lockstep."""
        import torch.distributed as dist

        if src is None:
            src = self.comm.last_stage_rank
        n = len(row_seqs)
        if logits is not None:
            token_ids = self.sampler.sample(logits, row_seqs)
            t = torch.tensor(token_ids, dtype=torch.long, device=self.device)
        else:
            t = torch.empty(n, dtype=torch.long, device=self.device)
        self.comm.broadcast_world(t, src=src)
        token_ids = t.tolist()
        self.last_logprobs = None
        self.last_top_logprobs = None
        if any(s.params.logprobs or s.params.top_logprobs
               for s in batch.seqs):
            if logits is not None:
                lf = logits.float()
                lse = torch.logsumexp(lf, dim=-1)
                chosen = lf.gather(
                    1, torch.as_tensor(token_ids, dtype=torch.long,
                                       device=logits.device).unsqueeze(1)
                ).squeeze(1)
                tops = None
                k = max((s.params.top_logprobs for s in batch.seqs),
                        default=0)
                if k > 0:
                    lsm = lf - lse.unsqueeze(1)
                    vals, idx = torch.topk(lsm, min(k, lf.shape[-1]), dim=-1)
                    tops = [list(zip(idx[r].tolist(), vals[r].tolist()))
                            for r in range(lf.shape[0])]
                obj = [((chosen - lse).tolist(), tops)]
            else:
                obj = [None]
            dist.broadcast_object_list(obj, src=src)
            if obj[0] is not None:
                self.last_logprobs, self.last_top_logprobs = obj[0]
        return token_ids
