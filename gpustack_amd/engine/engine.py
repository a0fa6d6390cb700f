"""LLMEngine: continuous-batching serving engine for MI355X.

The first-party replacement for the external engines the reference launches
in containers. One engine instance owns one GPU (or one
TP rank group); the worker's serve manager runs it in a subprocess behind
an OpenAI-compatible HTTP endpoint (gpustack_amd/worker/engine_server.py).
"""
from __future__ import annotations

import itertools
import logging

from ..parallel import Communicator
from .config import EngineConfig
from .model_runner import ModelRunner
from .scheduler import Scheduler
from .sequence import SamplingParams, Sequence, SeqStatus, StepOutput

logger = logging.getLogger(__name__)


class LLMEngine:
    def __init__(self, cfg: EngineConfig, comm: Communicator | None = None):
        self.cfg = cfg
        self.runner = ModelRunner(cfg, comm)
        self.comm = self.runner.comm
        if self.comm.cp_size > 1:
            if cfg.speculative and cfg.speculative.get("method") in (
                    "eagle", "eagle3", "mtp"):
                raise ValueError("draft-model speculative decoding is not "
                                 "supported with context parallelism")
        kv = self.runner.init_kv_cache()
        self.scheduler = Scheduler(cfg, kv)
        self.seqs: dict[str, Sequence] = {}
        self._counter = itertools.count()
        # TP: rank 0 buffers add/abort ops; step() broadcasts them so every
        # rank replays the same scheduler state deterministically
        self._pending_ops: list[tuple] = []
        # async decode pipeline: (seqs, event) of the submitted-but-unread
        # step; the host processes step N-1 while the GPU runs step N
        import os

        self._async_enabled = (
            os.environ.get("GPUSTACK_AMD_ASYNC", "1") == "1"
            and self.comm.world_size == 1
            and cfg.speculative is None
        )
        self._pending: tuple[list[Sequence], object] | None = None
        self._carry_outputs: list[StepOutput] = []
        logger.info(
            "engine ready: model=%s kv_blocks=%d (%.1f GiB KV pool)",
            cfg.model, kv.num_blocks,
            kv.num_blocks * cfg.block_size * cfg.spec.kv_bytes_per_token() / 2**30,
        )

    # -- API ---------------------------------------------------------------
    def add_request(
        self,
        prompt_token_ids: list[int],
        params: SamplingParams | None = None,
        request_id: str | None = None,
    ) -> str:
        rid = request_id or f"req-{next(self._counter)}"
        params = params or SamplingParams()
        if (params.guided_json is not None or params.guided_regex is not None
                or params.guided_grammar is not None) \
                and (self.comm.pp_size > 1 or self.comm.cp_size > 1):
            raise ValueError("guided decoding is not supported with pipeline "
                             "or context parallelism (prefill sampling runs "
                             "on a rank without the token table)")
        if self.comm.world_size > 1:
            assert self.comm.world_rank == 0, "requests enter through rank 0"
            self._pending_ops.append(("add", list(prompt_token_ids),
                                      params.__dict__.copy(), rid))
            return rid
        self._apply_add(prompt_token_ids, params, rid)
        return rid

    def _apply_add(self, prompt_token_ids, params: SamplingParams, rid: str) -> None:
        if len(prompt_token_ids) > self.cfg.max_model_len - 1:
            prompt_token_ids = prompt_token_ids[-(self.cfg.max_model_len - 1):]
        seq = Sequence(rid, list(prompt_token_ids), params)
        if params.lora_name:
            slot = self.runner.lora_bank.slot_of(params.lora_name)
            if slot is None:
                logger.warning("unknown LoRA adapter %r; serving base model",
                               params.lora_name)
            else:
                seq.lora_slot = slot
        self.seqs[rid] = seq
        self.scheduler.add(seq)

    # -- dynamic multi-LoRA (reference: vLLM /v1/load_lora_adapter, gpustack
    # per-LoRA model routes) ------------------------------------------------
    def add_lora(self, name: str, adapter_dir: str) -> None:
        if self.comm.world_size > 1:
            assert self.comm.world_rank == 0
            self._pending_ops.append(("lora_add", name, adapter_dir))
            return
        self.runner.add_lora(name, adapter_dir)

    def remove_lora(self, name: str) -> bool:
        if self.comm.world_size > 1:
            assert self.comm.world_rank == 0
            self._pending_ops.append(("lora_rm", name))
            return True
        return self.runner.remove_lora(name)

    def set_token_table(self, table: list[str]) -> None:
        """Per-token-id decoded strings for guided-JSON decoding (rank 0
        only; follower picks are overwritten by the rank-0 broadcast)."""
        self.runner.sampler.token_table = table

    def lora_names(self) -> list[str]:
        names = list(self.runner.lora_bank.names())
        names.extend(op[1] for op in self._pending_ops if op[0] == "lora_add")
        return names

    def abort_request(self, request_id: str) -> bool:
        if self.comm.world_size > 1:
            assert self.comm.world_rank == 0
            self._pending_ops.append(("abort", request_id))
            return True
        if self._pending is not None:
            # an in-flight graph still writes this seq's KV slot; drain
            # before the abort can release blocks for reuse
            self._carry_outputs.extend(self._drain())
        ok = self.scheduler.abort(request_id)
        seq = self.seqs.pop(request_id, None)
        if seq is not None and self.runner.eagle is not None:
            self.runner.eagle.drop(seq)
        return ok

    def _sync_tp_ops(self) -> None:
        """Broadcast buffered add/abort ops from rank 0 and apply on all
        ranks (rank bootstrap + op replication over the RCCL/gloo group)."""
        import torch.distributed as dist

        # world-group broadcast: every TP AND PP rank replays the same ops
        ops = [self._pending_ops] if self.comm.world_rank == 0 else [None]
        dist.broadcast_object_list(ops, src=0)
        self._pending_ops = []
        for op in ops[0]:
            if op[0] == "add":
                _, toks, params_d, rid = op
                self._apply_add(toks, SamplingParams(**params_d), rid)
            elif op[0] == "abort":
                rid = op[1]
                self.scheduler.abort(rid)
                seq = self.seqs.pop(rid, None)
                if seq is not None and self.runner.eagle is not None:
                    self.runner.eagle.drop(seq)
            elif op[0] == "lora_add":
                self.runner.add_lora(op[1], op[2])
            elif op[0] == "lora_rm":
                self.runner.remove_lora(op[1])

    def has_unfinished(self) -> bool:
        return self.scheduler.has_work()

    def tp_active(self) -> bool:
        """Coordinated loop condition: every rank (TP and PP) keeps
        stepping while rank 0 has work (requests enter only through rank 0)."""
        if self.comm.world_size == 1:
            return self.has_unfinished()
        import torch.distributed as dist

        if self.comm.world_rank == 0:
            flag = [self.has_unfinished() or bool(self._pending_ops)]
        else:
            flag = [None]
        dist.broadcast_object_list(flag, src=0)
        return bool(flag[0])

    @property
    def num_running(self) -> int:
        return len(self.scheduler.running)

    @property
    def num_waiting(self) -> int:
        return len(self.scheduler.waiting)

    # -- async decode pipeline --------------------------------------------

    def _must_drain_before_schedule(self) -> bool:
        sch = self.scheduler
        if sch.would_admit():
            return True  # admission changes composition
        if sch.kv.allocator.num_free < len(sch.running) + 8:
            return True  # preemption possible under pressure
        for seq in self._pending[0]:
            p = seq.params
            if not p.ignore_eos:
                return True  # unpredictable stop
            if len(seq.output_token_ids) + seq.pending_tokens >= p.max_tokens:
                return True  # imminent length finish
            if seq.num_tokens + 1 >= self.cfg.max_model_len:
                return True
        return False

    def _drain(self) -> list[StepOutput]:
        entry = self._pending
        self._pending = None
        return self._drain_entry(entry)

    def _drain_entry(self, entry) -> list[StepOutput]:
        seqs, ev = entry
        toks = self.runner.read_sampled(len(seqs), ev)
        outputs: list[StepOutput] = []
        for seq, tok in zip(seqs, toks):
            seq.pending_tokens -= 1
            if seq.status == SeqStatus.FINISHED:
                continue  # aborted/finished while in flight
            tok = int(tok)
            seq.record_first_token()
            seq.output_token_ids.append(tok)
            reason = self._finish_reason(seq, tok)
            outputs.append(StepOutput(seq.request_id, tok, reason is not None, reason))
            if reason:
                self.scheduler.finish_seq(seq, reason)
                self.seqs.pop(seq.request_id, None)
        return outputs

    @staticmethod
    def _same_seqs(a: list[Sequence], b: list[Sequence]) -> bool:
        return len(a) == len(b) and all(x is y for x, y in zip(a, b))

    def _refill_tokens(self, batch) -> None:
        """Replace placeholder tokens with the real (now drained) values."""
        if batch.is_suffix:
            return  # suffix/chunk rows carry prompt tokens, not placeholders
        if batch.is_prefill:
            base = batch.num_prefill_tokens
            dec = batch.seqs[batch.n_prefill_seqs:]
        else:
            base = 0
            dec = batch.seqs
        for i, seq in enumerate(dec):
            out = seq.output_token_ids
            batch.token_ids[base + i] = out[-1] if out else seq.prompt_token_ids[-1]

    def _async_ok(self, batch) -> bool:
        return (
            self._async_enabled
            and not batch.is_prefill
            and not batch.is_suffix
            and batch.rows_per_seq == 1
            and all(s.params.greedy and not s.params.logprobs
                    and not s.params.top_logprobs
                    and not s.params.needs_logit_processing
                    for s in batch.seqs)
        )

    def step(self) -> list[StepOutput]:
        if self.comm.world_size > 1:
            self._sync_tp_ops()
        outputs: list[StepOutput] = []
        if self._carry_outputs:
            outputs.extend(self._carry_outputs)
            self._carry_outputs = []
        if self._pending is not None and self._must_drain_before_schedule():
            outputs.extend(self._drain())
        batch = self.scheduler.schedule()
        if batch is None:
            if self._pending is not None:
                outputs.extend(self._drain())
            return outputs
        if self._async_ok(batch):
            reuse = self._pending is not None and self._same_seqs(self._pending[0], batch.seqs)
            if self._pending is not None and not reuse:
                outputs.extend(self._drain())
                self._refill_tokens(batch)
            old = self._pending if reuse else None
            ev = self.runner.execute_async(batch, reuse_tokens=reuse)
            for seq in batch.seqs:
                seq.pending_tokens += 1
            self._pending = (list(batch.seqs), ev)
            if old is not None:
                outputs.extend(self._drain_entry(old))
            return outputs
        if self._pending is not None:  # sync fallback with stale placeholders
            outputs.extend(self._drain())
            self._refill_tokens(batch)
        token_ids = self.runner.execute(batch)
        lps = getattr(self.runner, "last_logprobs", None)
        tops = getattr(self.runner, "last_top_logprobs", None)
        if batch.is_prefill or batch.is_suffix:
            self.scheduler.on_prefill_done(batch)
        rps = 1 if (batch.is_prefill or batch.is_suffix) else batch.rows_per_seq
        finals = batch.chunk_final
        emitted_all: list[list[int]] = []
        for i, seq in enumerate(batch.seqs):
            if finals is not None and i < len(finals) and not finals[i]:
                emitted_all.append([])
                continue  # intermediate prefill chunk: discard sampled token
            if rps == 1:
                emitted = [token_ids[i]]
                row0 = i
            else:
                rows = token_ids[i * rps:(i + 1) * rps]
                row0 = i * rps
                if seq.params.spec_safe and batch.k_eff[i] > 0:
                    from .spec import accept_tokens

                    ke = batch.k_eff[i]
                    emitted = accept_tokens(batch.drafts[i][:ke], rows[:ke + 1])
                else:
                    emitted = [rows[0]]
            emitted_all.append(emitted)
            seq.record_first_token()
            for j, tok in enumerate(emitted):
                seq.output_token_ids.append(tok)
                reason = self._finish_reason(seq, tok)
                lp = (lps[row0 + j] if (lps is not None and seq.params.logprobs)
                      else None)
                top = (tops[row0 + j][:seq.params.top_logprobs]
                       if (tops is not None and seq.params.top_logprobs)
                       else None)
                outputs.append(StepOutput(seq.request_id, tok,
                                          reason is not None, reason, lp, top))
                if reason:
                    self.scheduler.finish_seq(seq, reason)
                    self.seqs.pop(seq.request_id, None)
                    break
        self._eagle_post_step(batch, token_ids, emitted_all)
        return outputs

    def _eagle_post_step(self, batch, token_ids, emitted_all) -> None:
        """Draft-model speculative: extend draft KV + propose the next
        window from the hiddens the verify step produced."""
        eagle = self.runner.eagle
        if eagle is None:
            return
        alive = [s.status != SeqStatus.FINISHED for s in batch.seqs]
        hidden = self.runner.last_hidden
        if batch.is_prefill:
            n_pre = batch.n_prefill_seqs or len(batch.seqs)
            if n_pre == len(batch.seqs) and hidden is not None:
                eagle.seed_from_prefill(batch, hidden, token_ids, alive)
        elif batch.rows_per_seq > 1 and hidden is not None:
            eagle.step(batch, hidden, emitted_all, alive)
        for ok, seq in zip(alive, batch.seqs):
            if not ok:
                eagle.drop(seq)

    def _finish_reason(self, seq: Sequence, tok: int) -> str | None:
        p = seq.params
        # min_tokens suppresses EOS/stop finishes (vLLM semantics);
        # length finishes still apply
        if not p.ignore_eos and len(seq.output_token_ids) > p.min_tokens:
            if tok == self.cfg.spec.eos_token_id or tok in p.stop_token_ids:
                return "stop"
        if len(seq.output_token_ids) >= p.max_tokens:
            return "length"
        if seq.num_tokens >= self.cfg.max_model_len:
            return "length"
        return None

    # convenience for tests / offline use
    def generate(self, prompts: list[list[int]], params: SamplingParams | None = None):
        ids = [self.add_request(p, params) for p in prompts]
        results = {i: [] for i in ids}
        while self.has_unfinished():
            for out in self.step():
                if out.request_id in results:
                    results[out.request_id].append(out.token_id)
        return [results[i] for i in ids]
