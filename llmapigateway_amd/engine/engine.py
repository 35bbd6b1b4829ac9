"""LLMEngine: continuous-batching inference engine for one MI355X (or CPU).

The engine replaces the reference's remote upstream as the thing a
"provider" resolves to. One engine owns one GPU-resident
model + paged KV cache; the gateway's fallback loop treats engine errors
(including injected faults) exactly like upstream HTTP errors — raised
before the first streamed byte so fallback can engage.

Scheduling policy: each step() is either a prefill forward (admitted
prompts advance by up to prefill_budget tokens — chunked for long
prompts, with decode rows riding along while a prompt is mid-chunk so
decode never stalls) or a decode forward over all running sequences
(hipGraph-replayed on GPU with deferred sampling: step s+1 enqueues
before step s's tokens reach the host). Prompt prefixes shared with
earlier requests are served from the content-addressed prefix cache and
only the suffix is prefilled. Decode out-of-block conditions preempt the
youngest running sequence back to the waiting queue (its KV is freed;
re-admission re-prefills prompt+generated, re-hitting the prefix cache).
"""

from __future__ import annotations

import itertools
import logging
import os
import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

import numpy as np
import torch

from .. import ops
from ..models.configs import ModelConfig, get_model_config
from ..models.llama import ForwardBatch, LlamaModel
from .kvcache import PagedKVCache

logger = logging.getLogger(__name__)

_req_counter = itertools.count()


class _DropRow:
    """Placeholder for sampled-but-not-delivered rows in logprob handling."""

    class params:  # noqa: N801 — mimics SamplingParams statics
        logprobs = False
        top_logprobs = 0

    state = "discarded"


_DROP = _DropRow()


@dataclass
class SamplingParams:
    temperature: float = 0.0
    top_p: float = 1.0
    top_k: int = 0
    presence_penalty: float = 0.0   # flat penalty on already-generated tokens
    frequency_penalty: float = 0.0  # per-occurrence penalty (output tokens)
    logit_bias: Dict[int, float] = field(default_factory=dict)
    max_tokens: int = 128
    seed: Optional[int] = None
    stop: List[str] = field(default_factory=list)
    ignore_eos: bool = False
    logprobs: bool = False
    top_logprobs: int = 0

    @classmethod
    def from_payload(cls, payload: dict, default_max_tokens: int = 256) -> "SamplingParams":
        stop = payload.get("stop") or []
        if isinstance(stop, str):
            stop = [stop]
        return cls(
            temperature=float(payload.get("temperature", 0.0) or 0.0),
            top_p=float(payload.get("top_p", 1.0) or 1.0),
            top_k=int(payload.get("top_k", 0) or 0),
            presence_penalty=float(payload.get("presence_penalty", 0.0) or 0.0),
            frequency_penalty=float(payload.get("frequency_penalty", 0.0) or 0.0),
            logit_bias={
                int(k): float(v)
                for k, v in (payload.get("logit_bias") or {}).items()
            },
            max_tokens=int(
                payload.get("max_completion_tokens")
                or payload.get("max_tokens")
                or default_max_tokens
            ),
            seed=payload.get("seed"),
            stop=[s for s in stop if isinstance(s, str)],
            logprobs=bool(payload.get("logprobs")),
            top_logprobs=max(0, min(20, int(payload.get("top_logprobs") or 0))),
        )


class EngineRequest:
    def __init__(
        self,
        prompt_ids: List[int],
        params: SamplingParams,
        on_token: Optional[Callable[["EngineRequest", int], None]] = None,
        on_finish: Optional[Callable[["EngineRequest"], None]] = None,
    ):
        self.id = f"req-{next(_req_counter)}"
        self.prompt_ids = list(prompt_ids)
        self.params = params
        self.out_ids: List[int] = []
        # per emitted token, when params.logprobs: (logprob, [(id, lp), ...])
        self.out_logprobs: List[tuple] = []
        self.text = ""  # incrementally decoded output (stop-string detection)
        self.block_table: List[int] = []
        self.state = "waiting"  # waiting | running | finished | failed
        self.finish_reason: Optional[str] = None
        self.error: Optional[str] = None
        self.on_token = on_token
        self.on_finish = on_finish
        self.bt_slot: Optional[int] = None  # engine block-table row (running)
        self.num_cached = 0  # prompt tokens served from the prefix cache
        self.prefill_pos = 0  # prompt tokens already written to the KV cache
        self.created = time.monotonic()
        self.enqueued: float = self.created
        self.prefill_start_time: Optional[float] = None
        self.first_token_time: Optional[float] = None
        self.finished_time: Optional[float] = None

    def timings(self) -> Dict[str, float]:
        """Per-phase timing (ms) for tracing/usage reporting."""
        out: Dict[str, float] = {}
        if self.prefill_start_time:
            out["queue_ms"] = round((self.prefill_start_time - self.created) * 1e3, 2)
        if self.first_token_time:
            out["ttft_ms"] = round((self.first_token_time - self.created) * 1e3, 2)
        if self.finished_time and self.first_token_time and len(self.out_ids) > 1:
            dt = self.finished_time - self.first_token_time
            if dt > 0:
                out["decode_tok_per_s"] = round((len(self.out_ids) - 1) / dt, 1)
        return out

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_ids) + len(self.out_ids)

    def __repr__(self) -> str:
        return f"<EngineRequest {self.id} {self.state} {len(self.prompt_ids)}+{len(self.out_ids)}>"


class LLMEngine:
    def __init__(
        self,
        model: str | ModelConfig = "llama-3-8b",
        device: torch.device | str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        block_size: int = 64,
        max_batch_size: int = 64,
        num_blocks: Optional[int] = None,
        hbm_fraction: float = 0.90,
        max_model_len: Optional[int] = None,
        seed: int = 0,
        use_hipgraph: Optional[bool] = None,
        tp_group: Optional[object] = None,
        tp_rank: int = 0,
        tp_size: int = 1,
        prefix_caching: bool = False,
        prefill_budget: int = 4096,
        kv_dtype: str = "auto",
        tokenizer: Optional[object] = None,
        admit_min_batch: Optional[int] = None,
        admit_max_wait: Optional[float] = None,
    ):
        full_config = get_model_config(model) if isinstance(model, str) else model
        self.full_config = full_config
        config = full_config.scaled_for_tp(tp_size) if tp_size > 1 else full_config
        self.config = config
        self.device = torch.device(device)
        self.dtype = dtype
        self.max_model_len = min(
            max_model_len or full_config.max_positions, full_config.max_positions
        )
        self.max_batch_size = max_batch_size
        if tokenizer is None:
            from .tokenizer import get_tokenizer

            # real BPE for production presets, byte stand-in for tiny ones
            tokenizer = get_tokenizer("auto", full_config.vocab_size)
        self.tokenizer = tokenizer  # stop-string decode + EOS detection
        self._eos_np = np.asarray(sorted(tokenizer.eos_ids), dtype=np.int64)

        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        self.model = LlamaModel(
            config, device=self.device, dtype=dtype, seed=seed,
            tp_group=tp_group, tp_rank=tp_rank, tp_size=tp_size,
            full_config=full_config,
        )
        if kv_dtype == "auto":
            kv_dtype = os.environ.get("LLMAPI_KV_DTYPE", "auto")
        if num_blocks is None:
            num_blocks = PagedKVCache.fit_num_blocks(
                config, block_size, self.device, dtype, hbm_fraction,
                kv_dtype=kv_dtype,
            )
        self.kv = PagedKVCache(
            config, num_blocks, block_size, self.device, dtype,
            prefix_caching=prefix_caching, kv_dtype=kv_dtype,
        )
        self.prefix_caching = prefix_caching
        if prefill_budget == 4096:  # default: allow env tuning
            prefill_budget = int(os.environ.get("LLMAPI_PREFILL_BUDGET", "4096"))
        self.prefill_budget = prefill_budget
        if admit_min_batch is None:
            admit_min_batch = int(os.environ.get("LLMAPI_ADMIT_MIN", "64"))
        self.admit_min_batch = max(1, admit_min_batch)
        if admit_max_wait is None:
            admit_max_wait = float(os.environ.get("LLMAPI_ADMIT_WAIT", "0.1"))
        self.admit_max_wait = admit_max_wait

        # hipGraph-captured decode (GPU only; TP group ops are capturable
        # with RCCL but kept off by default under TP until validated)
        if use_hipgraph is None:
            use_hipgraph = (
                self.device.type == "cuda"
                and tp_size == 1
                and not os.environ.get("LLMAPI_NO_HIPGRAPH")
            )
        self.graph_runner = None
        if use_hipgraph and self.device.type == "cuda":
            from .graph_runner import DecodeGraphRunner

            self.graph_runner = DecodeGraphRunner(
                self.model,
                self.kv.k_caches,
                self.kv.v_caches,
                max_batch=max_batch_size,
                max_blocks=(self.max_model_len + block_size - 1) // block_size,
                k_scales=self.kv.k_scales if self.kv.fp8 else None,
                v_scales=self.kv.v_scales if self.kv.fp8 else None,
            )

        # deferred sampling (GPU): the sampled-token fetch of decode step s
        # resolves while step s+1 is already enqueued — tokens feed the
        # next step device-side, host delivery runs one step late. Flushed
        # at every boundary: prefill, preemption, length cap, idle.
        self.async_sampling = self.device.type == "cuda"
        self._pending: Optional[tuple] = None  # (reqs, n, event, want_lp)
        self._pend_tokens_dev: Optional[torch.Tensor] = None
        self._pend_pinned: Optional[torch.Tensor] = None
        self._lp_pinned = None
        self._topv_pinned = None
        self._topi_pinned = None
        if self.async_sampling:
            self._pend_pinned = torch.empty(
                max_batch_size, dtype=torch.long, pin_memory=True
            )
            self._lp_pinned = torch.empty(
                max_batch_size, dtype=torch.float32, pin_memory=True
            )
            self._topv_pinned = torch.empty(
                max_batch_size, 20, dtype=torch.float32, pin_memory=True
            )
            self._topi_pinned = torch.empty(
                max_batch_size, 20, dtype=torch.long, pin_memory=True
            )

        # host-side persistent block tables: one stable row per admitted
        # request, updated only when a block is appended — the decode loop
        # gathers rows with one vectorized numpy fancy-index instead of
        # rebuilding 256 small tensors per step
        maxb = (self.max_model_len + block_size - 1) // block_size
        self._bt_np = np.zeros((max_batch_size, maxb), dtype=np.int32)
        self._slot_pool = list(range(max_batch_size - 1, -1, -1))
        # vectorized per-slot bookkeeping (the python per-request loops in
        # the decode hot path measured ~0.4 ms/step at b256): committed
        # token count, last committed token, absolute caps, block count
        self._slot_ntok = np.zeros(max_batch_size, dtype=np.int64)
        self._slot_last = np.zeros(max_batch_size, dtype=np.int64)
        self._slot_maxt = np.zeros(max_batch_size, dtype=np.int64)  # prompt+max_tokens
        self._slot_nblk = np.zeros(max_batch_size, dtype=np.int32)
        # bit0 = ignore_eos, bit1 = has stop strings
        self._slot_flags = np.zeros(max_batch_size, dtype=np.int8)

        # optional batched token-event sink (set by the gateway registry):
        # called under the engine lock with [(req, token), ...] per step
        self.batch_notifier: Optional[Callable[[list], None]] = None

        self._gen = torch.Generator(device=self.device).manual_seed(seed ^ 0x5EED)
        # block tables of requests aborted while a forward may be in
        # flight: freed at the next step() top (stream-ordered after the
        # in-flight KV writes) instead of immediately — an immediate free
        # could hand the blocks to a newly admitted request while the old
        # forward still writes into them
        self._deferred_free: List[List[int]] = []
        self._lock = threading.Lock()
        self._work = threading.Condition(self._lock)
        self.waiting: deque[EngineRequest] = deque()
        self.prefilling: List[EngineRequest] = []  # chunked prefill in progress
        self.prefill_budget = prefill_budget
        self.running: List[EngineRequest] = []
        self.stats: Dict[str, float] = {
            "requests": 0,
            "finished": 0,
            "failed": 0,
            "prefill_tokens": 0,
            "decode_tokens": 0,
            "steps": 0,
            # step-kind anatomy (serving-regime diagnosis)
            "prefill_steps": 0,
            "mixed_steps": 0,
            "decode_steps": 0,
            "graph_steps": 0,
            "flushes": 0,
            "prefill_s": 0.0,
            "decode_s": 0.0,
        }

    # ---- request intake ----
    def add_request(self, req: EngineRequest) -> EngineRequest:
        if len(req.prompt_ids) >= self.max_model_len:
            raise ValueError(
                f"Prompt of {len(req.prompt_ids)} tokens exceeds max_model_len={self.max_model_len}"
            )
        with self._work:
            req.enqueued = time.monotonic()
            self.waiting.append(req)
            self.stats["requests"] += 1
            self._work.notify_all()
        return req

    def abort_request(self, req: EngineRequest) -> None:
        with self._work:
            if req in self.waiting:
                self.waiting.remove(req)
                self._finish(req, "aborted")
            else:
                # running/prefilling: a forward may be mid-flight on the
                # engine thread — defer the KV free to the next step top
                self._finish(req, "aborted", defer_free=True)

    def has_work(self) -> bool:
        with self._lock:
            return bool(self.waiting or self.prefilling or self.running)

    def wait_for_work(self, timeout: float = 0.2) -> bool:
        with self._work:
            if self.waiting or self.prefilling or self.running:
                return True
            self._work.wait(timeout)
            return bool(self.waiting or self.prefilling or self.running)

    # ---- scheduling ----
    def _admit(self) -> List[EngineRequest]:
        """Move waiting requests into the prefilling set (blocks for the
        whole prompt are allocated up front; KV fills chunk by chunk).

        Admission batching: while the engine is at least half-loaded with
        decodes, hold a trickle of arrivals back (up to admit_min_batch
        or admit_max_wait, whichever first) so open-loop traffic
        produces a few large prefill steps instead of many small ones.
        Measured at saturation (profiles/r02_serving_notes.md): equal
        throughput, p50 TTFT 445->301 ms, p95 758->538. A lightly-loaded
        or idle engine always admits immediately (no latency tax at low
        traffic)."""
        if (
            self.waiting
            and len(self.running) >= self.max_batch_size // 2
            and len(self.waiting) < self.admit_min_batch
            and (time.monotonic() - self.waiting[0].enqueued) < self.admit_max_wait
        ):
            return []
        admitted: List[EngineRequest] = []
        while (
            self.waiting
            and len(self.running) + len(self.prefilling) + len(admitted)
            < self.max_batch_size
        ):
            req = self.waiting[0]
            need = len(req.prompt_ids)
            if not self.kv.manager.can_allocate(need + 1):
                break
            self.waiting.popleft()
            if self.prefix_caching:
                req.block_table, req.num_cached = (
                    self.kv.manager.allocate_with_prefix(req.prompt_ids)
                )
            else:
                req.block_table = self.kv.manager.allocate(need)
                req.num_cached = 0
            req.state = "running"
            req.prefill_start_time = time.monotonic()
            req.prefill_pos = req.num_cached
            req.bt_slot = self._slot_pool.pop()
            self._bt_np[req.bt_slot, : len(req.block_table)] = req.block_table
            self._slot_ntok[req.bt_slot] = len(req.prompt_ids)
            self._slot_last[req.bt_slot] = req.prompt_ids[-1]
            self._slot_maxt[req.bt_slot] = len(req.prompt_ids) + req.params.max_tokens
            self._slot_nblk[req.bt_slot] = len(req.block_table)
            self._slot_flags[req.bt_slot] = (
                (1 if req.params.ignore_eos else 0)
                | (2 if req.params.stop else 0)
            )
            admitted.append(req)
        return admitted

    def _free_slot(self, req: EngineRequest) -> None:
        if req.bt_slot is not None:
            self._slot_pool.append(req.bt_slot)
            req.bt_slot = None

    def _preempt_youngest(self) -> bool:
        if not self.running:
            return False
        victim = self.running.pop()
        self.kv.manager.free(victim.block_table)
        victim.block_table = []
        self._free_slot(victim)
        victim.state = "waiting"
        # re-admission re-prefills prompt + generated so far
        victim.prompt_ids = victim.prompt_ids + victim.out_ids
        victim.out_ids = []
        self.waiting.appendleft(victim)
        logger.warning("Preempted %s (KV blocks exhausted)", victim.id)
        return True

    # ---- deferred-sampling flush ----
    def _flush_pending(self) -> None:
        if self._pending is None:
            return
        reqs, n, event, want_lp = self._pending
        self._pending = None
        self.stats["flushes"] += 1
        event.synchronize()
        tokens = self._pend_pinned[:n].tolist()
        with self._lock:
            if want_lp:
                self._attach_logprobs(
                    reqs,
                    self._lp_pinned[:n].tolist(),
                    self._topv_pinned[:n].tolist(),
                    self._topi_pinned[:n].tolist(),
                )
            self._deliver(reqs, tokens)

    # ---- the step ----
    def step(self) -> int:
        """Run one engine iteration. Returns number of tokens produced."""
        with self._lock:
            if self._deferred_free:
                # previous step's forward has been issued; aborted
                # requests' blocks are now safe to reuse
                for bt in self._deferred_free:
                    self.kv.manager.free(bt)
                self._deferred_free.clear()
            self.prefilling.extend(self._admit())
            has_prefill = bool(self.prefilling)
        try:
            if has_prefill:
                self._flush_pending()  # running set is about to change
                t0 = time.monotonic()
                produced = self._prefill_step()
                self.stats["prefill_s"] += time.monotonic() - t0
            else:
                with self._lock:
                    idle = not self.running
                if idle:
                    self._flush_pending()  # deliver the final in-flight step
                    return 0
                t0 = time.monotonic()
                produced = self._decode_step()
                self.stats["decode_s"] += time.monotonic() - t0
        except Exception as e:
            logger.exception("engine step failed")
            with self._lock:
                for req in list(self.running) + list(self.prefilling):
                    req.error = f"engine error: {e}"
                    self._finish(req, "error")
                self.running.clear()
                self.prefilling.clear()
            raise
        self.stats["steps"] += 1
        return produced

    def _prefill_step(self) -> int:
        """One forward over up to prefill_budget prompt tokens of the
        prefilling set PLUS one decode row for every running sequence (a
        "mixed" step): decode never stalls behind prefill. Each prefill
        chunk attends to its previously written context (prefix-cache hits
        and earlier chunks alike) through the paged cache; a request whose
        chunk reaches the end of its prompt samples its first token and
        joins the running set. Deferred-sampling state is always flushed
        before this step, so decode inputs are host-visible."""
        device = self.device
        bs = self.kv.block_size

        with self._lock:
            budget = self.prefill_budget
            work: List[tuple] = []  # (req, start, end, final)
            for req in self.prefilling:
                if budget <= 0:
                    break
                L = len(req.prompt_ids)
                chunk = min(L - req.prefill_pos, budget)
                work.append((req, req.prefill_pos, req.prefill_pos + chunk,
                             req.prefill_pos + chunk == L))
                budget -= chunk
            # decode rows ride along whenever any are running: a prefill
            # step without them stalls the whole decode batch for its
            # duration, which dominates under continuous (open-loop)
            # arrivals. LLMAPI_MIXED_RIDE=chunking restores the round-1
            # gate (ride only while a prompt is mid-chunk) for A/B.
            ride = os.environ.get("LLMAPI_MIXED_RIDE", "chunking")
            chunking = ride == "always" or any(
                (start > req.num_cached) or (not final)
                for (req, start, end, final) in work
            )
            dec_reqs: List[EngineRequest] = []
            if chunking:
                i = 0
                while i < len(self.running):
                    req = self.running[i]
                    try:
                        before = len(req.block_table)
                        self.kv.manager.extend(req.block_table, req.num_tokens, req.num_tokens + 1)
                        if len(req.block_table) != before:
                            self._bt_np[req.bt_slot, before : len(req.block_table)] = (
                                req.block_table[before:]
                            )
                            self._slot_nblk[req.bt_slot] = len(req.block_table)
                        i += 1
                    except RuntimeError:
                        if not self._preempt_youngest():
                            raise
                dec_reqs = list(self.running)
        if not work and not dec_reqs:
            return 0

        reqs = [w[0] for w in work]
        starts = [w[1] for w in work]
        lens = [w[2] - w[1] for w in work]
        cu = np.zeros(len(reqs) + 1, dtype=np.int32)
        np.cumsum(lens, out=cu[1:])
        token_ids = np.concatenate(
            [np.asarray(r.prompt_ids[s : s + L], dtype=np.int64)
             for r, s, L in zip(reqs, starts, lens)]
        )
        positions = np.concatenate(
            [np.arange(s, s + L, dtype=np.int64) for s, L in zip(starts, lens)]
        )
        slots = np.empty(int(cu[-1]), dtype=np.int64)
        for i, (req, s, L) in enumerate(zip(reqs, starts, lens)):
            p = np.arange(s, s + L, dtype=np.int64)
            bt = np.asarray(req.block_table, dtype=np.int64)
            slots[cu[i] : cu[i + 1]] = bt[p // bs] * bs + p % bs
        logits_idx = (cu[1:] - 1).astype(np.int64)

        cached_lens_t = None
        block_tables_t = None
        if any(starts):
            rows = np.fromiter((req.bt_slot for req in reqs), dtype=np.intp, count=len(reqs))
            block_tables_t = torch.from_numpy(self._bt_np[rows]).to(device)
            cached_lens_t = torch.from_numpy(
                np.asarray(starts, dtype=np.int32)
            ).to(device)

        # decode rows appended after the prefill rows (mixed step)
        Tp = int(cu[-1])
        nd = len(dec_reqs)
        dec_bt_t = dec_ctx_t = None
        if nd:
            d_tokens = np.fromiter(
                ((r.out_ids[-1] if r.out_ids else r.prompt_ids[-1]) for r in dec_reqs),
                dtype=np.int64, count=nd,
            )
            d_pos = np.fromiter((r.num_tokens - 1 for r in dec_reqs), dtype=np.int64, count=nd)
            d_rows = np.fromiter((r.bt_slot for r in dec_reqs), dtype=np.intp, count=nd)
            d_tables = self._bt_np[d_rows]
            d_slots = d_tables[np.arange(nd), d_pos // bs].astype(np.int64) * bs + d_pos % bs
            token_ids = np.concatenate([token_ids, d_tokens])
            positions = np.concatenate([positions, d_pos])
            slots = np.concatenate([slots, d_slots])
            logits_idx = np.concatenate([logits_idx, Tp + np.arange(nd, dtype=np.int64)])
            dec_bt_t = torch.from_numpy(d_tables).to(device)
            dec_ctx_t = torch.from_numpy((d_pos + 1).astype(np.int32)).to(device)

        tile_seq, tile_off = ops.build_prefill_tiles(lens, device)
        batch = ForwardBatch(
            kind="mixed" if nd else "prefill",
            token_ids=torch.from_numpy(token_ids).to(device),
            positions=torch.from_numpy(positions).to(device),
            slot_mapping=torch.from_numpy(slots).to(device),
            cu_seqlens=torch.from_numpy(cu).to(device),
            max_seqlen=max(lens),
            tile_seq=tile_seq,
            tile_off=tile_off,
            block_tables=block_tables_t,
            cached_lens=cached_lens_t,
            n_prefill_tokens=Tp,
            dec_block_tables=dec_bt_t,
            dec_context_lens=dec_ctx_t,
            logits_indices=torch.from_numpy(logits_idx).to(device),
        )
        self.stats["mixed_steps" if nd else "prefill_steps"] += 1
        logits = self.model.forward(
            batch, self.kv.k_caches, self.kv.v_caches,
            k_scales=self.kv.k_scales if self.kv.fp8 else None,
            v_scales=self.kv.v_scales if self.kv.fp8 else None,
        )
        sample_reqs = reqs + dec_reqs
        deliver_mask = [w[3] for w in work] + [True] * nd
        tokens = self._sample(logits, sample_reqs, deliver_mask)
        dec_tokens = tokens[len(reqs):]
        self.stats["prefill_tokens"] += Tp
        self.stats["decode_tokens"] += nd
        with self._lock:
            # requests aborted while the forward was in flight are gone
            # from self.prefilling and must not be revived (their blocks
            # sit in _deferred_free — never register them as prefixes)
            live = [
                (req, s, e, final)
                for (req, s, e, final) in work
                if req.state == "running"
            ]
            finals = [req for (req, s, e, final) in live if final]
            if self.prefix_caching:
                # KV for this chunk is now written (stream-ordered before
                # any later forward): make completed full prompt blocks
                # reusable. Under the lock — abort/_finish on gateway
                # threads mutate the same refcount tables via kv.manager.
                for req in finals:
                    self.kv.manager.register_prefix(req.prompt_ids, req.block_table)
            for req, s, e, final in live:
                req.prefill_pos = e
                if final:
                    self.prefilling.remove(req)
            final_tokens = [
                t for t, (req, _, _, final) in zip(tokens, work)
                if final and req.state == "running"
            ]
            if nd:
                self._deliver(dec_reqs, dec_tokens)
            self.running.extend(finals)
            self._deliver(finals, final_tokens)
        return len(finals) + nd

    def _decode_step(self) -> int:
        device = self.device
        bs = self.kv.block_size
        with self._lock:
            reqs_now = list(self.running)
        if not reqs_now:
            self._flush_pending()
            return 0

        # deferred-sampling bookkeeping: if the previous decode step's
        # tokens are still in flight, they belong to EXACTLY this request
        # list; any mismatch or limit-crossing request forces a flush
        if self._pending is not None:
            same = self._pending[0] == reqs_now
            if same:
                rows0 = np.fromiter(
                    (r.bt_slot for r in reqs_now), dtype=np.intp, count=len(reqs_now)
                )
                nt0 = self._slot_ntok[rows0]
                capped = bool(
                    ((nt0 + 1 >= self._slot_maxt[rows0])
                     | (nt0 + 2 > self.max_model_len)).any()
                )
            else:
                capped = False
            if not same or capped:
                self._flush_pending()
                with self._lock:
                    reqs_now = list(self.running)
                if not reqs_now:
                    return 0
        inflight = 1 if self._pending is not None else 0

        with self._lock:
            # ensure every running seq has a block for the incoming token:
            # vectorized need-check, python only for the rows that grow
            nrun = len(self.running)
            rows_all = np.fromiter(
                (r.bt_slot for r in self.running), dtype=np.intp, count=nrun
            )
            nt_all = self._slot_ntok[rows_all] + inflight
            need = (nt_all + bs) // bs  # blocks_needed(nt + 1)
            grow = np.nonzero(need > self._slot_nblk[rows_all])[0]
            i = -1
            for gi in grow.tolist():
                req = self.running[gi] if gi < len(self.running) else None
                if req is None or req.bt_slot is None:
                    continue
                nt = int(self._slot_ntok[req.bt_slot]) + inflight
                try:
                    before = len(req.block_table)
                    self.kv.manager.extend(req.block_table, nt, nt + 1)
                    if len(req.block_table) != before:
                        self._bt_np[req.bt_slot, before : len(req.block_table)] = (
                            req.block_table[before:]
                        )
                        self._slot_nblk[req.bt_slot] = len(req.block_table)
                except RuntimeError:
                    if inflight:
                        i = 0  # flush below, then retry the whole step
                        break
                    # rare pressure path: fall back to the full python
                    # loop with preemption (unchanged semantics)
                    j = 0
                    while j < len(self.running):
                        r2 = self.running[j]
                        nt2 = r2.num_tokens + inflight
                        try:
                            b2 = len(r2.block_table)
                            self.kv.manager.extend(r2.block_table, nt2, nt2 + 1)
                            if len(r2.block_table) != b2:
                                self._bt_np[r2.bt_slot, b2 : len(r2.block_table)] = (
                                    r2.block_table[b2:]
                                )
                                self._slot_nblk[r2.bt_slot] = len(r2.block_table)
                            j += 1
                        except RuntimeError:
                            if not self._preempt_youngest():
                                raise
                    break
            reqs = list(self.running)
        if i != -1:  # needed preemption while tokens were in flight
            self._flush_pending()
            return self._decode_step()
        if not reqs:
            return 0

        n = len(reqs)
        slot_rows = np.fromiter((req.bt_slot for req in reqs), dtype=np.intp, count=n)
        pos = self._slot_ntok[slot_rows] + (inflight - 1)
        tables_np = self._bt_np[slot_rows]  # [n, maxb] vectorized gather
        blk = pos // bs
        slots = tables_np[np.arange(n), blk].astype(np.int64) * bs + pos % bs
        ctx = (pos + 1).astype(np.int32)
        if inflight:
            last_tokens = self._pend_tokens_dev  # device int64 [n], aligned
        else:
            last_tokens = self._slot_last[slot_rows].copy()

        self.stats["decode_steps"] += 1
        logits = None
        if self.graph_runner is not None:
            try:
                logits = self.graph_runner.run(last_tokens, pos, slots, tables_np, ctx)
                self.stats["graph_steps"] += 1
            except Exception:
                logger.exception("hipGraph decode failed; falling back to eager")
                self.graph_runner = None
        if logits is None:
            tok_t = (
                last_tokens
                if isinstance(last_tokens, torch.Tensor)
                else torch.from_numpy(last_tokens).to(device)
            )
            batch = ForwardBatch(
                kind="decode",
                token_ids=tok_t,
                positions=torch.from_numpy(pos).to(device),
                slot_mapping=torch.from_numpy(slots).to(device),
                block_tables=torch.from_numpy(tables_np).to(device),
                context_lens=torch.from_numpy(ctx).to(device),
                logits_indices=None,
            )
            logits = self.model.forward(
                batch, self.kv.k_caches, self.kv.v_caches,
                k_scales=self.kv.k_scales if self.kv.fp8 else None,
                v_scales=self.kv.v_scales if self.kv.fp8 else None,
            )

        tokens_dev = self._sample_dev(
            logits, reqs, noise_pos=[len(r.out_ids) + inflight for r in reqs]
        )
        lp = self._logprobs_dev(logits, tokens_dev, reqs)
        self.stats["decode_tokens"] += n
        if self.async_sampling:
            # previous pending was either flushed or belongs to these same
            # reqs and is already delivered-by-flush above when needed
            self._flush_pending()
            self._pend_pinned[:n].copy_(tokens_dev, non_blocking=True)
            if lp is not None:
                chosen, tv, ti = lp
                k = tv.shape[1]
                self._lp_pinned[:n].copy_(chosen, non_blocking=True)
                self._topv_pinned[:n, :k].copy_(tv, non_blocking=True)
                self._topi_pinned[:n, :k].copy_(ti, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
            self._pending = (reqs, n, ev, lp is not None)
            self._pend_tokens_dev = tokens_dev
            return n
        tokens = tokens_dev.tolist()
        with self._lock:
            if lp is not None:
                self._attach_logprobs(reqs, *(t.tolist() for t in lp))
            self._deliver(reqs, tokens)
        return n

    # ---- sampling ----
    def _apply_penalties(self, logits: torch.Tensor, reqs: List[EngineRequest]) -> torch.Tensor:
        """OpenAI presence/frequency penalties over the OUTPUT tokens so
        far (with deferred sampling the in-flight token lags one step —
        an accepted approximation). Returns logits, cloned only if any
        request uses penalties."""
        # fast path: the common all-greedy/plain batch skips the per-row
        # python loop entirely (this runs every decode step at b256)
        if not any(
            r.params.presence_penalty != 0.0
            or r.params.frequency_penalty != 0.0
            or r.params.logit_bias
            for r in reqs
        ):
            return logits
        rows, idxs, vals = [], [], []
        from collections import Counter

        for i, r in enumerate(reqs):
            pp, fp = r.params.presence_penalty, r.params.frequency_penalty
            if (pp != 0.0 or fp != 0.0) and r.out_ids:
                for tok, c in Counter(r.out_ids).items():
                    rows.append(i)
                    idxs.append(tok)
                    vals.append(pp + fp * c)
            for tok, bias in r.params.logit_bias.items():
                rows.append(i)
                idxs.append(tok)
                vals.append(-bias)  # subtracted below: negate to ADD the bias
        if not rows:
            return logits
        logits = logits.clone()
        dev = logits.device
        logits[
            torch.tensor(rows, dtype=torch.long, device=dev),
            torch.tensor(idxs, dtype=torch.long, device=dev),
        ] -= torch.tensor(vals, dtype=logits.dtype, device=dev)
        return logits

    def _sample_dev(
        self,
        logits: torch.Tensor,
        reqs: List[EngineRequest],
        noise_pos: Optional[List[int]] = None,
    ) -> torch.Tensor:
        """Sample next tokens; returns an int64 device tensor (no host sync).

        ``noise_pos[i]`` is the output position being sampled for request i —
        callers on the deferred-sampling decode path pass
        ``len(out_ids) + inflight`` because the previous step's token has not
        reached ``out_ids`` yet (keying on bare len(out_ids) would reuse the
        same seeded noise vector for two consecutive tokens after every
        flush boundary)."""
        logits = self._apply_penalties(logits, reqs)
        any_temp = any(r.params.temperature > 0 for r in reqs)
        temps = torch.tensor(
            [r.params.temperature for r in reqs], dtype=torch.float32, device=logits.device
        )
        filtered = logits
        if any_temp:
            filtered = self._apply_topk_topp(logits.clone(), reqs)
            noise = torch.rand(
                logits.shape, generator=self._gen, device=logits.device, dtype=torch.float32
            )
            # per-request seeds (OpenAI `seed`): deterministic per output
            # position, independent of batch composition
            seeded = any(r.params.seed is not None for r in reqs)
            for i, r in enumerate(reqs) if seeded else ():
                if r.params.seed is not None and r.params.temperature > 0:
                    pos = noise_pos[i] if noise_pos is not None else len(r.out_ids)
                    g = torch.Generator(device=logits.device).manual_seed(
                        (int(r.params.seed) << 20) ^ pos
                    )
                    noise[i] = torch.rand(
                        logits.shape[-1], generator=g, device=logits.device,
                        dtype=torch.float32,
                    )
        else:
            noise = None
        return ops.sample(filtered, temps, noise)

    def _sample(
        self,
        logits: torch.Tensor,
        reqs: List[EngineRequest],
        deliver_mask: Optional[List[bool]] = None,
    ) -> List[int]:
        """Sample + attach logprobs for the rows whose token will actually
        be delivered (non-final chunked-prefill rows sample a discarded
        token and must not accumulate logprobs)."""
        tokens = self._sample_dev(logits, reqs)
        lp_reqs = (
            reqs if deliver_mask is None
            else [r if m else _DROP for r, m in zip(reqs, deliver_mask)]
        )
        lp = self._logprobs_dev(logits, tokens, lp_reqs)
        toks = tokens.tolist()
        if lp is not None:
            self._attach_logprobs(lp_reqs, *(t.tolist() for t in lp))
        return toks

    def _logprobs_dev(self, logits, tokens, reqs):
        """(chosen_lp [B], top_v [B,K], top_i [B,K]) device tensors, or
        None when no request in the batch asked for logprobs. Raw-model
        log-softmax (pre-temperature), OpenAI semantics."""
        if not any(r.params.logprobs for r in reqs):
            return None
        lf = logits.float()
        lse = torch.logsumexp(lf, dim=-1)
        chosen = lf.gather(1, tokens.unsqueeze(1)).squeeze(1) - lse
        k = max([r.params.top_logprobs for r in reqs] + [0])
        if k > 0:
            tv, ti = torch.topk(lf, k, dim=-1)
            tv = tv - lse.unsqueeze(1)
        else:
            B = lf.shape[0]
            tv = torch.empty(B, 0, device=lf.device)
            ti = torch.empty(B, 0, dtype=torch.long, device=lf.device)
        return chosen, tv, ti

    def _attach_logprobs(self, reqs, chosen, top_v, top_i) -> None:
        for i, r in enumerate(reqs):
            if not r.params.logprobs or r.state != "running":
                continue
            k = r.params.top_logprobs
            tops = list(zip(top_i[i][:k], top_v[i][:k])) if k else []
            r.out_logprobs.append((chosen[i], tops))

    def _apply_topk_topp(self, logits: torch.Tensor, reqs: List[EngineRequest]) -> torch.Tensor:
        """In-place top-k/top-p filtering over the whole batch: the GPU
        path is the histogram-threshold kernel (ops/csrc/sampling.hip —
        no full-vocab sort, no per-row host loop); CPU uses the batched
        torch reference. Rows that need no filtering are untouched."""
        any_filter = False
        topp = []
        topk = []
        for r in reqs:
            active = r.params.temperature > 0 and (
                r.params.top_k > 0 or r.params.top_p < 1.0
            )
            any_filter = any_filter or active
            topp.append(r.params.top_p if active else 1.0)
            topk.append(r.params.top_k if active else 0)
        if not any_filter:
            return logits
        dev = logits.device
        return ops.topk_topp_filter(
            logits,
            torch.tensor(topp, dtype=torch.float32, device=dev),
            torch.tensor(topk, dtype=torch.int32, device=dev),
        )

    # ---- delivery / lifecycle (call with lock held) ----
    def _deliver(self, reqs: List[EngineRequest], tokens: List[int]) -> None:
        now = time.monotonic()
        # pass 1: record tokens; batch-notify consumers in ONE cross-thread
        # event (the gateway registers batch_notifier — per-token
        # call_soon_threadsafe wakeups saturate the event loop ~10K/s)
        events = [] if self.batch_notifier is not None else None
        rows: List[int] = []
        row_toks: List[int] = []
        for req, tok in zip(reqs, tokens):
            if req.state != "running":
                continue
            req.out_ids.append(int(tok))
            if req.bt_slot is not None:
                rows.append(req.bt_slot)
                row_toks.append(int(tok))
            if req.params.stop:
                req.text += self.tokenizer.decode([int(tok)])
            if req.first_token_time is None:
                req.first_token_time = now
            if req.on_token is not None:
                try:
                    req.on_token(req, int(tok))
                except Exception:
                    logger.exception("on_token callback failed for %s", req.id)
            elif events is not None:
                events.append((req, int(tok)))
        if rows:
            r = np.asarray(rows, dtype=np.intp)
            self._slot_ntok[r] += 1
            self._slot_last[r] = row_toks
        if events:
            try:
                self.batch_notifier(events)
            except Exception:
                logger.exception("batch_notifier failed")
        # pass 2: finish checks. The vectorized prefilter covers the
        # common finish causes (length caps, EOS for non-ignore_eos);
        # requests with stop strings always take the python path.
        if rows and len(reqs) == len(rows):
            nt = self._slot_ntok[r]
            flags = self._slot_flags[r]
            tarr = np.asarray(row_toks, dtype=np.int64)
            needs_py = (
                (nt >= self._slot_maxt[r])
                | (nt >= self.max_model_len)
                | (np.isin(tarr, self._eos_np) & ((flags & 1) == 0))
                | ((flags & 2) != 0)
            )
            idxs = np.nonzero(needs_py)[0]
            if idxs.size == 0:
                return
            for i in idxs.tolist():
                req, tok = reqs[i], tokens[i]
                if req.state != "running":
                    continue
                reason = self._finish_reason(req, int(tok))
                if reason:
                    self.running.remove(req)
                    self._finish(req, reason)
            return
        for req, tok in zip(reqs, tokens):
            if req.state != "running":
                continue
            reason = self._finish_reason(req, int(tok))
            if reason:
                self.running.remove(req)
                self._finish(req, reason)

    def _finish_reason(self, req: EngineRequest, tok: int) -> Optional[str]:
        if not req.params.ignore_eos and tok in self.tokenizer.eos_ids:
            return "stop"
        if req.params.stop and any(s in req.text for s in req.params.stop):
            return "stop"
        if len(req.out_ids) >= req.params.max_tokens:
            return "length"
        if req.num_tokens >= self.max_model_len:
            return "length"
        return None

    def _finish(self, req: EngineRequest, reason: str, defer_free: bool = False) -> None:
        if req.state in ("finished", "failed"):
            return
        if req in self.running:  # e.g. abort of a running request
            self.running.remove(req)
        if req in self.prefilling:
            self.prefilling.remove(req)
        req.state = "failed" if reason == "error" else "finished"
        req.finish_reason = reason
        req.finished_time = time.monotonic()
        self.stats["finished" if req.state == "finished" else "failed"] += 1
        if req.block_table:
            if defer_free:
                self._deferred_free.append(req.block_table)
            else:
                self.kv.manager.free(req.block_table)
            req.block_table = []
        self._free_slot(req)
        if req.on_finish is not None:
            try:
                req.on_finish(req)
            except Exception:
                logger.exception("on_finish callback failed for %s", req.id)

    # ---- convenience driver (tests / bench) ----
    def generate(self, prompt_ids: List[int], params: SamplingParams) -> EngineRequest:
        req = EngineRequest(prompt_ids, params)
        self.add_request(req)
        while req.state in ("waiting", "running"):
            if self.step() == 0 and req.state == "waiting":
                raise RuntimeError("engine made no progress")
        return req
