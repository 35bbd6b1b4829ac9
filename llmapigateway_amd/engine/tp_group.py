"""Tensor-parallel engine worker group: one provider entry -> TP=N over xGMI.

Serves BASELINE configs[4] (Llama-3-70B TP=8 as a single provider entry in
models_fallback_rules.json). The reference has no equivalent — its only
"distribution" is HTTP to remote providers — so this is the
MI355X-native design:

- N worker processes, one per GPU, joined by a torch.distributed group
  (backend "nccl" == RCCL over xGMI on ROCm; "gloo" on CPU for tests).
- Every rank runs an identical LLMEngine (tp_rank=r) in LOCKSTEP: rank 0
  drains the gateway's command queue, broadcasts the command batch, every
  rank applies it identically and calls engine.step(). Scheduling is a pure
  function of the (identical) request stream, and sampled tokens are
  identical on all ranks because the final hidden state is identical after
  the last row-parallel all-reduce (full, unsharded lm_head) and every rank
  seeds the same sampling generator — so no per-token broadcast is needed:
  the only hot-path collectives are the model's own 2 all-reduces/layer.
- Rank 0 streams tokens back over an mp.Queue; TPEngineClient mirrors them
  onto local EngineRequest objects so EngineRegistry.make_request can treat
  a TP group exactly like a single-GPU engine (duck-typed: add_request /
  abort_request / max_model_len / stats).

Failure semantics keep the gateway's first-chunk contract
(/root/reference/llm_gateway_core/services/request_handler.py:67-100): a
dead worker group fails all pending requests with an error *before* any
bytes reach the client, so the chat fallback loop can engage.
"""

from __future__ import annotations

import logging
import multiprocessing as mp
import os
import queue as queue_mod
import socket
import threading
import time
from typing import Dict, List, Optional

logger = logging.getLogger(__name__)


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_main(
    rank: int,
    world: int,
    port: int,
    model: str,
    dtype_name: str,
    max_batch_size: int,
    kv_block_size: int,
    num_blocks: Optional[int],
    max_model_len: Optional[int],
    seed: int,
    inbox,
    outbox,
) -> None:
    # fresh interpreter (spawn): do the heavy imports here
    import torch
    import torch.distributed as dist

    from .engine import EngineRequest, LLMEngine, SamplingParams

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    have_cuda = torch.cuda.is_available()
    backend = "nccl" if have_cuda else "gloo"
    dist.init_process_group(backend, rank=rank, world_size=world)
    device = f"cuda:{rank}" if have_cuda else "cpu"
    if have_cuda:
        torch.cuda.set_device(rank)
    dtype = getattr(torch, dtype_name) if have_cuda else torch.float32

    engine = LLMEngine(
        model=model,
        device=device,
        dtype=dtype,
        block_size=kv_block_size,
        max_batch_size=max_batch_size,
        num_blocks=num_blocks if num_blocks is not None else (None if have_cuda else 256),
        max_model_len=max_model_len,
        seed=seed,  # same seed on every rank: identical sampling decisions
        # hipGraph decode under TP: RCCL collectives are capturable on
        # ROCm; a capture failure falls back to eager (engine.step's
        # graph_runner try/except). LLMAPI_NO_TP_HIPGRAPH=1 opts out.
        use_hipgraph=(
            None if have_cuda and not os.environ.get("LLMAPI_NO_TP_HIPGRAPH")
            else False
        ),
        tp_group=dist.group.WORLD,
        tp_rank=rank,
        tp_size=world,
        # lockstep determinism: admission must be a pure function of the
        # broadcast command stream — the wall-clock admission hold-back
        # would diverge across ranks
        admit_min_batch=1,
        admit_max_wait=0.0,
    )
    reqs: Dict[str, EngineRequest] = {}

    if rank == 0:
        outbox.put(("ready", None, None))

    def on_token(req: EngineRequest, tok: int) -> None:
        if rank == 0:
            if req.params.logprobs and req.out_logprobs:
                outbox.put(("tokenlp", req.id, (tok, req.out_logprobs[-1])))
            else:
                outbox.put(("token", req.id, tok))

    def on_finish(req: EngineRequest) -> None:
        reqs.pop(req.id, None)
        if rank == 0:
            outbox.put(
                (
                    "finish",
                    req.id,
                    {
                        "state": req.state,
                        "finish_reason": req.finish_reason,
                        "error": req.error,
                        "n_out": len(req.out_ids),
                    },
                )
            )

    # Lockstep command channel: rank 0 drains the gateway queue and the
    # group agrees on each step's command batch through ONE fixed-size
    # int32 broadcast (the pickled blob follows only when non-empty) —
    # the round-1 per-step broadcast_object_list cost 2 object
    # collectives + pickle on EVERY decode step. Checked only every
    # CMD_STRIDE busy steps (deterministic across ranks: the step counter
    # advances in lockstep), so steady-state decode pays one tiny
    # broadcast per stride.
    import pickle

    CMD_STRIDE = 4
    flag = torch.zeros(1, dtype=torch.int32, device=device)
    since_check = CMD_STRIDE  # first loop always checks
    t_bcast = 0.0  # busy-path time inside the command collective
    t_step = 0.0
    stop = False
    while not stop:
        cmds: List[tuple] = []
        # `since_check` advances only on lockstep busy steps and resets at
        # every collective, so all ranks agree on which iterations check
        # (has_work() is identical across ranks by the lockstep invariant)
        if since_check >= CMD_STRIDE or not engine.has_work():
            blob = b""
            if rank == 0:
                raw: List[tuple] = []
                busy = engine.has_work()
                try:
                    raw.append(inbox.get(timeout=0.0 if busy else 0.05))
                except queue_mod.Empty:
                    pass
                while True:
                    try:
                        raw.append(inbox.get_nowait())
                    except queue_mod.Empty:
                        break
                if not busy and not raw:
                    continue  # idle: workers are parked in the broadcast
                blob = pickle.dumps(raw) if raw else b""
                flag[0] = len(blob)
            tb0 = time.monotonic() if engine.has_work() else None
            dist.broadcast(flag, src=0)
            since_check = 0
            n = int(flag.item())
            if tb0 is not None:
                t_bcast += time.monotonic() - tb0
            if n:
                buf = torch.empty(n, dtype=torch.uint8, device=device)
                if rank == 0:
                    buf.copy_(torch.frombuffer(bytearray(blob), dtype=torch.uint8))
                dist.broadcast(buf, src=0)
                cmds = pickle.loads(bytes(buf.cpu().numpy().tobytes()))
        else:
            since_check += 1
        for cmd in cmds:
            kind = cmd[0]
            if kind == "add":
                _, rid, prompt_ids, params = cmd
                req = EngineRequest(
                    prompt_ids,
                    SamplingParams(**params),
                    on_token=on_token,
                    on_finish=on_finish,
                )
                req.id = rid  # gateway-assigned id, identical on all ranks
                reqs[rid] = req
                try:
                    engine.add_request(req)
                except Exception as e:  # identical on all ranks (same check)
                    req.error = str(e)
                    req.state = "failed"
                    reqs.pop(rid, None)
                    if rank == 0:
                        outbox.put(
                            ("finish", rid,
                             {"state": "failed", "finish_reason": "error",
                              "error": str(e), "n_out": 0})
                        )
            elif kind == "abort":
                req = reqs.get(cmd[1])
                if req is not None:
                    engine.abort_request(req)
            elif kind == "stop":
                stop = True
        if stop:
            break
        if engine.has_work():
            ts0 = time.monotonic()
            try:
                engine.step()
            except Exception:
                # step() already failed + reported the affected requests
                logger.exception("tp rank %d step error", rank)
            t_step += time.monotonic() - ts0

    if rank == 0:
        # lockstep overhead report (world-4 CPU test asserts bcast < 5%)
        outbox.put(("lockstep_stats", None, {"bcast_s": t_bcast, "step_s": t_step}))
    dist.destroy_process_group()


class TPEngineClient:
    """Gateway-side handle to a TP worker group (duck-types LLMEngine)."""

    def __init__(
        self,
        model: str,
        tp: int,
        dtype: str = "bfloat16",
        max_batch_size: int = 64,
        kv_block_size: int = 64,
        num_blocks: Optional[int] = None,
        max_model_len: Optional[int] = None,
        seed: int = 0,
        start_timeout: float = 600.0,
    ):
        from ..models.configs import get_model_config
        from .tokenizer import get_tokenizer

        self.full_config = get_model_config(model) if isinstance(model, str) else model
        self.tokenizer = get_tokenizer("auto", self.full_config.vocab_size)
        self.max_model_len = max_model_len or self.full_config.max_positions
        self.tp = tp
        ctx = mp.get_context("spawn")
        self.inbox = ctx.Queue()
        self.outbox = ctx.Queue()
        port = _free_port()
        self.procs = [
            ctx.Process(
                target=_worker_main,
                args=(
                    r, tp, port, model, dtype, max_batch_size, kv_block_size,
                    num_blocks, self.max_model_len, seed, self.inbox, self.outbox,
                ),
                daemon=True,
                name=f"tp-worker-{r}",
            )
            for r in range(tp)
        ]
        for p in self.procs:
            p.start()
        self._reqs: Dict[str, object] = {}
        self.lockstep_stats: Optional[dict] = None
        self._lock = threading.Lock()
        self._dead: Optional[str] = None
        self.stats: Dict[str, float] = {"requests": 0, "finished": 0, "failed": 0}
        # wait for rank0's ready (engine built on every rank)
        deadline = time.monotonic() + start_timeout
        while True:
            try:
                kind, _, _ = self.outbox.get(timeout=1.0)
                if kind == "ready":
                    break
            except queue_mod.Empty:
                if any(not p.is_alive() for p in self.procs):
                    self._terminate()
                    raise RuntimeError("TP worker died during startup")
                if time.monotonic() > deadline:
                    self._terminate()
                    raise RuntimeError("TP worker group startup timed out")
        self._reader = threading.Thread(target=self._drain, daemon=True, name="tp-reader")
        self._reader.start()
        self._watchdog = threading.Thread(target=self._watch, daemon=True, name="tp-watchdog")
        self._watchdog.start()

    # ---- LLMEngine duck-type surface used by EngineRegistry ----
    def add_request(self, req) -> None:
        if len(req.prompt_ids) >= self.max_model_len:
            raise ValueError(
                f"Prompt of {len(req.prompt_ids)} tokens exceeds max_model_len={self.max_model_len}"
            )
        if self._dead:
            raise RuntimeError(self._dead)
        with self._lock:
            self._reqs[req.id] = req
            self.stats["requests"] += 1
        req.state = "running"
        self.inbox.put(("add", req.id, req.prompt_ids, _params_dict(req.params)))

    def abort_request(self, req) -> None:
        self.inbox.put(("abort", req.id))
        self._finish_local(req, "aborted", None)

    # ---- internals ----
    def _drain(self) -> None:
        while True:
            try:
                kind, rid, value = self.outbox.get(timeout=0.5)
            except queue_mod.Empty:
                if self._dead:
                    return
                continue
            except (EOFError, OSError):
                return
            if kind == "lockstep_stats":
                self.lockstep_stats = value
                continue
            with self._lock:
                req = self._reqs.get(rid)
            if req is None:
                continue
            if kind in ("token", "tokenlp"):
                if kind == "tokenlp":
                    value, lp = value
                    if not hasattr(req, "out_logprobs"):
                        req.out_logprobs = []
                    req.out_logprobs.append(lp)
                req.out_ids.append(int(value))
                if req.first_token_time is None:
                    req.first_token_time = time.monotonic()
                if req.on_token is not None:
                    try:
                        req.on_token(req, int(value))
                    except Exception:
                        logger.exception("on_token failed for %s", rid)
            elif kind == "finish":
                reason = value.get("finish_reason") or "stop"
                req.error = value.get("error")
                state = value.get("state")
                self._finish_local(req, reason, state)

    def _finish_local(self, req, reason: str, state: Optional[str]) -> None:
        with self._lock:
            if self._reqs.pop(req.id, None) is None:
                return
            req.state = state or ("failed" if reason == "error" else "finished")
            req.finish_reason = reason
            req.finished_time = time.monotonic()
            self.stats["finished" if req.state == "finished" else "failed"] += 1
        if req.on_finish is not None:
            try:
                req.on_finish(req)
            except Exception:
                logger.exception("on_finish failed for %s", req.id)

    def _watch(self) -> None:
        while self._dead is None:
            for p in self.procs:
                if not p.is_alive() and p.exitcode not in (0, None):
                    self._dead = (
                        f"TP worker rank {p.name} died (exit {p.exitcode}); engine group down"
                    )
                    with self._lock:
                        pending = list(self._reqs.values())
                    for req in pending:
                        req.error = self._dead
                        self._finish_local(req, "error", "failed")
                    return
            time.sleep(0.5)

    def _terminate(self) -> None:
        for p in self.procs:
            if p.is_alive():
                p.terminate()
        for p in self.procs:
            p.join(timeout=5.0)

    def stop(self) -> None:
        self._dead = self._dead or "engine group stopped"
        try:
            self.inbox.put(("stop",))
        except Exception:
            pass
        # fail in-flight requests NOW (mirrors _watch): registry.prune()
        # calls stop() while clients may still be streaming — without this
        # they would hang until the first-token timeout
        with self._lock:
            pending = list(self._reqs.values())
        for req in pending:
            req.error = self._dead
            self._finish_local(req, "error", "failed")
        for p in self.procs:
            p.join(timeout=10.0)
        self._terminate()


def _params_dict(params) -> dict:
    return {
        "temperature": params.temperature,
        "top_p": params.top_p,
        "top_k": params.top_k,
        "presence_penalty": params.presence_penalty,
        "frequency_penalty": params.frequency_penalty,
        "logit_bias": dict(params.logit_bias),
        "max_tokens": params.max_tokens,
        "seed": params.seed,
        "stop": list(params.stop),
        "ignore_eos": params.ignore_eos,
        "logprobs": params.logprobs,
        "top_logprobs": params.top_logprobs,
    }
