"""EngineRegistry: local providers -> GPU-resident engines.

The registry is the local-backend half of the dispatch service (SURVEY.md
§2b): it keeps one LLMEngine per (model, device, tp) key, each driven by its
own daemon thread, and turns OpenAI chat payloads into engine requests with
the reference's make_llm_request semantics
(/root/reference/llm_gateway_core/services/request_handler.py:8): returns
(response, None) on success or (None, error_detail) on any failure that
happens *before the first streamed byte* — injected faults, engine build
errors, OOM — so the chat fallback loop can move to the next model.

Failure injection (BASELINE configs[2]) is first-class: per-provider
fail_rate / fail_requests from the EngineSpec, overridable at runtime via
set_failure() (exposed by the admin API route).
"""

from __future__ import annotations

import asyncio
import json
import logging
import threading
import time
import uuid
from typing import Any, Dict, List, Optional, Tuple

import torch
from fastapi.responses import StreamingResponse

from ..config.loader import EngineSpec
from ..config.settings import Settings
from .engine import EngineRequest, LLMEngine, SamplingParams
from .tokenizer import ByteTokenizer

logger = logging.getLogger(__name__)

STREAM_HEADERS = {"X-Accel-Buffering": "no"}


class _EngineHandle:
    def __init__(self, engine: LLMEngine, name: str):
        self.engine = engine
        self.name = name
        self.stop_event = threading.Event()
        self.thread = threading.Thread(target=self._loop, name=f"engine-{name}", daemon=True)
        self.thread.start()

    def _loop(self) -> None:
        logger.info("engine loop started: %s", self.name)
        while not self.stop_event.is_set():
            try:
                if self.engine.wait_for_work(timeout=0.05):
                    self.engine.step()
            except Exception:
                # step() already failed the affected requests; keep serving
                logger.exception("engine %s step error", self.name)
                time.sleep(0.01)

    def stop(self) -> None:
        self.stop_event.set()
        self.thread.join(timeout=5.0)


class _TPGroupHandle:
    """Handle for a multi-process TP engine group (no local step loop —
    the lockstep loop runs inside the worker processes)."""

    def __init__(self, client, name: str):
        self.engine = client
        self.name = name

    def stop(self) -> None:
        self.engine.stop()


class _FailureState:
    def __init__(self, fail_rate: float = 0.0, fail_requests: Optional[int] = None):
        self.fail_rate = fail_rate
        self.fail_requests = fail_requests
        self.counter = 0

    def should_fail(self) -> bool:
        self.counter += 1
        if self.fail_requests is not None and self.counter <= self.fail_requests:
            return True
        if self.fail_rate > 0:
            # deterministic pseudo-random by counter (reproducible tests)
            x = (self.counter * 2654435761) & 0xFFFFFFFF
            return (x / 0xFFFFFFFF) < self.fail_rate
        return False


def _format_logprobs(tokenizer, ids, lps) -> Dict[str, Any]:
    """OpenAI chat `logprobs` object from delivered token ids and the
    engine's per-token (logprob, [(id, lp), ...]) records."""
    content = []
    for tok_id, (lp, tops) in zip(ids, lps):
        t = tokenizer.decode([int(tok_id)])
        content.append({
            "token": t,
            "logprob": float(lp),
            "bytes": list(t.encode("utf-8")),
            "top_logprobs": [
                {
                    "token": tokenizer.decode([int(i)]),
                    "logprob": float(v),
                    "bytes": list(tokenizer.decode([int(i)]).encode("utf-8")),
                }
                for i, v in tops
            ],
        })
    return {"content": content}


class EngineRegistry:
    def __init__(self, settings: Optional[Settings] = None):
        self.settings = settings or Settings()
        self._engines: Dict[str, _EngineHandle] = {}
        self._failures: Dict[str, _FailureState] = {}
        self._create_lock = threading.Lock()
        self.tokenizer = ByteTokenizer()

    # ---- engine lifecycle ----
    def _engine_key(self, spec: EngineSpec) -> str:
        # identical specs share one engine (replica provider entries on one
        # GPU); explicit batch/block/dtype overrides make the spec distinct
        key = f"{spec.model}@{spec.device}x{spec.tp}"
        if spec.max_batch_size:
            key += f"/b{spec.max_batch_size}"
        if spec.kv_block_size:
            key += f"/k{spec.kv_block_size}"
        if spec.kv_dtype not in ("auto", None):
            key += f"/kv{spec.kv_dtype}"
        if spec.dtype not in ("bfloat16", None):
            key += f"/{spec.dtype}"
        return key

    def _resolve_device(self, spec: EngineSpec) -> str:
        if torch.cuda.is_available():
            n = torch.cuda.device_count()
            if spec.device >= n:
                raise RuntimeError(
                    f"Engine spec asks for GPU {spec.device} but only {n} are visible"
                )
            return f"cuda:{spec.device}"
        return "cpu"

    def get_engine(self, spec: EngineSpec) -> _EngineHandle:
        key = self._engine_key(spec)
        handle = self._engines.get(key)
        if handle is not None:
            return handle
        with self._create_lock:
            handle = self._engines.get(key)
            if handle is not None:
                return handle
            if spec.tp > 1:
                from .tp_group import TPEngineClient

                if torch.cuda.is_available() and torch.cuda.device_count() < spec.tp:
                    raise RuntimeError(
                        f"Engine spec asks for TP={spec.tp} but only "
                        f"{torch.cuda.device_count()} GPUs are visible"
                    )
                client = TPEngineClient(
                    model=spec.model,
                    tp=spec.tp,
                    dtype="float16" if spec.dtype in ("float16", "fp16") else "bfloat16",
                    max_batch_size=spec.max_batch_size or self.settings.engine_max_batch_size,
                    kv_block_size=spec.kv_block_size or self.settings.engine_kv_block_size,
                )
                handle = _TPGroupHandle(client, key)
                self._engines[key] = handle
                logger.info("created TP=%d engine group %s", spec.tp, key)
                return handle
            device = self._resolve_device(spec)
            dtype = torch.bfloat16 if device.startswith("cuda") else torch.float32
            if spec.dtype in ("float16", "fp16"):
                dtype = torch.float16
            engine = LLMEngine(
                model=spec.model,
                device=device,
                dtype=dtype,
                block_size=spec.kv_block_size or self.settings.engine_kv_block_size,
                max_batch_size=spec.max_batch_size or self.settings.engine_max_batch_size,
                hbm_fraction=self.settings.engine_hbm_fraction,
                num_blocks=None if device.startswith("cuda") else 256,
                prefix_caching=self.settings.engine_prefix_caching,
                use_hipgraph=None if self.settings.engine_use_hipgraph else False,
                kv_dtype=spec.kv_dtype,
            )
            if engine.graph_runner is not None:
                logger.info("pre-capturing decode hipGraphs for %s", key)
                engine.graph_runner.capture_all()
            handle = _EngineHandle(engine, key)
            self._engines[key] = handle
            logger.info("created engine %s on %s", key, device)
            return handle

    def prune(self, active_specs: List[EngineSpec]) -> None:
        """Stop and release engines no longer referenced by any local
        provider entry (config reload): a removed 70B engine would
        otherwise pin its weights in HBM indefinitely."""
        keep = {self._engine_key(spec) for spec in active_specs}
        with self._create_lock:
            dead = [k for k in self._engines if k not in keep]
            for key in dead:
                handle = self._engines.pop(key)
                logger.info("pruning engine %s (no longer configured)", key)
                eng = handle.engine
                for req in (
                    list(getattr(eng, "waiting", []))
                    + list(getattr(eng, "prefilling", []))
                    + list(getattr(eng, "running", []))
                ):
                    try:  # fail in-flight work loudly instead of stranding it
                        eng.abort_request(req)
                    except Exception:
                        pass
                try:
                    handle.stop()
                except Exception:
                    logger.exception("failed to stop engine %s", key)
        if dead and torch.cuda.is_available():
            torch.cuda.empty_cache()

    def _failure_state(self, provider: str, spec: EngineSpec) -> _FailureState:
        st = self._failures.get(provider)
        if st is None:
            st = _FailureState(spec.fail_rate, spec.fail_requests)
            self._failures[provider] = st
        return st

    def set_failure(
        self, provider: str, fail_rate: float = 0.0, fail_requests: Optional[int] = None
    ) -> None:
        st = _FailureState(fail_rate, fail_requests)
        self._failures[provider] = st

    @staticmethod
    def _ensure_batch_notifier(engine, loop) -> None:
        """Install (or re-bind after a loop change, e.g. between test
        asyncio.run calls) the engine's batched token-event dispatcher."""
        if getattr(engine, "_notifier_loop", None) is loop:
            return

        def dispatch(events):  # runs on the event loop
            for req, tok in events:
                q = getattr(req, "_aq", None)
                if q is not None:
                    q.put_nowait(("token", tok))

        def notifier(events):
            try:
                loop.call_soon_threadsafe(dispatch, events)
            except RuntimeError:  # loop closed; requests fail via timeout
                pass

        engine.batch_notifier = notifier
        engine._notifier_loop = loop

    # ---- the request path ----
    async def make_request(
        self,
        provider_name: str,
        spec: EngineSpec,
        payload: Dict[str, Any],
        is_streaming: bool,
    ) -> Tuple[Optional[Any], Optional[str]]:
        fail = self._failure_state(provider_name, spec)
        if fail.should_fail():
            detail = f"Injected failure on local provider '{provider_name}' (request #{fail.counter})"
            logger.warning(detail)
            return None, detail

        try:
            handle = await asyncio.to_thread(self.get_engine, spec)
        except Exception as e:
            return None, f"Engine for provider '{provider_name}' failed to start: {e}"
        engine = handle.engine

        # admission control: fail fast into the fallback chain instead of
        # queueing unboundedly under overload (the q300 loadgen run showed
        # requests waiting >100 s before erroring without this)
        depth = len(getattr(engine, "waiting", None) or []) + len(
            getattr(engine, "_reqs", None) or []
        )
        if depth >= self.settings.engine_max_queue:
            return None, (
                f"Engine for provider '{provider_name}' is overloaded "
                f"(queue depth {depth} >= {self.settings.engine_max_queue})"
            )

        tokenizer = getattr(engine, "tokenizer", None) or self.tokenizer
        messages = payload.get("messages")
        if messages:
            prompt_text = tokenizer.render_chat(messages)
        else:
            prompt_text = str(payload.get("prompt", ""))
        prompt_ids = tokenizer.encode(prompt_text)
        if len(prompt_ids) >= engine.max_model_len:
            return None, (
                f"Prompt of {len(prompt_ids)} tokens exceeds model context "
                f"({engine.max_model_len}) on '{provider_name}'"
            )
        params = SamplingParams.from_payload(payload)
        params.max_tokens = min(params.max_tokens, engine.max_model_len - len(prompt_ids) - 1)
        # random-init weights rarely emit EOS; respect explicit ignore_eos
        if "ignore_eos" in payload:
            params.ignore_eos = bool(payload["ignore_eos"])

        try:
            n_choices = int(payload.get("n") or 1)
        except (TypeError, ValueError):
            n_choices = 1
        if n_choices > 16:
            return None, "n > 16 is not supported"
        if n_choices > 1:
            return await self._make_request_multi(
                provider_name, engine, prompt_ids, params, payload,
                is_streaming, n_choices,
            )

        loop = asyncio.get_running_loop()
        queue: asyncio.Queue = asyncio.Queue()

        def on_token(req: EngineRequest, tok: int) -> None:
            loop.call_soon_threadsafe(queue.put_nowait, ("token", tok))

        def on_finish(req: EngineRequest) -> None:
            loop.call_soon_threadsafe(queue.put_nowait, ("finish", req))

        if hasattr(engine, "batch_notifier"):
            # single-process engine: one cross-thread wakeup per STEP for
            # all requests (engine._deliver batches), not one per token
            self._ensure_batch_notifier(engine, loop)
            req = EngineRequest(prompt_ids, params, on_finish=on_finish)
            req._aq = queue
        else:  # TP worker-group client: token rate is per-group, keep simple
            req = EngineRequest(prompt_ids, params, on_token=on_token, on_finish=on_finish)
        try:
            engine.add_request(req)
        except Exception as e:
            return None, f"Engine '{provider_name}' rejected request: {e}"

        model_name = payload.get("model", spec.model)
        completion_id = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())

        # prime: wait for the first token (or failure) BEFORE returning, so a
        # failing request yields (None, error) with zero bytes sent —
        # first-chunk semantics parity (request_handler.py:67-100)
        try:
            kind, value = await asyncio.wait_for(queue.get(), timeout=300.0)
        except asyncio.TimeoutError:
            engine.abort_request(req)
            return None, f"Engine '{provider_name}' timed out before first token"
        if kind == "finish" and value.state == "failed":
            return None, value.error or "engine failure"

        if not is_streaming:
            return await self._collect_nonstream(
                req, engine, queue, kind, value, completion_id, created,
                model_name, provider_name,
            )
        return (
            self._stream_response(
                req, engine, queue, kind, value, completion_id, created, model_name
            ),
            None,
        )

    async def _make_request_multi(
        self, provider_name, engine, prompt_ids, params, payload,
        is_streaming, n,
    ):
        """OpenAI `n` choices: n engine requests share the prompt (the
        prefix cache dedups its KV). Greedy runs return identical choices
        (matching upstream behavior); temperature>0 diverges per choice.
        Stop-string holdback trimming is not applied to multi-choice
        streams (the engine still terminates on the stop string)."""
        import dataclasses

        loop = asyncio.get_running_loop()
        shared: asyncio.Queue = asyncio.Queue()

        class _Tagged:
            __slots__ = ("q", "i")

            def __init__(self, q, i):
                self.q, self.i = q, i

            def put_nowait(self, item):
                self.q.put_nowait((self.i, item))

        use_batch = hasattr(engine, "batch_notifier")
        if use_batch:
            self._ensure_batch_notifier(engine, loop)
        reqs: List[EngineRequest] = []
        for i in range(n):
            def mk_finish(idx):
                def onf(req):
                    loop.call_soon_threadsafe(shared.put_nowait, (idx, ("finish", req)))
                return onf

            p_i = dataclasses.replace(params, stop=list(params.stop))
            if use_batch:
                req = EngineRequest(prompt_ids, p_i, on_finish=mk_finish(i))
                req._aq = _Tagged(shared, i)
            else:
                def mk_tok(idx):
                    def ont(req, tok):
                        loop.call_soon_threadsafe(shared.put_nowait, (idx, ("token", tok)))
                    return ont

                req = EngineRequest(
                    prompt_ids, p_i, on_token=mk_tok(i), on_finish=mk_finish(i)
                )
            try:
                engine.add_request(req)
            except Exception as e:
                for r in reqs:
                    engine.abort_request(r)
                return None, f"Engine '{provider_name}' rejected request: {e}"
            reqs.append(req)

        model_name = payload.get("model", "unknown")
        completion_id = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())
        tokenizer = getattr(engine, "tokenizer", None) or self.tokenizer

        try:
            first = await asyncio.wait_for(shared.get(), timeout=300.0)
        except asyncio.TimeoutError:
            for r in reqs:
                engine.abort_request(r)
            return None, f"Engine '{provider_name}' timed out before first token"
        if first[1][0] == "finish" and first[1][1].state == "failed":
            for r in reqs:
                engine.abort_request(r)
            return None, first[1][1].error or "engine failure"

        def agg_usage():
            comp = sum(len(r.out_ids) for r in reqs)
            return {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": comp,
                "total_tokens": len(prompt_ids) + comp,
            }

        if not is_streaming:
            remaining = n
            ev = first
            while True:
                _, (kind, value) = ev
                if kind == "finish":
                    remaining -= 1
                    if remaining == 0:
                        break
                ev = await shared.get()
            choices = []
            for i, req in enumerate(reqs):
                text = tokenizer.decode(req.out_ids)
                for s in req.params.stop:
                    cut = text.find(s)
                    if cut >= 0:
                        text = text[:cut]
                choices.append(
                    {
                        "index": i,
                        "message": {"role": "assistant", "content": text},
                        "finish_reason": req.finish_reason or "stop",
                    }
                )
            return (
                {
                    "id": completion_id,
                    "object": "chat.completion",
                    "created": created,
                    "model": model_name,
                    "choices": choices,
                    "usage": agg_usage(),
                },
                None,
            )

        def chunk(idx, delta, finish=None, usage=None) -> bytes:
            obj: Dict[str, Any] = {
                "id": completion_id,
                "object": "chat.completion.chunk",
                "created": created,
                "model": model_name,
                "choices": [{"index": idx, "delta": delta, "finish_reason": finish}],
            }
            if usage is not None:
                obj["usage"] = usage
            return b"data: " + json.dumps(obj, separators=(",", ":")).encode() + b"\n\n"

        async def gen():
            try:
                for i in range(n):
                    yield chunk(i, {"role": "assistant", "content": ""})
                remaining = n
                ev = first
                while True:
                    idx, (kind, value) = ev
                    if kind == "token":
                        yield chunk(idx, {"content": tokenizer.decode([value])})
                    else:
                        remaining -= 1
                        yield chunk(
                            idx, {}, finish=value.finish_reason or "stop",
                            usage=agg_usage() if remaining == 0 else None,
                        )
                        if remaining == 0:
                            yield b"data: [DONE]\n\n"
                            return
                    ev = await shared.get()
            finally:
                for r in reqs:
                    if r.state in ("waiting", "running"):
                        engine.abort_request(r)

        return (
            StreamingResponse(
                gen(), media_type="text/event-stream", headers=dict(STREAM_HEADERS)
            ),
            None,
        )

    async def _collect_nonstream(
        self, req, engine, queue, kind, value, completion_id, created,
        model_name, provider_name,
    ):
        tokens: List[int] = []
        finished: Optional[EngineRequest] = None
        while True:
            if kind == "token":
                tokens.append(value)
            else:
                finished = value
                break
            kind, value = await queue.get()
        if finished is not None and finished.state == "failed":
            return None, finished.error or "engine failure"
        text = (getattr(engine, "tokenizer", None) or self.tokenizer).decode(tokens)
        for s in req.params.stop:  # OpenAI semantics: stop string excluded
            idx = text.find(s)
            if idx >= 0:
                text = text[:idx]
        usage = {
            "prompt_tokens": len(req.prompt_ids),
            "completion_tokens": len(req.out_ids),
            "total_tokens": len(req.prompt_ids) + len(req.out_ids),
            # per-phase tracing (extra field; OpenAI clients ignore it)
            "timings": req.timings() if hasattr(req, "timings") else {},
        }
        return (
            {
                "id": completion_id,
                "object": "chat.completion",
                "created": created,
                "model": model_name,
                "choices": [
                    {
                        "index": 0,
                        "message": {"role": "assistant", "content": text},
                        "logprobs": _format_logprobs(
                            getattr(engine, "tokenizer", None) or self.tokenizer,
                            req.out_ids[: len(req.out_logprobs)],
                            req.out_logprobs,
                        )
                        if req.params.logprobs and getattr(req, "out_logprobs", None)
                        else None,
                        "finish_reason": req.finish_reason or "stop",
                    }
                ],
                "usage": usage,
            },
            None,
        )

    def _stream_response(
        self, req, engine, queue, first_kind, first_value, completion_id, created, model_name
    ) -> StreamingResponse:
        tokenizer = getattr(engine, "tokenizer", None) or self.tokenizer

        def chunk(delta: Dict[str, Any], finish: Optional[str] = None, usage=None,
                  logprobs=None) -> bytes:
            obj: Dict[str, Any] = {
                "id": completion_id,
                "object": "chat.completion.chunk",
                "created": created,
                "model": model_name,
                "choices": [{"index": 0, "delta": delta, "finish_reason": finish,
                             "logprobs": logprobs}],
            }
            if usage is not None:
                obj["usage"] = usage
            return b"data: " + json.dumps(obj, separators=(",", ":")).encode() + b"\n\n"

        stops = req.params.stop
        holdback = max((len(s) for s in stops), default=1) - 1 if stops else 0

        want_lp = bool(req.params.logprobs)
        lp_pos = {"i": 0}

        def chunk_logprobs(toks):
            if not want_lp:
                return None
            i0 = lp_pos["i"]
            lps = req.out_logprobs[i0 : i0 + len(toks)]
            lp_pos["i"] = i0 + len(toks)
            return _format_logprobs(tokenizer, toks[: len(lps)], lps)

        async def gen():
            pending = ""
            stopped = False
            try:
                yield chunk({"role": "assistant", "content": ""})
                kind, value = first_kind, first_value
                while True:
                    if kind == "token":
                        # coalesce every token already queued into ONE SSE
                        # chunk: per-token chunks saturate the event loop
                        # near ~10K tokens/s (json+send per token)
                        toks = [value]
                        nxt = None
                        while True:
                            try:
                                k2, v2 = queue.get_nowait()
                            except asyncio.QueueEmpty:
                                break
                            if k2 == "token":
                                toks.append(v2)
                            else:
                                nxt = (k2, v2)
                                break
                        text = tokenizer.decode(toks)
                        lp_obj = chunk_logprobs(toks)
                        if stops and not stopped:
                            # hold back enough text to cleanly cut a stop
                            # string before it reaches the client
                            pending += text
                            cut = -1
                            for s in stops:
                                i = pending.find(s)
                                if i >= 0 and (cut < 0 or i < cut):
                                    cut = i
                            if cut >= 0:
                                if pending[:cut]:
                                    yield chunk({"content": pending[:cut]})
                                pending = ""
                                stopped = True
                            elif len(pending) > holdback:
                                emit = pending[: len(pending) - holdback]
                                pending = pending[len(pending) - holdback :]
                                yield chunk({"content": emit})
                        elif not stopped:
                            if text:
                                yield chunk({"content": text}, logprobs=lp_obj)
                        if nxt is not None:
                            kind, value = nxt
                            continue
                    else:
                        fin: EngineRequest = value
                        if pending and not stopped:
                            yield chunk({"content": pending})
                            pending = ""
                        usage = {
                            "prompt_tokens": len(fin.prompt_ids),
                            "completion_tokens": len(fin.out_ids),
                            "total_tokens": len(fin.prompt_ids) + len(fin.out_ids),
                            "timings": fin.timings() if hasattr(fin, "timings") else {},
                        }
                        yield chunk({}, finish=fin.finish_reason or "stop", usage=usage)
                        yield b"data: [DONE]\n\n"
                        return
                    kind, value = await queue.get()
            finally:
                if req.state in ("waiting", "running"):
                    engine.abort_request(req)

        return StreamingResponse(
            gen(), media_type="text/event-stream", headers=dict(STREAM_HEADERS)
        )

    # ---- introspection ----
    def list_models(self, spec: EngineSpec) -> List[Dict[str, Any]]:
        from ..models.configs import MODEL_PRESETS

        out = []
        for name, cfg in sorted(MODEL_PRESETS.items()):
            out.append(
                {
                    "id": name,
                    "object": "model",
                    "owned_by": "llmapigateway-amd-engine",
                    "context_length": cfg.max_positions,
                }
            )
        return out

    def stats(self) -> List[Dict[str, Any]]:
        out = []
        for key, handle in self._engines.items():
            eng = handle.engine
            row = {
                "engine": key,
                "model": eng.full_config.name,
                **{k: int(v) for k, v in eng.stats.items()},
            }
            if hasattr(eng, "kv"):  # single-process engine
                row.update(
                    device=str(eng.device),
                    waiting=len(eng.waiting),
                    prefilling=len(eng.prefilling),
                    running=len(eng.running),
                    kv_blocks_free=eng.kv.manager.num_free_blocks,
                    kv_blocks_total=eng.kv.num_blocks,
                    kv_dtype="fp8" if eng.kv.fp8 else str(eng.dtype).replace("torch.", ""),
                    prefix_cache_hits=eng.kv.manager.stats_prefix_hits,
                    prefix_cached_tokens=eng.kv.manager.stats_prefix_tokens,
                )
            else:  # TP worker group (state lives in the worker processes)
                row.update(device=f"tp{eng.tp}", in_flight=len(eng._reqs))
            out.append(row)
        return out

    async def aclose(self) -> None:
        for handle in self._engines.values():
            handle.stop()
        self._engines.clear()
