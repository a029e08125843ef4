"""OpenAI-compatible HTTP frontend (FastAPI).

Endpoints (parity with the reference's vllm-rs frontend + backend API surface):
POST /v1/chat/completions (stream + non-stream), POST /v1/completions,
GET /v1/models, GET /health, GET /stats.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import List, Optional

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..utils.logging_config import get_logger
from .engine_server import EngineServer
from .sampling_params import SamplingParams
from .tokenizer_util import IncrementalDetokenizer, TokenizerWrapper

logger = get_logger("server.http_frontend")


def create_app(
    server: EngineServer,
    tokenizer: TokenizerWrapper,
    model_name: str = "parallax-amd-model",
) -> FastAPI:
    app = FastAPI(title="parallax_amd", version="0.1.0")

    def _params(body: dict) -> SamplingParams:
        sp = SamplingParams.from_openai(body)
        if (tokenizer.eos_token_id is not None and not sp.stop_token_ids
                and not sp.ignore_eos):
            # default stop tokens to the tokenizer's EOS (it may differ from the
            # model-config eos_token_ids the engine applies); ignore_eos
            # requests must stay stop-free or they lose async decode
            # pipelining (Engine._async_eligible)
            sp.stop_token_ids = [tokenizer.eos_token_id]
        if sp.json_schema and server.engine.sampler.grammar_vocab is None:
            # one-time id->text table for constrained decoding
            server.engine.set_grammar_vocab(tokenizer.vocab_strings())
        return sp

    def _find_stop(text: str, stops: List[str]):
        """Earliest stop-string hit in the decoded text, or -1."""
        best = -1
        for st in stops:
            i = text.find(st)
            if i >= 0 and (best < 0 or i < best):
                best = i
        return best

    async def _collect(stream, sp: SamplingParams):
        """Drain a request stream fully (non-streaming path). Stop STRINGS
        are matched here on the detokenized text (token-id stops happen in
        the engine) — on a hit the request is aborted and the text truncated
        (reference: frontend-side stop matching)."""
        token_ids: List[int] = []
        logprobs: List[float] = []
        finish_reason = "stop"
        stop_at: int = -1
        while True:
            out = await stream.aget()
            if out is None:
                break
            if out.token_id < 0:  # abort/timeout terminator
                finish_reason = out.finish_reason or "abort"
                continue
            token_ids.append(out.token_id)
            if out.logprob is not None:
                logprobs.append(out.logprob)
            if sp.stop and stop_at < 0:
                hit = _find_stop(tokenizer.decode(token_ids), sp.stop)
                if hit >= 0:
                    stop_at = hit
                    server.abort(out.rid)
            if out.finished:
                finish_reason = out.finish_reason or "stop"
        return token_ids, finish_reason, logprobs, stop_at

    def _chat_logprobs(token_ids, logprobs):
        """OpenAI chat logprobs block: one entry per sampled token."""
        return {
            "content": [
                {"token": tokenizer.decode([t]), "logprob": lp}
                for t, lp in zip(token_ids, logprobs)
            ]
        }

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/v1/models")
    async def models():
        return {
            "object": "list",
            "data": [{"id": model_name, "object": "model", "owned_by": "parallax_amd"}],
        }

    @app.get("/stats")
    async def stats():
        return server.stats()

    @app.post("/profile")
    async def profile(request: Request):
        """Trace the next N engine steps to a chrome trace file."""
        body = await request.json()
        path = body.get("path", "/tmp/parallax_trace.json")
        server.profile_next(int(body.get("steps", 10)), path)
        return {"tracing_next_steps": int(body.get("steps", 10)), "path": path}

    @app.post("/update_weights")
    async def update_weights(request: Request):
        body = await request.json()
        path = body.get("model_path")
        if not path:
            raise HTTPException(400, "model_path required")
        try:
            with server._lock:
                n = server.engine.update_weights_from_disk(path)
        except (FileNotFoundError, OSError, ValueError) as e:
            raise HTTPException(400, f"refit failed: {e}")
        return {"updated_tensors": n}

    async def _sse_multi(streams, all_rids, make_chunk, make_usage,
                         stop_strings=None):
        """Merge n request streams into one SSE stream (OpenAI streams n>1
        as interleaved chunks labeled by choice index). Each pump drains its
        queue greedily and detokenizes incrementally; stop STRINGS truncate
        the stream at the match (token-id stops happen in the engine);
        client disconnect aborts every underlying request."""
        q: asyncio.Queue = asyncio.Queue()
        stops = stop_strings or []
        tail_keep = max((len(x) for x in stops), default=0)

        async def pump(idx, st):
            detok = IncrementalDetokenizer(tokenizer)
            emitted = ""  # tail window for cross-chunk stop-string hits
            first_t = None
            done = False
            while not done:
                out = await st.aget()
                if out is None:
                    break
                # greedy drain: under load several tokens are already queued
                # — one chunk per wakeup keeps the loop at O(steps), not
                # O(tokens) (per-token SSE capped the loop at ~3k tok/s)
                batch = [out]
                while True:
                    try:
                        nxt = st.aio_queue.get_nowait()
                    except asyncio.QueueEmpty:
                        break
                    if nxt is None:
                        done = True
                        break
                    batch.append(nxt)
                if first_t is None:
                    first_t = time.monotonic()
                finish = None
                new_ids = []
                for o in batch:
                    if o.token_id >= 0:
                        new_ids.append(o.token_id)
                    if o.finished:
                        finish = o.finish_reason or "stop"
                delta = detok.push(new_ids) if new_ids else ""
                if delta and stops:
                    tail = emitted[-tail_keep:]
                    hit = _find_stop(tail + delta, stops)
                    if hit >= 0:
                        # truncate at the match (a hit inside the already-
                        # emitted tail keeps nothing more) and stop
                        delta = delta[: max(0, hit - len(tail))]
                        server.abort(st.rid)
                        await q.put(("chunk", idx, delta, "stop"))
                        break
                    emitted = (emitted + delta)[-tail_keep:] \
                        if tail_keep else ""
                if delta or finish is not None:
                    await q.put(("chunk", idx, delta, finish))
            await q.put(("end", idx, len(detok.ids), first_t))

        tasks = [asyncio.create_task(pump(i, st))
                 for i, st in enumerate(streams)]
        t_start = time.monotonic()
        first_any = None
        total_tokens = 0
        active = len(streams)
        try:
            while active:
                kind, idx, a, b = await q.get()
                if kind == "chunk":
                    if first_any is None:
                        first_any = time.monotonic()
                    yield make_chunk(idx, a, b)
                else:
                    active -= 1
                    total_tokens += a
                    if first_any is None and b is not None:
                        first_any = b
            yield make_usage(total_tokens,
                             first_any or time.monotonic(), t_start)
            yield "data: [DONE]\n\n"
        except asyncio.CancelledError:
            # client went away mid-stream: stop generating everywhere
            for t in tasks:
                t.cancel()
            for r in all_rids:
                server.abort(r)
            raise

    def _submit_n(prompt_ids, sp, rid, n):
        """Submit the primary + n-1 extra same-prompt requests (the prefix
        cache makes the extra prefills nearly free)."""
        loop = asyncio.get_running_loop()
        streams = [server.submit(prompt_ids, sp, rid=rid,
                                 aio_loop=loop, aio_queue=asyncio.Queue())]
        for k in range(1, n):
            streams.append(server.submit(
                prompt_ids, sp, rid=f"{rid}-{k}",
                aio_loop=loop, aio_queue=asyncio.Queue()))
        return streams

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        messages = body.get("messages")
        if not messages:
            raise HTTPException(400, "messages required")
        try:
            prompt_ids = tokenizer.chat_prompt_ids(messages)
        except (TypeError, ValueError, KeyError, AttributeError) as e:
            raise HTTPException(400, f"bad messages: {e}")
        try:
            sp = _params(body)
        except (ValueError, TypeError) as e:  # bad sampling params -> 400
            raise HTTPException(400, str(e))
        rid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        n = max(1, int(body.get("n", 1)))
        try:
            streams = _submit_n(prompt_ids, sp, rid, n)
        except ValueError as e:  # over context limit etc -> clean client error
            raise HTTPException(400, str(e))
        created = int(time.time())

        if body.get("stream"):
            def make_chunk(idx, delta, finish):
                chunk = {
                    "id": rid, "object": "chat.completion.chunk",
                    "created": created, "model": model_name,
                    "choices": [{
                        "index": idx,
                        "delta": {"content": delta},
                        "finish_reason": finish,
                    }],
                }
                return f"data: {json.dumps(chunk)}\n\n"

            def make_usage(total, first_t, t_start):
                # final usage chunk (reference logs TPS/TTFT from this)
                elapsed = time.monotonic() - t_start
                usage = {
                    "prompt_tokens": len(prompt_ids),
                    "completion_tokens": total,
                    "total_tokens": len(prompt_ids) + total,
                    "ttft_ms": round((first_t - t_start) * 1e3, 2),
                    "tps": round(total / max(elapsed, 1e-6), 2),
                }
                return "data: " + json.dumps({
                    "id": rid, "object": "chat.completion.chunk",
                    "created": created, "model": model_name,
                    "choices": [], "usage": usage,
                }) + "\n\n"

            return StreamingResponse(
                _sse_multi(streams, [st.rid for st in streams],
                           make_chunk, make_usage, stop_strings=sp.stop),
                media_type="text/event-stream")

        results = await asyncio.gather(*(_collect(st, sp) for st in streams))
        choices, total_completion = [], 0
        for idx, (token_ids, finish_reason, logprobs, stop_at) in \
                enumerate(results):
            content = tokenizer.decode(token_ids)
            if stop_at >= 0:
                content = content[:stop_at]
                finish_reason = "stop"
            total_completion += len(token_ids)
            choices.append({
                "index": idx,
                "message": {"role": "assistant", "content": content},
                "logprobs": _chat_logprobs(token_ids, logprobs)
                if sp.logprobs and logprobs else None,
                "finish_reason": finish_reason,
            })
        return JSONResponse({
            "id": rid, "object": "chat.completion", "created": created,
            "model": model_name,
            "choices": choices,
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": total_completion,
                "total_tokens": len(prompt_ids) + total_completion,
            },
        })

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        prompt = body.get("prompt")
        if prompt is None:
            raise HTTPException(400, "prompt required")
        try:
            if isinstance(prompt, list) and prompt \
                    and isinstance(prompt[0], int):
                prompt_ids = [int(t) for t in prompt]
            else:
                prompt_ids = tokenizer.encode(prompt)
        except (TypeError, ValueError, AttributeError) as e:
            raise HTTPException(400, f"bad prompt: {e}")
        try:
            sp = _params(body)
        except (ValueError, TypeError) as e:  # bad sampling params -> 400
            raise HTTPException(400, str(e))
        rid = f"cmpl-{uuid.uuid4().hex[:24]}"
        n = max(1, int(body.get("n", 1)))
        try:
            streams = _submit_n(prompt_ids, sp, rid, n)
        except ValueError as e:  # over context limit etc -> clean client error
            raise HTTPException(400, str(e))
        created = int(time.time())

        if body.get("stream"):
            def make_chunk(idx, delta, finish):
                chunk = {
                    "id": rid, "object": "text_completion",
                    "created": created, "model": model_name,
                    "choices": [{
                        "index": idx, "text": delta,
                        "finish_reason": finish,
                    }],
                }
                return f"data: {json.dumps(chunk)}\n\n"

            def make_usage(total, first_t, t_start):
                elapsed = time.monotonic() - t_start
                usage = {
                    "prompt_tokens": len(prompt_ids),
                    "completion_tokens": total,
                    "total_tokens": len(prompt_ids) + total,
                    "ttft_ms": round((first_t - t_start) * 1e3, 2),
                    "tps": round(total / max(elapsed, 1e-6), 2),
                }
                return "data: " + json.dumps({
                    "id": rid, "object": "text_completion",
                    "created": created, "model": model_name,
                    "choices": [], "usage": usage,
                }) + "\n\n"

            return StreamingResponse(
                _sse_multi(streams, [st.rid for st in streams],
                           make_chunk, make_usage, stop_strings=sp.stop),
                media_type="text/event-stream")

        results = await asyncio.gather(*(_collect(st, sp) for st in streams))
        choices, total_completion = [], 0
        echo_text = tokenizer.decode(prompt_ids) if body.get("echo") else ""
        for idx, (token_ids, finish_reason, logprobs, stop_at) in \
                enumerate(results):
            text = tokenizer.decode(token_ids)
            if stop_at >= 0:
                text = text[:stop_at]
                finish_reason = "stop"
            total_completion += len(token_ids)
            choices.append({
                "index": idx, "text": echo_text + text,
                "logprobs": {"token_logprobs": logprobs}
                if sp.logprobs and logprobs else None,
                "finish_reason": finish_reason,
            })
        return JSONResponse({
            "id": rid, "object": "text_completion", "created": created,
            "model": model_name,
            "choices": choices,
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": total_completion,
                "total_tokens": len(prompt_ids) + total_completion,
            },
        })

    return app
