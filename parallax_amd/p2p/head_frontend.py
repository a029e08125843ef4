"""OpenAI frontend served by the HEAD peer of a decentralized pipeline.

The gateway (backend/service.py) routes a request to this app with the chosen
routing_table in the body; tokens stream back from the last stage through the
node agent's output demux. Reference analogue: the vllm-rs frontend launched on
the head peer (start_layer == 0, launch.py)."""

from __future__ import annotations

import asyncio
import time
import uuid
from typing import List

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..server.sampling_params import SamplingParams
from ..server.tokenizer_util import TokenizerWrapper
from ..utils.logging_config import get_logger
from .node_agent import NodeAgent

logger = get_logger("p2p.head_frontend")


def _find_stop(text: str, stops) -> int:
    best = -1
    for st in stops or []:
        i = text.find(st)
        if i >= 0 and (best < 0 or i < best):
            best = i
    return best


def create_head_app(
    agent: NodeAgent, tokenizer: TokenizerWrapper, model_name: str = "model"
) -> FastAPI:
    app = FastAPI(title="parallax_amd-head", version="0.1.0")

    @app.get("/health")
    async def health():
        return {"status": "ok", "node_id": agent.node_id}

    @app.get("/v1/models")
    async def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model",
                          "owned_by": "parallax_amd"}]}

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        prompt = body.get("prompt")
        if prompt is None:
            raise HTTPException(400, "prompt required")
        if isinstance(prompt, list) and prompt and isinstance(prompt[0], int):
            prompt_ids = prompt
        else:
            prompt_ids = tokenizer.encode(prompt)
        routing_table: List[str] = body.get("routing_table") or [agent.node_id]
        sp = SamplingParams.from_openai(body)
        aq = asyncio.Queue()
        rid, _ = agent.submit(prompt_ids, sp, routing_table,
                              aio_loop=asyncio.get_running_loop(), aio_queue=aq)
        token_ids, finish_reason = [], "stop"
        while True:
            out = await aq.get()
            if out is None:
                break
            if out.token_id >= 0:
                token_ids.append(out.token_id)
            if out.finished:
                finish_reason = out.finish_reason or "stop"
        return JSONResponse({
            "id": rid, "object": "text_completion",
            "created": int(time.time()), "model": model_name,
            "choices": [{
                "index": 0, "text": tokenizer.decode(token_ids),
                "finish_reason": finish_reason,
            }],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": len(token_ids),
                "total_tokens": len(prompt_ids) + len(token_ids),
            },
        })

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        messages = body.get("messages")
        if not messages:
            raise HTTPException(400, "messages required")
        routing_table: List[str] = body.get("routing_table") or [agent.node_id]
        prompt_ids = tokenizer.chat_prompt_ids(messages)
        sp = SamplingParams.from_openai(body)
        aq = asyncio.Queue()
        rid, _ = agent.submit(prompt_ids, sp, routing_table,
                              aio_loop=asyncio.get_running_loop(), aio_queue=aq)
        created = int(time.time())

        if body.get("stream"):
            async def sse():
                import json as _json

                token_ids: List[int] = []
                sent_len = 0
                t0 = time.monotonic()
                first_t = None
                while True:
                    out = await aq.get()
                    if out is None:
                        break
                    if first_t is None:
                        first_t = time.monotonic()
                    if out.token_id < 0:
                        continue
                    token_ids.append(out.token_id)
                    text = tokenizer.decode(token_ids)
                    delta, nl = text[sent_len:], len(text)
                    sent_len = nl
                    if sp.stop:
                        hit = _find_stop(text, sp.stop)
                        if hit >= 0:
                            # truncate at the stop string and terminate
                            delta = text[min(sent_len - len(delta), hit):hit]
                            agent.executor.abort(rid)
                            chunk = {
                                "id": rid,
                                "object": "chat.completion.chunk",
                                "created": created, "model": model_name,
                                "choices": [{
                                    "index": 0,
                                    "delta": {"content": delta},
                                    "finish_reason": "stop",
                                }],
                            }
                            yield f"data: {_json.dumps(chunk)}\n\n"
                            break
                    chunk = {
                        "id": rid, "object": "chat.completion.chunk",
                        "created": created, "model": model_name,
                        "choices": [{
                            "index": 0, "delta": {"content": delta},
                            "finish_reason": out.finish_reason
                            if out.finished else None,
                        }],
                    }
                    yield f"data: {_json.dumps(chunk)}\n\n"
                elapsed = time.monotonic() - t0
                usage = {
                    "prompt_tokens": len(prompt_ids),
                    "completion_tokens": len(token_ids),
                    "total_tokens": len(prompt_ids) + len(token_ids),
                    "ttft_ms": round(
                        ((first_t or time.monotonic()) - t0) * 1e3, 2),
                    "tps": round(len(token_ids) / max(elapsed, 1e-6), 2),
                }
                yield "data: " + _json.dumps({
                    "id": rid, "object": "chat.completion.chunk",
                    "created": created, "model": model_name,
                    "choices": [], "usage": usage,
                }) + "\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(sse(), media_type="text/event-stream")

        token_ids, finish_reason = [], "stop"
        t0 = time.monotonic()
        first_t = None
        stop_at = -1
        while True:
            out = await aq.get()
            if out is None:
                break
            if first_t is None:
                first_t = time.monotonic()
            if out.token_id >= 0:
                token_ids.append(out.token_id)
                if sp.stop and stop_at < 0:
                    hit = _find_stop(tokenizer.decode(token_ids), sp.stop)
                    if hit >= 0:
                        stop_at = hit
                        agent.abort(out.rid) if hasattr(agent, "abort") else None
            if out.finished:
                finish_reason = out.finish_reason or "stop"
        elapsed = time.monotonic() - t0
        return JSONResponse({
            "id": f"chatcmpl-{uuid.uuid4().hex[:16]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": model_name,
            "choices": [{
                "index": 0,
                "message": {"role": "assistant",
                            "content": tokenizer.decode(token_ids)[:stop_at]
                            if stop_at >= 0 else tokenizer.decode(token_ids)},
                "finish_reason": "stop" if stop_at >= 0 else finish_reason,
            }],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": len(token_ids),
                "total_tokens": len(prompt_ids) + len(token_ids),
                "ttft_ms": round(((first_t or time.monotonic()) - t0) * 1e3, 2),
                "tps": round(len(token_ids) / max(elapsed, 1e-6), 2),
            },
        })

    return app
