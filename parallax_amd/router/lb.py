"""Multi-cluster HTTP load balancer.

Reference analogue: src/router/ (round_robin / random / performance strategies
with EWMA TTFT/TPOT scoring, TTL health polling of /cluster/status, runtime
reconfiguration over HTTP). Fresh asyncio/FastAPI implementation."""

from __future__ import annotations

import asyncio
import random
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import httpx
from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import (HTMLResponse, JSONResponse,
                                StreamingResponse)

from ..utils.logging_config import get_logger

logger = get_logger("router.lb")


@dataclass
class ClusterEndpoint:
    url: str
    healthy: bool = True
    last_check: float = 0.0
    # EWMA performance metrics (reference lb_strategy.py:26-110)
    ewma_ttft_ms: Optional[float] = None
    ewma_tps: Optional[float] = None
    inflight: int = 0

    def score(self) -> float:
        """Lower is better."""
        ttft = self.ewma_ttft_ms if self.ewma_ttft_ms is not None else 1000.0
        tps = self.ewma_tps if self.ewma_tps else 1.0
        return (ttft + 1000.0 / tps) * (1 + 0.2 * self.inflight)

    def update_metrics(self, ttft_ms: Optional[float], tps: Optional[float],
                       alpha: float = 0.3) -> None:
        if ttft_ms is not None:
            self.ewma_ttft_ms = (
                ttft_ms if self.ewma_ttft_ms is None
                else alpha * ttft_ms + (1 - alpha) * self.ewma_ttft_ms
            )
        if tps is not None:
            self.ewma_tps = (
                tps if self.ewma_tps is None
                else alpha * tps + (1 - alpha) * self.ewma_tps
            )


class LoadBalancer:
    def __init__(self, endpoints: List[str], strategy: str = "round_robin",
                 health_ttl_s: float = 10.0):
        assert strategy in ("round_robin", "random", "performance")
        self.endpoints: Dict[str, ClusterEndpoint] = {
            u: ClusterEndpoint(url=u.rstrip("/")) for u in endpoints
        }
        self.strategy = strategy
        self.health_ttl_s = health_ttl_s
        self._rr = 0
        self._rng = random.Random(0)
        self._lock = threading.Lock()

    # -- configuration (runtime reconfigurable, reference router/main.py) ---------

    def add_endpoint(self, url: str) -> None:
        with self._lock:
            self.endpoints.setdefault(url.rstrip("/"), ClusterEndpoint(url.rstrip("/")))

    def remove_endpoint(self, url: str) -> None:
        with self._lock:
            self.endpoints.pop(url.rstrip("/"), None)

    def set_strategy(self, strategy: str) -> None:
        assert strategy in ("round_robin", "random", "performance")
        self.strategy = strategy

    # -- selection ----------------------------------------------------------------

    def pick(self) -> Optional[ClusterEndpoint]:
        with self._lock:
            healthy = [e for e in self.endpoints.values() if e.healthy]
            if not healthy:
                return None
            if self.strategy == "random":
                return self._rng.choice(healthy)
            if self.strategy == "performance":
                return min(healthy, key=lambda e: e.score())
            self._rr = (self._rr + 1) % len(healthy)
            return healthy[self._rr]

    # -- health -------------------------------------------------------------------

    async def check_health(self) -> None:
        async with httpx.AsyncClient(timeout=5.0) as client:
            for ep in list(self.endpoints.values()):
                if time.monotonic() - ep.last_check < self.health_ttl_s:
                    continue
                try:
                    r = await client.get(f"{ep.url}/cluster/status")
                    ep.healthy = r.status_code == 200 and r.json().get(
                        "bootstrapped", r.json().get("initialized", False)
                    )
                except httpx.HTTPError:
                    ep.healthy = False
                ep.last_check = time.monotonic()


def create_router_app(lb: LoadBalancer) -> FastAPI:
    app = FastAPI(title="parallax_amd-router", version="0.1.0")
    app.state.lb = lb

    @app.get("/health")
    async def health():
        return {
            "endpoints": {
                u: {"healthy": e.healthy, "ttft": e.ewma_ttft_ms,
                    "tps": e.ewma_tps, "inflight": e.inflight}
                for u, e in lb.endpoints.items()
            },
            "strategy": lb.strategy,
        }

    @app.post("/config/endpoints")
    async def config_endpoints(request: Request):
        body = await request.json()
        for u in body.get("add", []):
            lb.add_endpoint(u)
        for u in body.get("remove", []):
            lb.remove_endpoint(u)
        if body.get("strategy"):
            lb.set_strategy(body["strategy"])
        return {"status": "ok"}

    async def _relay(request: Request, path: str):
        """Proxy one OpenAI request to the picked cluster. Streaming requests
        relay the SSE bytes as they arrive (the reference router relays SSE,
        request_handler.py:190-245); metrics update from the usage chunk."""
        body = await request.json()
        await lb.check_health()
        ep = lb.pick()
        if ep is None:
            raise HTTPException(503, "no healthy cluster")
        ep.inflight += 1
        t0 = time.monotonic()
        if not body.get("stream"):
            try:
                async with httpx.AsyncClient(timeout=600.0) as client:
                    r = await client.post(f"{ep.url}{path}", json=body)
                data = r.json()
                usage = data.get("usage", {})
                ep.update_metrics(
                    usage.get("ttft_ms", (time.monotonic() - t0) * 1e3),
                    usage.get("tps"),
                )
                return JSONResponse(data, status_code=r.status_code)
            except httpx.HTTPError as e:
                ep.healthy = False
                raise HTTPException(502, f"cluster {ep.url} failed: {e}")
            finally:
                ep.inflight -= 1

        async def sse():
            import json as _json

            first_t = None
            try:
                async with httpx.AsyncClient(timeout=600.0) as client:
                    async with client.stream(
                        "POST", f"{ep.url}{path}", json=body
                    ) as r:
                        async for line in r.aiter_lines():
                            if first_t is None and line.startswith("data:"):
                                first_t = time.monotonic()
                            if line.startswith("data: ") and '"usage"' in line:
                                try:
                                    msg = _json.loads(line[6:])
                                    u = msg.get("usage") or {}
                                    ep.update_metrics(
                                        u.get("ttft_ms",
                                              ((first_t or time.monotonic())
                                               - t0) * 1e3),
                                        u.get("tps"),
                                    )
                                except ValueError:
                                    pass
                            yield line + "\n"
            except httpx.HTTPError:
                ep.healthy = False
                yield 'data: {"error": "upstream failed"}\n\n'
            finally:
                ep.inflight -= 1

        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        return await _relay(request, "/v1/chat/completions")

    @app.post("/v1/completions")
    async def completions(request: Request):
        return await _relay(request, "/v1/completions")

    @app.get("/")
    async def dashboard():
        rows = "".join(
            f"<tr><td>{u}</td><td>{'up' if e.healthy else 'DOWN'}</td>"
            f"<td>{e.ewma_ttft_ms and round(e.ewma_ttft_ms, 1)}</td>"
            f"<td>{e.ewma_tps and round(e.ewma_tps, 1)}</td>"
            f"<td>{e.inflight}</td></tr>"
            for u, e in lb.endpoints.items()
        )
        html = (
            "<html><head><title>parallax-amd router</title>"
            "<meta http-equiv=refresh content=3>"
            "<style>body{font-family:monospace;background:#111;color:#ddd}"
            "td,th{padding:4px 12px;border-bottom:1px solid #333}</style>"
            f"</head><body><h2>router · strategy: {lb.strategy}</h2>"
            "<table><tr><th>endpoint</th><th>health</th><th>ttft ms</th>"
            f"<th>tps</th><th>inflight</th></tr>{rows}</table></body></html>"
        )
        return HTMLResponse(html)

    return app
