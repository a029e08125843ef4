"""Cluster scheduler service: the HTTP control plane.

Reference analogue: src/backend/main.py + scheduler_manage.py +
rpc_connection_handler.py (FastAPI + Lattica RPC). Fresh design: all control
traffic is HTTP (offline environment has no libp2p); workers join/heartbeat via
REST, chat completions are proxied to the head peer of the routed pipeline with
the reference's retry/backoff ladder (request_handler.py:33-36 behavior).

Endpoints:
  POST /node/join     {node_id, host, port(frontend), p2p_port, hardware{...}}
  POST /node/update   heartbeat {node_id, layer_latency_ms, current_requests}
  POST /node/leave    {node_id}
  POST /scheduler/init {model_name, hf_config, min_nodes}
  GET  /cluster/status
  POST /weight/refit
  POST /v1/chat/completions  (proxied to the routed head peer)
"""

from __future__ import annotations

import asyncio
import os
import threading
import time
from typing import Dict, Optional

import httpx
from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..models.config import ModelConfig
from ..scheduling.model_info import ModelInfo
from ..scheduling.node import Node, NodeHardware
from ..scheduling.scheduler import ClusterScheduler
from ..utils.logging_config import get_logger

logger = get_logger("backend.service")

ROUTING_RETRIES = 20        # reference: 20 retries @5s on empty table -> 429
ROUTING_RETRY_DELAY_S = 0.5
FORWARD_RETRIES = 3         # reference: 10 @10s -> 502 (shorter here)
FORWARD_RETRY_DELAY_S = 1.0


class SchedulerService:
    def __init__(self):
        self.scheduler: Optional[ClusterScheduler] = None
        self.model_info: Optional[ModelInfo] = None
        self.node_endpoints: Dict[str, dict] = {}  # node_id -> {host, port, p2p_port}
        self._lock = threading.Lock()
        self._sweeper: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def init_model(self, model_name: str, hf_config: dict, min_nodes: int = 1,
                   allocator: str = "greedy", routing: str = "round_robin") -> None:
        cfg = ModelConfig.from_hf_config(hf_config)
        self.model_info = ModelInfo.from_config(model_name, cfg)
        self.scheduler = ClusterScheduler(
            self.model_info, min_nodes_bootstrapping=min_nodes,
            allocator=allocator, routing_strategy=routing,
        )
        if self._sweeper is None:
            self._sweeper = threading.Thread(target=self._sweep_loop, daemon=True)
            self._sweeper.start()
        logger.info("scheduler initialized for %s (min_nodes=%d)", model_name, min_nodes)

    def _sweep_loop(self) -> None:
        while not self._stop.is_set():
            time.sleep(5.0)
            with self._lock:
                if self.scheduler is not None:
                    self.scheduler.sweep_heartbeats()

    def shutdown(self) -> None:
        self._stop.set()


def create_backend_app(service: Optional[SchedulerService] = None) -> FastAPI:
    svc = service or SchedulerService()
    app = FastAPI(title="parallax_amd-scheduler", version="0.1.0")
    app.state.service = svc

    def _require(body: dict, *keys):
        missing = [k for k in keys if k not in body]
        if missing:
            raise HTTPException(400, f"missing field(s): {missing}")

    @app.post("/scheduler/init")
    async def scheduler_init(request: Request):
        body = await request.json()
        _require(body, "hf_config")
        svc.init_model(
            body.get("model_name", "model"),
            body["hf_config"],
            min_nodes=int(body.get("min_nodes", 1)),
            allocator=body.get("allocator", "greedy"),
            routing=body.get("routing", "round_robin"),
        )
        return {"status": "ok"}

    @app.post("/node/join")
    async def node_join(request: Request):
        if svc.scheduler is None:
            raise HTTPException(503, "scheduler not initialized")
        body = await request.json()
        _require(body, "node_id")
        hw = body.get("hardware", {}) or {}
        node = Node(
            node_id=body["node_id"],
            hardware=NodeHardware(
                name=hw.get("name", "MI355X"),
                num_gpus=int(hw.get("num_gpus", 1)),
                memory_gb=float(hw.get("memory_gb", 288.0)),
                tflops_bf16=float(hw.get("tflops_bf16", 2500.0)),
                memory_bandwidth_gbps=float(hw.get("memory_bandwidth_gbps", 8000.0)),
            ),
        )
        with svc._lock:
            assignment = svc.scheduler.node_join(node)
            svc.node_endpoints[body["node_id"]] = {
                "host": body.get("host", "127.0.0.1"),
                "port": body.get("port"),
                "p2p_port": body.get("p2p_port"),
            }
        return {
            "assignment": assignment.__dict__ if assignment else None,
            "model_name": svc.model_info.name,
            "hf_config": svc.model_info.cfg.raw or None,
            "peers": svc.node_endpoints,
        }

    @app.post("/node/update")
    async def node_update(request: Request):
        if svc.scheduler is None:
            raise HTTPException(503, "scheduler not initialized")
        body = await request.json()
        _require(body, "node_id")
        with svc._lock:
            assignment = svc.scheduler.node_update(
                body["node_id"],
                layer_latency_ms=body.get("layer_latency_ms"),
                current_requests=body.get("current_requests"),
                rtt_ms=body.get("rtt_ms"),
                last_refit_time=body.get("last_refit_time"),
            )
        return {
            "assignment": assignment.__dict__ if assignment else None,
            "last_refit_time": svc.scheduler.last_refit_time,
            "peers": svc.node_endpoints,
        }

    @app.post("/node/leave")
    async def node_leave(request: Request):
        body = await request.json()
        _require(body, "node_id")
        with svc._lock:
            if svc.scheduler is not None:
                svc.scheduler.node_leave(body["node_id"])
            svc.node_endpoints.pop(body["node_id"], None)
        return {"status": "ok"}

    @app.get("/")
    async def dashboard():
        from fastapi.responses import HTMLResponse

        # full single-file app (setup wizard + node monitor + chat UI — the
        # dependency-free equivalent of the reference's React dashboard);
        # the inline page below is the fallback if the static file is missing
        static = os.path.join(os.path.dirname(__file__), "static",
                              "dashboard.html")
        if os.path.exists(static):
            with open(static) as f:
                return HTMLResponse(f.read())
        return HTMLResponse("""<!doctype html><html><head>
<title>parallax_amd cluster</title>
<style>body{font-family:monospace;margin:2em;background:#111;color:#eee}
table{border-collapse:collapse}td,th{border:1px solid #444;padding:4px 10px}
.ok{color:#6f6}.bad{color:#f66}</style></head><body>
<h2>parallax_amd cluster</h2><div id=s>loading...</div>
<script>
async function load(){
  const r = await fetch('/cluster/status'); const d = await r.json();
  if(!d.initialized){document.getElementById('s').innerText='scheduler not initialized';return;}
  let h = `<p>model: <b>${d.model}</b> | bootstrapped: ${d.bootstrapped} | `+
          `nodes: ${d.num_nodes} | pipelines: ${d.num_pipelines}</p>`;
  h += '<table><tr><th>node</th><th>hw</th><th>layers</th><th>inflight</th><th>ms/layer</th><th>alive</th></tr>';
  for(const n of d.nodes){h += `<tr><td>${n.node_id}</td><td>${n.hardware}</td>`+
    `<td>[${n.start_layer}, ${n.end_layer})</td><td>${n.current_requests}</td>`+
    `<td>${(n.layer_latency_ms||0).toFixed(3)}</td>`+
    `<td class="${n.active?'ok':'bad'}">${n.active?'yes':'NO'}</td></tr>`;}
  h += '</table><p>pipelines:</p><ul>';
  for(const p of d.pipelines){h += `<li>${p.join(' &rarr; ')}</li>`;}
  h += '</ul>';
  document.getElementById('s').innerHTML = h;
}
load(); setInterval(load, 3000);
</script></body></html>""")

    def _status() -> dict:
        if svc.scheduler is None:
            return {"initialized": False}
        with svc._lock:
            status = svc.scheduler.cluster_status()
        status["initialized"] = True
        status["endpoints"] = svc.node_endpoints
        return status

    @app.get("/cluster/status")
    async def cluster_status():
        return _status()

    @app.get("/cluster/status_stream")
    async def cluster_status_stream(interval_s: float = 2.0, count: int = 0):
        """NDJSON status stream (reference /cluster/status NDJSON,
        backend/main.py:172-193): one JSON line per tick; count=0 streams
        until the client disconnects."""
        import json as _json

        from fastapi.responses import StreamingResponse

        async def ndjson():
            n = 0
            while True:
                yield _json.dumps(_status()) + "\n"
                n += 1
                if count and n >= count:
                    break
                await asyncio.sleep(interval_s)

        return StreamingResponse(ndjson(), media_type="application/x-ndjson")

    @app.post("/weight/refit")
    async def weight_refit():
        if svc.scheduler is None:
            raise HTTPException(503, "scheduler not initialized")
        with svc._lock:
            t = svc.scheduler.update_last_refit_time()
        return {"last_refit_time": t}

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        """Route to a pipeline, proxy to the head peer's OpenAI frontend
        (reference RequestHandler retry ladder)."""
        if svc.scheduler is None:
            raise HTTPException(503, "scheduler not initialized")
        body = await request.json()
        decision = None
        for _ in range(ROUTING_RETRIES):
            with svc._lock:
                decision = svc.scheduler.dispatch_next_request()
            if decision is not None:
                break
            await asyncio.sleep(ROUTING_RETRY_DELAY_S)
        if decision is None:
            raise HTTPException(429, "no pipeline available")
        head = decision.routing_table[0]
        ep = svc.node_endpoints.get(head)
        if ep is None or not ep.get("port"):
            svc.scheduler.complete_request(decision.routing_table)
            raise HTTPException(502, f"head peer {head} has no HTTP endpoint")
        url = f"http://{ep['host']}:{ep['port']}/v1/chat/completions"
        body.setdefault("routing_table", decision.routing_table)
        t0 = time.monotonic()

        def _log_usage(usage: dict) -> None:
            # reference gateway parses the final SSE usage chunk and logs
            # per-request TPS/TTFT (request_handler.py:190-202)
            if not usage:
                return
            logger.info(
                "request via %s: in=%s out=%s ttft_ms=%s tps=%s e2e_ms=%.0f",
                head, usage.get("prompt_tokens"), usage.get("completion_tokens"),
                usage.get("ttft_ms"), usage.get("tps"),
                (time.monotonic() - t0) * 1e3,
            )

        if body.get("stream"):
            # SSE relay: stream chunks through while holding the routing slot
            async def relay():
                last_usage = {}
                try:
                    async with httpx.AsyncClient(timeout=600.0) as client:
                        async with client.stream("POST", url, json=body) as resp:
                            async for chunk in resp.aiter_bytes():
                                if b'"usage"' in chunk:
                                    try:
                                        import json as _json

                                        for line in chunk.decode().splitlines():
                                            if line.startswith("data: {"):
                                                d = _json.loads(line[6:])
                                                if d.get("usage"):
                                                    last_usage = d["usage"]
                                    except Exception:
                                        pass
                                yield chunk
                finally:
                    _log_usage(last_usage)
                    with svc._lock:
                        svc.scheduler.complete_request(decision.routing_table)

            return StreamingResponse(relay(), media_type="text/event-stream")

        try:
            async with httpx.AsyncClient(timeout=600.0) as client:
                for attempt in range(FORWARD_RETRIES):
                    try:
                        resp = await client.post(url, json=body)
                        payload = resp.json()
                        _log_usage(payload.get("usage") or {})
                        return JSONResponse(payload, status_code=resp.status_code)
                    except httpx.HTTPError:
                        if attempt == FORWARD_RETRIES - 1:
                            raise
                        await asyncio.sleep(FORWARD_RETRY_DELAY_S)
        except httpx.HTTPError as e:
            raise HTTPException(502, f"forward to {head} failed: {e}")
        finally:
            with svc._lock:
                svc.scheduler.complete_request(decision.routing_table)

    return app
