"""Worker node agent: join -> assignment -> serve -> heartbeat -> elastic reload.

Reference analogue: parallax/launch.py + p2p/server.py GradientServer (join via
node_join RPC, 10 s announcer heartbeat, layer-reallocation detection that
restarts the executors, :757-874). Fresh design over HTTP + TcpTransport.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Dict, List, Optional

import httpx
import torch

from ..models.config import ModelConfig
from ..server.request import new_request_id
from ..server.sampling_params import SamplingParams
from ..utils.logging_config import get_logger
from .peer_executor import PeerExecutor, PeerOutput
from .transport import TcpTransport

logger = get_logger("p2p.node_agent")


class _AioBridgeQueue:
    """Queue facade whose put() hops into an asyncio loop."""

    def __init__(self, loop, aio_queue):
        self.loop = loop
        self.aio_queue = aio_queue

    def put(self, item) -> None:
        self.loop.call_soon_threadsafe(self.aio_queue.put_nowait, item)


class NodeAgent:
    def __init__(
        self,
        scheduler_url: str,
        node_id: Optional[str] = None,
        host: str = "127.0.0.1",
        http_port: Optional[int] = None,
        hardware: Optional[dict] = None,
        model_path: Optional[str] = None,
        random_weights: bool = True,
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.float32,
        num_kv_blocks: int = 1024,
        block_size: int = 16,
        heartbeat_interval_s: float = 10.0,
    ):
        self.scheduler_url = scheduler_url.rstrip("/")
        self.node_id = node_id or f"node-{new_request_id()[:8]}"
        self.host = host
        self.http_port = http_port
        if hardware is None and device is not None and device.type == "cuda":
            from ..scheduling.node import detect_hardware

            hw = detect_hardware()
            hardware = {"name": hw.name, "num_gpus": hw.num_gpus,
                        "memory_gb": hw.memory_gb,
                        "tflops_bf16": hw.tflops_bf16,
                        "memory_bandwidth_gbps": hw.memory_bandwidth_gbps}
        self.hardware = hardware or {"name": "MI355X", "num_gpus": 1,
                                     "memory_gb": 288.0}
        self.model_path = model_path
        self.random_weights = random_weights
        self.device = device
        self.dtype = dtype
        self.num_kv_blocks = num_kv_blocks
        self.block_size = block_size
        self.heartbeat_interval_s = heartbeat_interval_s

        self.transport = TcpTransport(self.node_id, host, 0)
        self.executor: Optional[PeerExecutor] = None
        self.assignment: Optional[dict] = None
        self.cfg: Optional[ModelConfig] = None
        self._streams: Dict[str, "queue.Queue"] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._step_ms_ewma: Optional[float] = None
        self.reload_count = 0

    # -- lifecycle --------------------------------------------------------------

    def join(self, timeout_s: float = 60.0) -> dict:
        payload = {
            "node_id": self.node_id, "host": self.host, "port": self.http_port,
            "p2p_port": self.transport.port, "hardware": self.hardware,
        }
        deadline = time.monotonic() + timeout_s
        with httpx.Client(timeout=10.0) as client:
            while time.monotonic() < deadline:
                r = client.post(f"{self.scheduler_url}/node/join", json=payload)
                if r.status_code == 200:
                    resp = r.json()
                    self.cfg = ModelConfig.from_hf_config(resp["hf_config"])
                    if resp.get("assignment"):
                        self._apply_assignment(resp["assignment"], resp["peers"])
                        return resp
                    # not yet bootstrapped: wait via heartbeat
                    got = self._wait_for_assignment(client, deadline)
                    if got:
                        return got
                time.sleep(1.0)
        raise TimeoutError("no layer assignment from scheduler")

    def _wait_for_assignment(self, client: httpx.Client, deadline: float) -> Optional[dict]:
        while time.monotonic() < deadline:
            r = client.post(
                f"{self.scheduler_url}/node/update", json={"node_id": self.node_id}
            )
            if r.status_code == 200:
                resp = r.json()
                if resp.get("assignment"):
                    self._apply_assignment(resp["assignment"], resp["peers"])
                    return resp
            time.sleep(0.5)
        return None

    def _apply_assignment(self, assignment: dict, peers: Dict[str, dict]) -> None:
        logger.info("%s assigned layers [%d,%d)", self.node_id,
                    assignment["start_layer"], assignment["end_layer"])
        self.assignment = assignment
        for pid, ep in peers.items():
            if pid != self.node_id and ep.get("p2p_port"):
                self.transport.set_peer_addr(pid, ep["host"], ep["p2p_port"])
        self.executor = PeerExecutor(
            self.cfg, assignment["start_layer"], assignment["end_layer"],
            self.node_id, self.transport, device=self.device, dtype=self.dtype,
            num_kv_blocks=self.num_kv_blocks, block_size=self.block_size,
            random_weights=self.random_weights, model_path=self.model_path,
        )

    def start(self) -> None:
        assert self.executor is not None, "join() first"
        if self.device is not None and self.device.type == "cuda":
            # WAN nodes run the eager forward path (no graph runner): cap the
            # one-time per-shape hipBLASLt search so a previously-unseen
            # prefill shape stalls the pipeline stage by ~0.1 s, not ~1.5 s
            # (RPC latency dominates end-to-end here anyway)
            os.environ.setdefault("PARALLAX_LT_TUNE_MS", "100")
        t1 = threading.Thread(target=self._step_loop, daemon=True, name="peer-step")
        t2 = threading.Thread(target=self._heartbeat_loop, daemon=True,
                              name="peer-heartbeat")
        self._threads = [t1, t2]
        t1.start()
        t2.start()

    def abort(self, rid: str) -> None:
        if self.executor is not None and self.executor.is_head:
            self.executor.abort(rid)

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=5)
        try:
            with httpx.Client(timeout=5.0) as client:
                client.post(f"{self.scheduler_url}/node/leave",
                            json={"node_id": self.node_id})
        except httpx.HTTPError:
            pass
        self.transport.close()

    # -- loops --------------------------------------------------------------------

    def _step_loop(self) -> None:
        import queue as queue_mod

        while not self._stop.is_set():
            t0 = time.monotonic()
            with self._lock:
                ex = self.executor
            if ex is None:
                time.sleep(0.05)
                continue
            try:
                ex.step(recv_timeout=0.005)
            except Exception:
                # a step must never kill the node (reference keeps serving on
                # batch errors); affected requests abort via the sweeps
                logger.exception("executor step failed; node keeps serving")
                time.sleep(0.1)
                continue
            if ex.is_head:
                for out in ex.drain_outputs():
                    q = self._streams.get(out.rid)
                    if q is not None:
                        q.put(out)
                        if out.finished:
                            q.put(None)
                            self._streams.pop(out.rid, None)
            dt = (time.monotonic() - t0) * 1e3
            n_layers = self.assignment["end_layer"] - self.assignment["start_layer"]
            per_layer = dt / max(1, n_layers)
            self._step_ms_ewma = (
                per_layer if self._step_ms_ewma is None
                else 0.2 * per_layer + 0.8 * self._step_ms_ewma
            )

    def _heartbeat_loop(self) -> None:
        while not self._stop.is_set():
            try:
                with httpx.Client(timeout=10.0) as client:
                    r = client.post(
                        f"{self.scheduler_url}/node/update",
                        json={
                            "node_id": self.node_id,
                            "layer_latency_ms": self._step_ms_ewma,
                            "current_requests": (
                                self.executor.scheduler.num_running
                                if self.executor and self.executor.is_head else 0
                            ),
                        },
                    )
                if r.status_code == 200:
                    resp = r.json()
                    a = resp.get("assignment")
                    if a and self.assignment and (
                        a["start_layer"] != self.assignment["start_layer"]
                        or a["end_layer"] != self.assignment["end_layer"]
                    ):
                        # elastic reload (reference launch.py:251-299): rebuild
                        # the executor on the new layer range
                        logger.warning("%s reassigned to [%d,%d) — reloading",
                                       self.node_id, a["start_layer"], a["end_layer"])
                        with self._lock:
                            self._apply_assignment(a, resp.get("peers", {}))
                            self.reload_count += 1
            except httpx.HTTPError as e:
                logger.warning("heartbeat failed: %s", e)
            self._stop.wait(self.heartbeat_interval_s)

    # -- head request API (used by the head HTTP frontend) ----------------------------

    def submit(self, prompt_ids: List[int], sp: SamplingParams,
               routing_table: List[str], aio_loop=None, aio_queue=None):
        """aio_loop/aio_queue: asyncio delivery bridge — outputs are pushed
        with call_soon_threadsafe so async consumers never block an executor
        thread (see engine_server.RequestStream.deliver)."""
        import queue as queue_mod

        assert self.executor is not None and self.executor.is_head
        rid = new_request_id()
        if aio_queue is not None:
            q = _AioBridgeQueue(aio_loop, aio_queue)
        else:
            q = queue_mod.Queue()
        with self._lock:
            self._streams[rid] = q
            self.executor.submit(prompt_ids, sp, routing_table, rid=rid)
        return rid, q
