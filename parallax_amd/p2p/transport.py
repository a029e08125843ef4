"""Peer-to-peer transports for the decentralized layer.

Reference analogue: Lattica (libp2p RPC) in p2p/server.py. Fresh design with
two implementations behind one interface:
- LoopbackTransport: in-process queues — the reference's own test strategy
  ("loopback in-process transport so N-stage pipelines are
  testable in one process").
- TcpTransport: length-prefixed msgpack frames over TCP sockets between hosts
  (offline environment: no libp2p; NAT traversal is out of scope, peers
  address each other host:port as assigned by the scheduler service).
"""

from __future__ import annotations

import queue
import socket
import struct
import threading
from typing import Callable, Dict, Optional

from ..utils.logging_config import get_logger

logger = get_logger("p2p.transport")


class Transport:
    """send(peer_id, payload bytes); incoming payloads arrive on recv()."""

    def send(self, peer_id: str, payload: bytes) -> None:
        raise NotImplementedError

    def recv(self, timeout: Optional[float] = None) -> Optional[bytes]:
        raise NotImplementedError

    def close(self) -> None:
        pass


class LoopbackTransport(Transport):
    """All peers in one process; a shared registry of inbox queues."""

    def __init__(self, peer_id: str, registry: Dict[str, "LoopbackTransport"]):
        self.peer_id = peer_id
        self.inbox: "queue.Queue[bytes]" = queue.Queue()
        self.registry = registry
        registry[peer_id] = self

    def send(self, peer_id: str, payload: bytes) -> None:
        self.registry[peer_id].inbox.put(payload)

    def recv(self, timeout: Optional[float] = None) -> Optional[bytes]:
        try:
            return self.inbox.get(timeout=timeout)
        except queue.Empty:
            return None


class TcpTransport(Transport):
    """Length-prefixed frames; one listening socket, lazy outbound connections
    (reconnect on failure). peer addresses are set via set_peer_addr (from the
    scheduler's cluster view)."""

    #: refuse frames claiming more than this many bytes (a corrupt or malicious
    #: peer must not be able to drive unbounded allocation on the receiver)
    DEFAULT_MAX_FRAME_BYTES = 512 * 1024 * 1024
    #: inbox backpressure: a reader that stalls makes senders block in
    #: sendall (TCP flow control) instead of growing receiver memory
    DEFAULT_MAX_INBOX = 4096

    def __init__(self, peer_id: str, host: str = "0.0.0.0", port: int = 0,
                 max_frame_bytes: int = DEFAULT_MAX_FRAME_BYTES,
                 max_inbox: int = DEFAULT_MAX_INBOX,
                 auth_token: Optional[str] = None):
        self.peer_id = peer_id
        self.max_frame_bytes = max_frame_bytes
        # shared-secret handshake: when set, every outbound connection sends
        # an AUTH frame first and inbound connections must present it before
        # any payload frame is accepted
        self.auth_token = auth_token
        self.inbox: "queue.Queue[bytes]" = queue.Queue(maxsize=max_inbox)
        self._peers: Dict[str, tuple] = {}
        self._conns: Dict[str, socket.socket] = {}
        self._lock = threading.Lock()
        self._server = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._server.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._server.bind((host, port))
        self._server.listen(64)
        self.port = self._server.getsockname()[1]
        self._stop = threading.Event()
        self._accept_thread = threading.Thread(target=self._accept_loop, daemon=True)
        self._accept_thread.start()

    def set_peer_addr(self, peer_id: str, host: str, port: int) -> None:
        self._peers[peer_id] = (host, port)

    # -- outbound ---------------------------------------------------------------

    def send(self, peer_id: str, payload: bytes) -> None:
        with self._lock:
            conn = self._conns.get(peer_id)
        for attempt in range(2):
            try:
                if conn is None:
                    host, port = self._peers[peer_id]
                    conn = socket.create_connection((host, port), timeout=10)
                    conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                    if self.auth_token is not None:
                        tok = self.auth_token.encode()
                        conn.sendall(b"AUTH" + struct.pack("<I", len(tok)) + tok)
                    with self._lock:
                        self._conns[peer_id] = conn
                conn.sendall(struct.pack("<Q", len(payload)) + payload)
                return
            except OSError as e:
                with self._lock:
                    self._conns.pop(peer_id, None)
                conn = None
                if attempt == 1:
                    raise ConnectionError(f"send to {peer_id} failed: {e}") from e

    # -- inbound -----------------------------------------------------------------

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                self._server.settimeout(0.5)
                conn, _ = self._server.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            threading.Thread(
                target=self._read_loop, args=(conn,), daemon=True
            ).start()

    def _read_loop(self, conn: socket.socket) -> None:
        try:
            if self.auth_token is not None:
                magic = self._read_exact(conn, 4)
                if magic != b"AUTH":
                    logger.warning("dropping connection: no auth handshake")
                    return
                hdr = self._read_exact(conn, 4)
                if hdr is None:
                    return
                (tl,) = struct.unpack("<I", hdr)
                if tl > 4096:
                    logger.warning("dropping connection: oversized auth token")
                    return
                tok = self._read_exact(conn, tl)
                if tok != self.auth_token.encode():
                    logger.warning("dropping connection: bad auth token")
                    return
            while not self._stop.is_set():
                header = self._read_exact(conn, 8)
                if header is None:
                    break
                (n,) = struct.unpack("<Q", header)
                if n > self.max_frame_bytes:
                    logger.warning(
                        "dropping connection: frame of %d bytes exceeds max %d",
                        n, self.max_frame_bytes,
                    )
                    break
                payload = self._read_exact(conn, n)
                if payload is None:
                    break
                # bounded inbox: blocking put -> the TCP window closes and the
                # SENDER stalls (backpressure) rather than this side ballooning
                self.inbox.put(payload)
        finally:
            conn.close()

    @staticmethod
    def _read_exact(conn: socket.socket, n: int) -> Optional[bytes]:
        buf = b""
        while len(buf) < n:
            chunk = conn.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf

    def recv(self, timeout: Optional[float] = None) -> Optional[bytes]:
        try:
            return self.inbox.get(timeout=timeout)
        except queue.Empty:
            return None

    def close(self) -> None:
        self._stop.set()
        try:
            self._server.close()
        except OSError:
            pass
        with self._lock:
            for c in self._conns.values():
                try:
                    c.close()
                except OSError:
                    pass
