"""Weight-refit chunk distribution over the P2P transport.

Reference parity: the reference streams CID-checksummed weight chunks over the
Lattica block store, filters them to the node's layer range, concatenates
partitions and keeps 3 checkpoint versions on disk with GC
(/root/reference/src/parallax/p2p/server.py:224-338,
utils/weight_refit_utils.py:20-160). Fresh MI355X-native design: the publisher
(trainer/scheduler side) chunks a checkpoint directory into sha256-addressed
pieces; fetchers pull only the files covering their layer range through the
same length-prefixed transport the pipeline packets use, verify every chunk,
reassemble into a version directory and garbage-collect old versions.
"""

from __future__ import annotations

import hashlib
import json
import os
import shutil
from typing import Callable, Dict, List, Optional, Tuple

from ..utils.logging_config import get_logger

logger = get_logger("p2p.refit")

DEFAULT_CHUNK_SIZE = 4 << 20
KEEP_VERSIONS = 3


class RefitError(RuntimeError):
    pass


def build_manifest(
    ckpt_dir: str, version: int, chunk_size: int = DEFAULT_CHUNK_SIZE
) -> dict:
    """Chunk every file of a checkpoint dir into sha256-addressed pieces.
    Files covering only layers outside a node's range can be skipped by the
    fetcher (per-file `layers` span from the safetensors weight map)."""
    files = []
    layer_map: Dict[str, List[int]] = {}
    idx_path = os.path.join(ckpt_dir, "model.safetensors.index.json")
    if os.path.exists(idx_path):
        with open(idx_path) as f:
            weight_map = json.load(f).get("weight_map", {})
        for name, fname in weight_map.items():
            if name.startswith("model.layers."):
                g = int(name.split(".")[2])
                span = layer_map.setdefault(fname, [g, g + 1])
                span[0] = min(span[0], g)
                span[1] = max(span[1], g + 1)
            else:
                layer_map[fname] = [0, 1 << 30]  # endpoint weights: all ranges
    for name in sorted(os.listdir(ckpt_dir)):
        path = os.path.join(ckpt_dir, name)
        if not os.path.isfile(path):
            continue
        size = os.path.getsize(path)
        chunks = []
        with open(path, "rb") as f:
            while True:
                data = f.read(chunk_size)
                if not data:
                    break
                chunks.append(
                    {"cid": hashlib.sha256(data).hexdigest(), "size": len(data)}
                )
        files.append({
            "name": name,
            "size": size,
            "chunks": chunks,
            "layers": layer_map.get(name),  # None = always fetch
        })
    return {"version": version, "chunk_size": chunk_size, "files": files}


class RefitPublisher:
    """Serves chunk reads for a manifest it built."""

    def __init__(self, ckpt_dir: str, version: int,
                 chunk_size: int = DEFAULT_CHUNK_SIZE):
        self.ckpt_dir = ckpt_dir
        self.manifest = build_manifest(ckpt_dir, version, chunk_size)

    def get_chunk(self, name: str, idx: int) -> bytes:
        entry = next(f for f in self.manifest["files"] if f["name"] == name)
        if not (0 <= idx < len(entry["chunks"])):
            raise RefitError(f"chunk index {idx} out of range for {name}")
        cs = self.manifest["chunk_size"]
        with open(os.path.join(self.ckpt_dir, name), "rb") as f:
            f.seek(idx * cs)
            return f.read(entry["chunks"][idx]["size"])


class RefitFetcher:
    """Pulls a manifest's chunks (layer-range filtered), verifies each CID,
    reassembles files under dest_root/v{version}/ and keeps the newest
    KEEP_VERSIONS version directories."""

    def __init__(self, dest_root: str, keep_versions: int = KEEP_VERSIONS):
        self.dest_root = dest_root
        self.keep_versions = keep_versions
        os.makedirs(dest_root, exist_ok=True)

    def wanted_files(
        self, manifest: dict,
        layer_range: Optional[Tuple[int, int]] = None,
    ) -> List[dict]:
        out = []
        for f in manifest["files"]:
            span = f.get("layers")
            if layer_range is not None and span is not None:
                lo, hi = layer_range
                if span[1] <= lo or span[0] >= hi:
                    continue
            out.append(f)
        return out

    def fetch(
        self,
        manifest: dict,
        get_chunk: Callable[[str, int], bytes],
        layer_range: Optional[Tuple[int, int]] = None,
        max_retries: int = 2,
    ) -> str:
        """get_chunk(name, idx) -> bytes (over whatever transport). Returns
        the completed version directory. Raises RefitError when a chunk keeps
        failing its checksum (corrupted / malicious source)."""
        vdir = os.path.join(self.dest_root, f"v{manifest['version']}")
        tmp = vdir + ".partial"
        os.makedirs(tmp, exist_ok=True)
        for f in self.wanted_files(manifest, layer_range):
            path = os.path.join(tmp, f["name"])
            with open(path, "wb") as out:
                for idx, ch in enumerate(f["chunks"]):
                    ok = False
                    for attempt in range(max_retries + 1):
                        data = get_chunk(f["name"], idx)
                        got = hashlib.sha256(data).hexdigest()
                        if got == ch["cid"] and len(data) == ch["size"]:
                            out.write(data)
                            ok = True
                            break
                        logger.warning(
                            "refit chunk %s[%d] checksum mismatch "
                            "(attempt %d): %s != %s",
                            f["name"], idx, attempt, got[:12], ch["cid"][:12],
                        )
                    if not ok:
                        shutil.rmtree(tmp, ignore_errors=True)
                        raise RefitError(
                            f"chunk {f['name']}[{idx}] failed checksum "
                            f"after {max_retries + 1} attempts"
                        )
        if os.path.exists(vdir):
            shutil.rmtree(vdir)
        os.rename(tmp, vdir)
        self.gc()
        return vdir

    def gc(self) -> List[str]:
        """Keep the newest keep_versions version dirs (reference keeps 3,
        sglang/model_runner.py:434-446)."""
        vers = []
        for name in os.listdir(self.dest_root):
            if name.startswith("v") and name[1:].isdigit():
                vers.append(int(name[1:]))
        vers.sort(reverse=True)
        removed = []
        for v in vers[self.keep_versions:]:
            path = os.path.join(self.dest_root, f"v{v}")
            shutil.rmtree(path, ignore_errors=True)
            removed.append(path)
            logger.info("refit GC: removed %s", path)
        return removed


# -- transport plumbing ------------------------------------------------------------
#
# Packet framing (msgpack, same wire as the pipeline packets): the publisher
# answers {"kind": "refit_get", ...} with {"kind": "refit_chunk", ...}; the
# trigger side pushes {"kind": "refit_manifest", ...} to start a fetch.


def transport_get_chunk(
    transport, publisher_peer: str, reply_inbox, timeout: float = 30.0
) -> Callable[[str, int], bytes]:
    """Build a get_chunk() that round-trips over a Transport. reply_inbox is a
    queue.Queue the owner fills with decoded refit_chunk messages."""
    import msgpack

    def get(name: str, idx: int) -> bytes:
        transport.send(
            publisher_peer,
            msgpack.packb(
                {"kind": "refit_get", "name": name, "idx": idx,
                 "reply_to": getattr(transport, "peer_id", "")},
                use_bin_type=True,
            ),
        )
        msg = reply_inbox.get(timeout=timeout)
        if msg.get("name") != name or msg.get("idx") != idx:
            raise RefitError(
                f"out-of-order refit chunk: wanted {name}[{idx}], "
                f"got {msg.get('name')}[{msg.get('idx')}]"
            )
        return msg["data"]

    return get


def answer_refit_get(publisher: RefitPublisher, transport, msg: dict) -> None:
    """Publisher-side handler for a decoded refit_get packet."""
    import msgpack

    data = publisher.get_chunk(msg["name"], msg["idx"])
    transport.send(
        msg["reply_to"],
        msgpack.packb(
            {"kind": "refit_chunk", "name": msg["name"], "idx": msg["idx"],
             "data": data},
            use_bin_type=True,
        ),
    )
