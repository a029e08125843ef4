"""Runtime weight refit: reload from disk changes outputs and resets the
prefix cache (reference §3.5 weight-refit path)."""

import json
import os

import pytest
import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

transformers = pytest.importorskip("transformers")
safetensors = pytest.importorskip("safetensors")


def save_checkpoint(tmpdir, hf, hf_cfg):
    from safetensors.torch import save_file

    os.makedirs(tmpdir, exist_ok=True)
    with open(os.path.join(tmpdir, "config.json"), "w") as f:
        json.dump(hf_cfg.to_dict() | {"architectures": ["LlamaForCausalLM"]}, f)
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
              os.path.join(tmpdir, "model.safetensors"))


def test_refit_changes_output_and_resets_prefix_cache(tmp_path):
    torch.manual_seed(0)
    hf_cfg = transformers.LlamaConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, tie_word_embeddings=False,
    )
    hf_a = transformers.LlamaForCausalLM(hf_cfg).eval()
    torch.manual_seed(999)
    hf_b = transformers.LlamaForCausalLM(hf_cfg).eval()
    ckpt_a, ckpt_b = str(tmp_path / "a"), str(tmp_path / "b")
    save_checkpoint(ckpt_a, hf_a, hf_cfg)
    save_checkpoint(ckpt_b, hf_b, hf_cfg)

    cfg = ModelConfig.from_pretrained(ckpt_a)
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32), model_path=ckpt_a)
    prompt = list(range(10, 40))
    sp = SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)
    out_a = list(eng.generate([prompt], [sp]).values())[0]
    assert eng.cache_manager.radix.num_cached_blocks > 0  # prefix published

    n = eng.update_weights_from_disk(ckpt_b)
    assert n > 0
    assert eng.cache_manager.radix.num_cached_blocks == 0  # cache invalidated
    out_b = list(eng.generate([prompt], [sp]).values())[0]
    assert out_a != out_b  # new weights, new tokens

    # and back: refit restores the original behavior
    eng.update_weights_from_disk(ckpt_a)
    out_a2 = list(eng.generate([prompt], [sp]).values())[0]
    assert out_a2 == out_a


def test_shard_loader_selective_files(tmp_path):
    """selective_file_list picks only the shards covering a layer range."""
    from parallax_amd.server.shard_loader import selective_file_list

    idx = {
        "weight_map": {
            "model.embed_tokens.weight": "a.safetensors",
            "model.layers.0.mlp.up_proj.weight": "a.safetensors",
            "model.layers.1.mlp.up_proj.weight": "b.safetensors",
            "model.layers.2.mlp.up_proj.weight": "c.safetensors",
            "lm_head.weight": "c.safetensors",
        }
    }
    os.makedirs(tmp_path / "m", exist_ok=True)
    with open(tmp_path / "m" / "model.safetensors.index.json", "w") as f:
        json.dump(idx, f)
    files = selective_file_list(str(tmp_path / "m"), 1, 2)
    assert "b.safetensors" in files and "a.safetensors" in files  # endpoints travel


def test_http_update_weights_endpoint(tmp_path):
    """POST /update_weights reloads the checkpoint through the serving stack
    (reference /weight/refit -> node reload path, single-host variant)."""
    from fastapi.testclient import TestClient

    from parallax_amd.server.engine_server import EngineServer
    from parallax_amd.server.http_frontend import create_app
    from parallax_amd.server.tokenizer_util import TokenizerWrapper

    torch.manual_seed(5)
    hf_cfg = transformers.LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=64,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        tie_word_embeddings=False,
    )
    hf = transformers.LlamaForCausalLM(hf_cfg).eval()
    ckpt = str(tmp_path / "ckpt")
    save_checkpoint(ckpt, hf, hf_cfg)
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["LlamaForCausalLM"]}
    )
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=64,
                                 dtype=torch.float32),
                 model_path=ckpt)
    server = EngineServer(eng)
    server.start()
    tok = TokenizerWrapper(vocab_size=cfg.vocab_size)
    app = create_app(server, tok, "tiny")
    try:
        with TestClient(app) as c:
            r = c.post("/update_weights", json={"model_path": ckpt})
            assert r.status_code == 200
            assert r.json()["updated_tensors"] > 0
    finally:
        server.stop()


# -- chunked refit distribution over the P2P transport (reference
#    p2p/server.py:224-338 CID-checksummed block store + 3-version GC) -----------


def _tiny_cfg():
    return ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=128, hidden_size=32,
        num_layers=2, num_heads=2, num_kv_heads=2, head_dim=16,
        intermediate_size=64, max_position_embeddings=128, eos_token_ids=[],
    )


def _save_tiny_ckpt(tmpdir, seed):
    """Synthetic sharded checkpoint: per-layer safetensors + index map."""
    from safetensors.torch import save_file

    os.makedirs(tmpdir, exist_ok=True)
    g = torch.Generator().manual_seed(seed)
    cfg = _tiny_cfg()
    weight_map = {}
    sd0, sd1, shared = {}, {}, {}
    def rand(*s):
        return torch.randn(*s, generator=g) * 0.05
    for i, sd in enumerate((sd0, sd1)):
        p = f"model.layers.{i}."
        hd = cfg.num_heads * cfg.head_dim
        kvd = cfg.num_kv_heads * cfg.head_dim
        sd[p + "self_attn.q_proj.weight"] = rand(hd, cfg.hidden_size)
        sd[p + "self_attn.k_proj.weight"] = rand(kvd, cfg.hidden_size)
        sd[p + "self_attn.v_proj.weight"] = rand(kvd, cfg.hidden_size)
        sd[p + "self_attn.o_proj.weight"] = rand(cfg.hidden_size, hd)
        sd[p + "mlp.gate_proj.weight"] = rand(cfg.intermediate_size, cfg.hidden_size)
        sd[p + "mlp.up_proj.weight"] = rand(cfg.intermediate_size, cfg.hidden_size)
        sd[p + "mlp.down_proj.weight"] = rand(cfg.hidden_size, cfg.intermediate_size)
        sd[p + "input_layernorm.weight"] = torch.ones(cfg.hidden_size)
        sd[p + "post_attention_layernorm.weight"] = torch.ones(cfg.hidden_size)
    shared["model.embed_tokens.weight"] = rand(cfg.vocab_size, cfg.hidden_size)
    shared["model.norm.weight"] = torch.ones(cfg.hidden_size)
    shared["lm_head.weight"] = rand(cfg.vocab_size, cfg.hidden_size)
    for fname, sd in (("layer0.safetensors", sd0), ("layer1.safetensors", sd1),
                      ("shared.safetensors", shared)):
        save_file(sd, os.path.join(tmpdir, fname))
        for k in sd:
            weight_map[k] = fname
    with open(os.path.join(tmpdir, "model.safetensors.index.json"), "w") as f:
        json.dump({"weight_map": weight_map}, f)
    return cfg


def test_refit_chunks_over_tcp_transport(tmp_path):
    """Two peers over real TCP: the fetcher pulls only its layer range's
    files, verifies every chunk CID, reassembles and hot-reloads; a corrupted
    chunk stream is rejected with RefitError."""
    import threading

    from parallax_amd.p2p.peer_executor import PeerExecutor
    from parallax_amd.p2p.refit import RefitError, RefitFetcher, RefitPublisher
    from parallax_amd.p2p.transport import TcpTransport

    ckpt = str(tmp_path / "ckpt_v1")
    cfg = _save_tiny_ckpt(ckpt, seed=7)

    t_pub = TcpTransport("pub", host="127.0.0.1")
    t_sub = TcpTransport("sub", host="127.0.0.1")
    t_pub.set_peer_addr("sub", "127.0.0.1", t_sub.port)
    t_sub.set_peer_addr("pub", "127.0.0.1", t_pub.port)
    try:
        publisher = RefitPublisher(ckpt, version=1, chunk_size=1024)
        pub_peer = PeerExecutor(cfg, 0, 1, "pub", t_pub, random_weights=True)
        pub_peer.set_refit_publisher(publisher)
        sub_peer = PeerExecutor(cfg, 1, 2, "sub", t_sub, random_weights=True)

        stop = threading.Event()

        def pump():
            while not stop.is_set():
                pub_peer.step(recv_timeout=0.05)

        th = threading.Thread(target=pump, daemon=True)
        th.start()
        before = {k: v.clone() for k, v in sub_peer.model.state_dict().items()}
        vdir = sub_peer.refit_from_peer(
            "pub", publisher.manifest, str(tmp_path / "versions"), timeout=20
        )
        stop.set()
        th.join(timeout=5)

        # layer-range filter: only layer1 + shared files fetched
        got = sorted(os.listdir(vdir))
        assert "layer1.safetensors" in got and "shared.safetensors" in got
        assert "layer0.safetensors" not in got
        # weights actually changed to the checkpoint's values
        after = sub_peer.model.state_dict()
        changed = any(
            not torch.equal(before[k], after[k]) for k in before
        )
        assert changed

        # corrupted stream -> RefitError (checksum rejection after retries)
        bad = RefitFetcher(str(tmp_path / "bad"))
        def corrupt_get(name, idx):
            data = bytearray(publisher.get_chunk(name, idx))
            data[0] ^= 0xFF
            return bytes(data)
        with pytest.raises(RefitError):
            bad.fetch(publisher.manifest, corrupt_get, layer_range=(1, 2))
    finally:
        t_pub.close()
        t_sub.close()


def test_refit_version_gc(tmp_path):
    """The fetcher keeps only the newest 3 version directories."""
    from parallax_amd.p2p.refit import RefitFetcher, RefitPublisher

    ckpt = str(tmp_path / "ck")
    _save_tiny_ckpt(ckpt, seed=1)
    fetcher = RefitFetcher(str(tmp_path / "vers"))
    for v in range(1, 6):
        pub = RefitPublisher(ckpt, version=v, chunk_size=2048)
        fetcher.fetch(pub.manifest, pub.get_chunk)
    kept = sorted(os.listdir(str(tmp_path / "vers")))
    assert kept == ["v3", "v4", "v5"]


def test_refit_manifest_push_triggers_reload(tmp_path):
    """A pushed refit_manifest packet makes the peer fetch its shard's chunks
    and hot-reload on its next step (reference heartbeat-triggered refit)."""
    import threading

    import msgpack

    from parallax_amd.p2p.peer_executor import PeerExecutor
    from parallax_amd.p2p.refit import RefitPublisher
    from parallax_amd.p2p.transport import LoopbackTransport

    ckpt = str(tmp_path / "ckpt")
    cfg = _save_tiny_ckpt(ckpt, seed=11)
    registry = {}
    t_pub = LoopbackTransport("pub", registry)
    t_sub = LoopbackTransport("sub", registry)
    publisher = RefitPublisher(ckpt, version=3, chunk_size=512)
    pub_peer = PeerExecutor(cfg, 0, 1, "pub", t_pub, random_weights=True)
    pub_peer.set_refit_publisher(publisher)
    sub_peer = PeerExecutor(cfg, 1, 2, "sub", t_sub, random_weights=True,
                            refit_dir=str(tmp_path / "vers"))
    before = {k: v.clone() for k, v in sub_peer.model.state_dict().items()}

    stop = threading.Event()

    def pump():
        while not stop.is_set():
            pub_peer.step(recv_timeout=0.02)

    th = threading.Thread(target=pump, daemon=True)
    th.start()
    t_pub.send("sub", msgpack.packb(
        {"kind": "refit_manifest", "manifest": publisher.manifest,
         "publisher": "pub"}, use_bin_type=True))
    sub_peer.step(recv_timeout=0.2)   # receive the manifest
    sub_peer.step(recv_timeout=0.05)  # consume: fetch + reload
    stop.set()
    th.join(timeout=5)
    after = sub_peer.model.state_dict()
    assert any(not torch.equal(before[k], after[k]) for k in before)
    import os as _os

    assert _os.path.isdir(str(tmp_path / "vers" / "v3"))
