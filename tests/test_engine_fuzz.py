"""Property-based lifecycle fuzz: random interleavings of submit / step /
abort across variable prompt lengths, stop tokens, eos sensitivity and the
async pipelining envelope must preserve the engine's resource invariants:

- every request terminates with a finish reason;
- emitted token streams contain no placeholders;
- all KV blocks return to the pool once everything finishes (prefix cache
  disabled so retention is not expected);
- no requests remain running/waiting and no deferred frees are stranded.

(The reference relies on architecture for concurrency safety — SURVEY §5
"race detection: none"; this fuzz is the equivalent safety net here.)
"""

import pytest
import torch

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from parallax_amd.models.config import ModelConfig
from parallax_amd.parallel.comm import CommContext
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

NUM_BLOCKS = 96


def _engine(seed: int, prefix_cache: bool = False) -> Engine:
    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=97, hidden_size=32,
        num_layers=2, num_heads=2, num_kv_heads=1, head_dim=16,
        intermediate_size=64, max_position_embeddings=256,
        eos_token_ids=[5],
    )
    comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                       pp_rank=0, tp_rank=0, device=torch.device("cpu"))
    return Engine(
        cfg,
        EngineArgs(block_size=8, num_kv_blocks=NUM_BLOCKS,
                   dtype=torch.float32, seed=seed,
                   enable_prefix_cache=prefix_cache, max_batch_size=8),
        comm=comm, random_weights=True,
    )


req_strategy = st.fixed_dictionaries({
    "prompt_len": st.integers(min_value=1, max_value=24),
    "max_new": st.integers(min_value=1, max_value=10),
    "ignore_eos": st.booleans(),
    "stop_tok": st.sampled_from([None, 5, 17, 42]),
    "temperature": st.sampled_from([0.0, 0.8]),
    "abort_after": st.sampled_from([None, None, None, 0, 2, 5]),
    # a json_schema request rides the sync path; mixing it with async-
    # eligible requests churns batch membership through both paths
    "grammar": st.sampled_from([False, False, False, True]),
})

TINY_SCHEMA = '{"type": "object", "properties": {"k": {"type": "boolean"}}}'


def _sp(r):
    import json  # noqa: F401

    return SamplingParams(
        temperature=r["temperature"],
        max_new_tokens=r["max_new"],
        ignore_eos=r["ignore_eos"],
        stop_token_ids=[r["stop_tok"]] if r["stop_tok"] else [],
        json_schema=TINY_SCHEMA if r.get("grammar") else None,
    )


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(reqs=st.lists(req_strategy, min_size=1, max_size=6),
       submit_gaps=st.lists(st.integers(min_value=0, max_value=3),
                            min_size=6, max_size=6),
       seed=st.integers(min_value=0, max_value=3))
def test_lifecycle_invariants(reqs, submit_gaps, seed):
    eng = _engine(seed)
    if any(r.get("grammar") for r in reqs):
        eng.set_grammar_vocab(
            [""] + ['{"k":', "true", "false", "}", '{"k"', ":t", "rue}"]
            + [f"<{i}>" for i in range(90)]
        )
    free0 = eng.cache_manager.num_free_blocks
    assert free0 == NUM_BLOCKS

    pending = list(enumerate(reqs))
    tokens = {}
    finish = {}
    abort_at = {}
    submitted = set()
    steps = 0
    while (pending or eng.has_work) and steps < 600:
        if pending and (steps % (1 + submit_gaps[pending[0][0] % 6]) == 0):
            i, r = pending.pop(0)
            rid = f"f{i}"
            eng.submit(list(range(3, 3 + r["prompt_len"])), _sp(r), rid=rid)
            submitted.add(rid)
            if r["abort_after"] is not None:
                abort_at[rid] = steps + r["abort_after"]
        for rid, when in list(abort_at.items()):
            if steps >= when:
                eng.abort(rid)
                del abort_at[rid]
        for out in eng.step():
            if out.token_id >= 0:
                tokens.setdefault(out.rid, []).append(out.token_id)
            if out.finished:
                finish[out.rid] = out.finish_reason
        steps += 1
    assert steps < 600, "engine failed to drain"

    # every submitted request terminated with a reason
    assert set(finish) == submitted
    assert all(r is not None for r in finish.values())
    # no placeholder tokens leaked to consumers
    for toks in tokens.values():
        assert all(t >= 0 for t in toks)
    # stop-token semantics: a stream ending in a stop finish ends AT the stop
    for i, r in enumerate(reqs):
        rid = f"f{i}"
        if finish.get(rid) == "stop" and r["stop_tok"] is not None \
                and tokens.get(rid):
            last = tokens[rid][-1]
            assert last == r["stop_tok"] or last in (5,)
    # all resources returned
    assert eng.cache_manager.num_free_blocks == free0
    assert not eng.scheduler.running and not eng.scheduler.wait_queue
    assert eng._inflight is None and not eng._deferred_free


@settings(max_examples=20, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(reqs=st.lists(req_strategy, min_size=1, max_size=5),
       seed=st.integers(min_value=0, max_value=2),
       share_prefix=st.booleans())
def test_async_pipelining_equivalence(reqs, seed, share_prefix):
    """For any mix of eos-sensitive / stop-token GREEDY requests, the
    async-pipelined engine must emit exactly the sync engine's tokens and
    finish reasons (the one-step-late rollback is unobservable). Sampled
    requests are excluded here: a zombie draw for a late-finished request
    legitimately consumes RNG state, so sampled streams can differ from a
    hypothetical fully-sync run (within-envelope sampled equivalence is
    covered by test_async_decode.test_async_matches_sync_sampled)."""
    from parallax_amd.server import engine as engine_mod

    def run(async_on):
        old = engine_mod.ASYNC_DECODE
        engine_mod.ASYNC_DECODE = async_on
        try:
            eng = _engine(seed, prefix_cache=share_prefix)
            sps, prompts, rids = [], [], []
            for i, r in enumerate(reqs):
                sps.append(SamplingParams(
                    temperature=0.0,
                    max_new_tokens=r["max_new"],
                    ignore_eos=r["ignore_eos"],
                    stop_token_ids=[r["stop_tok"]] if r["stop_tok"] else [],
                ))
                if share_prefix:
                    # same long base + unique tail: later requests admit
                    # with most tokens prefix-cache hit (needs an engine
                    # with the radix enabled — see below)
                    prompts.append(list(range(3, 27)) + [60 + len(prompts)])
                else:
                    prompts.append(list(range(3, 3 + r["prompt_len"])))
            for i, (p, sp) in enumerate(zip(prompts, sps)):
                eng.submit(p, sp, rid=f"e{i}")
            tokens, finish = {}, {}
            for _ in range(400):
                if not eng.has_work:
                    break
                for out in eng.step():
                    if out.token_id >= 0:
                        tokens.setdefault(out.rid, []).append(out.token_id)
                    if out.finished:
                        finish[out.rid] = out.finish_reason
            assert not eng.has_work
            return tokens, finish
        finally:
            engine_mod.ASYNC_DECODE = old

    tokens_a, finish_a = run(True)
    tokens_b, finish_b = run(False)
    assert tokens_a == tokens_b
    assert finish_a == finish_b


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(reqs=st.lists(req_strategy, min_size=2, max_size=6),
       share_prefix=st.booleans(),
       seed=st.integers(min_value=0, max_value=2))
def test_prefix_cache_accounting(reqs, share_prefix, seed):
    """With the radix prefix cache ENABLED: after everything drains, every
    block is either free or held by the radix (free + cached = pool), and
    same-prefix requests must not corrupt accounting across async finishes."""
    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=97, hidden_size=32,
        num_layers=2, num_heads=2, num_kv_heads=1, head_dim=16,
        intermediate_size=64, max_position_embeddings=256,
        eos_token_ids=[5],
    )
    comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                       pp_rank=0, tp_rank=0, device=torch.device("cpu"))
    eng = Engine(
        cfg,
        EngineArgs(block_size=8, num_kv_blocks=NUM_BLOCKS,
                   dtype=torch.float32, seed=seed, max_batch_size=8),
        comm=comm, random_weights=True,
    )
    finish = {}
    for i, r in enumerate(reqs):
        sp = SamplingParams(
            temperature=r["temperature"], max_new_tokens=r["max_new"],
            ignore_eos=r["ignore_eos"],
            stop_token_ids=[r["stop_tok"]] if r["stop_tok"] else [],
        )
        base = list(range(3, 27)) if share_prefix \
            else list(range(3 + i, 3 + i + r["prompt_len"]))
        eng.submit(base + [50 + i], sp, rid=f"p{i}")
    for _ in range(500):
        if not eng.has_work:
            break
        for out in eng.step():
            if out.finished:
                finish[out.rid] = out.finish_reason
    assert not eng.has_work
    assert len(finish) == len(reqs)
    cm = eng.cache_manager
    assert cm.allocator.num_free_blocks + cm.radix.num_cached_blocks \
        == NUM_BLOCKS


@settings(max_examples=10, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(reqs=st.lists(req_strategy, min_size=1, max_size=4),
       n_stages=st.sampled_from([2, 3]))
def test_p2p_pipeline_fuzz(reqs, n_stages):
    """Random request mixes over a loopback PeerExecutor chain: every request
    terminates, tokens match the single-host engine (greedy), and every
    peer's cache state drains."""
    from parallax_amd.p2p.peer_executor import PeerExecutor
    from parallax_amd.p2p.transport import LoopbackTransport

    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=97, hidden_size=32,
        num_layers=4, num_heads=2, num_kv_heads=1, head_dim=16,
        intermediate_size=64, max_position_embeddings=256,
        eos_token_ids=[5],
    )
    from tests.test_pipeline_parallel import full_state_dict

    # one HF-style weight set shared by the single-host engine and the chain
    sd = full_state_dict(cfg)
    comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                       pp_rank=0, tp_rank=0, device=torch.device("cpu"))
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=96,
                                 dtype=torch.float32, seed=0), comm=comm)
    for name, w in sd.items():
        eng.model.load_hf_weight(name, w)

    sps = []
    for r in reqs:
        sps.append(SamplingParams(
            temperature=0.0, max_new_tokens=r["max_new"],
            ignore_eos=r["ignore_eos"],
            stop_token_ids=[r["stop_tok"]] if r["stop_tok"] else [],
        ))
    prompts = [list(range(3, 3 + r["prompt_len"])) for r in reqs]
    expected = eng.generate(prompts, sps)

    registry = {}
    peer_ids = [f"fz{i}" for i in range(n_stages)]
    base, rem = divmod(cfg.num_layers, n_stages)
    spans, pos = [], 0
    for i in range(n_stages):
        n = base + (1 if i < rem else 0)
        spans.append((pos, pos + n))
        pos += n
    peers = []
    for pid, (s0, e0) in zip(peer_ids, spans):
        t = LoopbackTransport(pid, registry)
        px = PeerExecutor(cfg, s0, e0, pid, t, dtype=torch.float32,
                          num_kv_blocks=96, block_size=8, seed=0)
        for name, w in sd.items():
            px.model.load_hf_weight(name, w)
        peers.append(px)
    head = peers[0]
    rids = [head.submit(p, sp, peer_ids) for p, sp in zip(prompts, sps)]
    got = {rid: [] for rid in rids}
    done = set()
    for _ in range(3000):
        for px in peers:
            px.step(recv_timeout=0.0005)
        for out in head.drain_outputs():
            if out.token_id >= 0:
                got[out.rid].append(out.token_id)
            if out.finished:
                done.add(out.rid)
        if len(done) == len(rids):
            break
    assert len(done) == len(rids), "pipeline failed to drain"
    assert [got[r] for r in rids] == list(expected.values())
    for _ in range(30):  # let release packets land
        for px in peers:
            px.step(recv_timeout=0.0005)
    for px in peers[1:]:
        assert not px._peer_positions
