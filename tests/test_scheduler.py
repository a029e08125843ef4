"""Continuous-batching scheduler tests (CPU, no model) — parity with the
reference's test_batch_scheduler.py coverage."""

from parallax_amd.server.cache_manager import CacheManager
from parallax_amd.server.request import InitialRequest, RequestStatus
from parallax_amd.server.sampling_params import SamplingParams
from parallax_amd.server.scheduler import Scheduler


def make_sched(num_blocks=64, block_size=4, **kw):
    cm = CacheManager(block_size=block_size, num_blocks=num_blocks)
    return Scheduler(cm, **kw), cm


def req(rid, n_prompt=6, max_new=4, **kw):
    return InitialRequest(
        rid=rid,
        prompt_token_ids=list(range(1, n_prompt + 1)),
        sampling_params=SamplingParams(max_new_tokens=max_new, ignore_eos=True, **kw),
    )


def test_admit_and_prefill_first():
    s, _ = make_sched()
    s.add_request(req("a"))
    s.add_request(req("b"))
    assert s.admit_requests() == 2
    batch = s.form_batch()
    assert len(batch.prefill_chunks) == 2 and not batch.decode_reqs


def test_prefill_to_decode_transition():
    s, _ = make_sched()
    s.add_request(req("a", n_prompt=6, max_new=3))
    s.admit_requests()
    b = s.form_batch()
    s.complete_prefill_chunk(b.prefill_chunks[0])
    assert s.commit_token("a", 42) is None
    b2 = s.form_batch()
    assert not b2.prefill_chunks and len(b2.decode_reqs) == 1
    s.commit_token("a", 43)
    fin = s.commit_token("a", 44)
    assert fin is not None and fin.status is RequestStatus.FINISHED_LENGTH
    assert s.num_running == 0


def test_chunked_prefill_page_aligned():
    s, _ = make_sched(num_blocks=256, block_size=4)
    s.prefill_chunk_size = 8
    s.add_request(req("a", n_prompt=19))
    s.admit_requests()
    chunks = []
    while True:
        b = s.form_batch()
        if not b.prefill_chunks:
            break
        c = b.prefill_chunks[0]
        chunks.append((c.start, c.num_tokens))
        s.complete_prefill_chunk(c)
    assert chunks == [(0, 8), (8, 8), (16, 3)]
    # non-final chunks page aligned
    assert all(start % 4 == 0 for start, _ in chunks)


def test_kv_budget_blocks_admission():
    s, cm = make_sched(num_blocks=4, block_size=4)  # 16 tokens of KV
    s.add_request(req("a", n_prompt=8))   # needs 2 blocks + slack
    s.add_request(req("b", n_prompt=8))
    assert s.admit_requests() == 1
    assert s.num_running == 1 and len(s.wait_queue) == 1


def test_token_budget_limits_batch():
    s, _ = make_sched(num_blocks=1024)
    s.max_num_tokens_per_batch = 10
    s.add_request(req("a", n_prompt=8))
    s.add_request(req("b", n_prompt=8))
    s.admit_requests()
    b = s.form_batch()
    assert sum(c.num_tokens for c in b.prefill_chunks) <= 10


def test_eos_finish():
    s, _ = make_sched()
    r = req("a", max_new=10)
    r.sampling_params.ignore_eos = False
    r.eos_token_ids = [2]
    s.add_request(r)
    s.admit_requests()
    b = s.form_batch()
    s.complete_prefill_chunk(b.prefill_chunks[0])
    fin = s.commit_token("a", 2)
    assert fin is not None and fin.status is RequestStatus.FINISHED_EOS


def test_abort():
    s, _ = make_sched()
    s.add_request(req("a"))
    s.admit_requests()
    assert s.abort_request("a")
    out = s.sweep_aborted()
    assert len(out) == 1 and out[0].status is RequestStatus.FINISHED_ABORT
    assert s.num_running == 0


def test_timeout_sweep():
    s, _ = make_sched(request_timeout_s=0.0)
    s.add_request(req("a"))
    s.admit_requests()
    timed = s.sweep_timeouts()
    assert len(timed) == 1 and s.num_running == 0


def test_decode_priority_ordering():
    """prefill_priority=False forms decodes before prefills: with a tight
    token budget the decode slots win."""
    import torch

    from parallax_amd.server.cache_manager import CacheManager
    from parallax_amd.server.request import InitialRequest, RequestStatus
    from parallax_amd.server.sampling_params import SamplingParams
    from parallax_amd.server.scheduler import Scheduler

    cache = CacheManager(8, 64, enable_prefix_cache=False)
    sched = Scheduler(cache, max_num_tokens_per_batch=4,
                      prefill_priority=False)
    # one decoding request + one big pending prefill
    r1 = InitialRequest(rid="d", prompt_token_ids=[1, 2, 3],
                        sampling_params=SamplingParams(max_new_tokens=8,
                                                       ignore_eos=True))
    r2 = InitialRequest(rid="p", prompt_token_ids=list(range(3, 30)),
                        sampling_params=SamplingParams(max_new_tokens=8,
                                                       ignore_eos=True))
    sched.add_request(r1)
    sched.add_request(r2)
    sched.admit_requests()
    r1.num_prefilled_tokens = r1.prompt_len
    r1.status = RequestStatus.DECODING
    r1.commit_new_token(9)
    batch = sched.form_batch()
    assert [r.rid for r in batch.decode_reqs] == ["d"]
    # decode took 1 of the 4 budget; prefill gets a page-aligned remainder
    assert sum(c.num_tokens for c in batch.prefill_chunks) <= 3
