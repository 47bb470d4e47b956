"""Continuous-batching serving engine (lstm_tensorspark_amd/serving.py):
the correctness bar is EXACT token parity with solo model.generate() for
every request, regardless of what else shares the batch or when it joins.
"""
import threading

import torch

from lstm_tensorspark_amd.config import ModelConfig
from lstm_tensorspark_amd.models.lstm import LSTMModel
from lstm_tensorspark_amd.serving import ContinuousBatcher


def _model(seed=12, vocab=48, hidden=24, layers=2):
    torch.manual_seed(seed)
    cfg = ModelConfig(input_size=hidden, hidden_size=hidden,
                      num_layers=layers, head="lm", vocab_size=vocab)
    return LSTMModel(cfg)


def _solo(m, tokens, steps):
    prompt = torch.tensor(tokens, dtype=torch.long).unsqueeze(1)
    return m.generate(prompt, steps=steps, temperature=0.0)[:, 0].tolist()


def test_batched_requests_match_solo_greedy():
    """Heterogeneous prompts/lengths submitted together: each result must
    equal its solo greedy decode token-for-token."""
    m = _model()
    eng = ContinuousBatcher(m)
    reqs = [([1, 2, 3], 7), ([5], 4), ([9, 8, 7, 6, 5], 9), ([40, 2], 1)]
    rids = [eng.submit(t, s) for t, s in reqs]
    eng.drain()
    for rid, (toks, steps) in zip(rids, reqs):
        got = eng.result(rid, timeout=0)
        assert got == _solo(m, toks, steps), (toks, steps)
    st = eng.stats()
    assert (st["active"], st["queued"], st["done_unclaimed"]) == (0, 0, 0)
    assert st["decode_steps"] > 0 and st["mean_occupancy"] >= 1.0


def test_midflight_join_and_leave():
    """A request admitted while another is mid-decode (and one leaving
    early) must not perturb anyone's tokens."""
    m = _model(seed=13)
    eng = ContinuousBatcher(m)
    ra = eng.submit([1, 2], 10)
    for _ in range(3):
        eng.step()                      # A is 3 tokens in
    rb = eng.submit([3, 4, 5], 4)       # joins mid-flight, leaves early
    rc = eng.submit([7], 12)
    eng.drain()
    assert eng.result(ra, timeout=0) == _solo(m, [1, 2], 10)
    assert eng.result(rb, timeout=0) == _solo(m, [3, 4, 5], 4)
    assert eng.result(rc, timeout=0) == _solo(m, [7], 12)


def test_max_batch_queueing():
    """More requests than max_batch: the overflow queues and still
    finishes correctly."""
    m = _model(seed=14)
    eng = ContinuousBatcher(m, max_batch=2)
    reqs = [([i + 1], 3 + i) for i in range(5)]
    rids = [eng.submit(t, s) for t, s in reqs]
    eng.step()
    assert eng.stats()["active"] <= 2 and eng.stats()["queued"] >= 2
    eng.drain()
    for rid, (toks, steps) in zip(rids, reqs):
        assert eng.result(rid, timeout=0) == _solo(m, toks, steps)


def test_engine_thread_concurrent_submitters():
    """Background engine thread + concurrent submitter threads: every
    result still matches solo greedy."""
    m = _model(seed=15)
    eng = ContinuousBatcher(m)
    eng.start()
    reqs = [([i + 1, i + 2], 3 + (i % 4)) for i in range(8)]
    results = {}
    lock = threading.Lock()

    def worker(i):
        toks, steps = reqs[i]
        rid = eng.submit(toks, steps)
        out = eng.result(rid, timeout=30)
        with lock:
            results[i] = out

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    eng.stop()
    for i, (toks, steps) in enumerate(reqs):
        assert results[i] == _solo(m, toks, steps), i


def test_submit_validation():
    m = _model()
    eng = ContinuousBatcher(m, max_steps=16)
    import pytest
    with pytest.raises(ValueError):
        eng.submit([], 4)
    with pytest.raises(ValueError):
        eng.submit([999], 4)          # out of vocab
    with pytest.raises(ValueError):
        eng.submit([1], 17)           # over max_steps


def test_serving_endpoint_with_batcher():
    """/generate through the HTTP handler in continuous mode returns the
    same tokens as solo greedy."""
    from fastapi.testclient import TestClient
    from examples.serve import build_app

    m = _model(seed=16)
    eng = ContinuousBatcher(m)
    eng.start()
    try:
        app = build_app(m, torch.device("cpu"), batcher=eng)
        client = TestClient(app)
        h = client.get("/health").json()
        assert h["continuous_batching"] and "engine" in h
        assert h["hidden"] == 24 and h["layers"] == 2
        r = client.post("/generate", json={"tokens": [1, 2, 3], "steps": 5})
        assert r.status_code == 200
        assert r.json()["tokens"] == _solo(m, [1, 2, 3], 5)
        # non-greedy requests bypass the batcher and still work
        r2 = client.post("/generate", json={"tokens": [1], "steps": 3,
                                            "temperature": 1.0})
        assert r2.status_code == 200
    finally:
        eng.stop()


def test_eos_early_stop_matches_generate():
    """Per-request eos: the batcher stops a request at its eos token
    exactly like model.generate(eos_token=...), and neighbors continue."""
    m = _model(seed=17)
    eng = ContinuousBatcher(m)
    # pick an eos that the greedy path actually emits: the 2nd generated
    # token of this prompt
    probe = _solo(m, [1, 2], 6)
    eos = probe[3]
    prompt = torch.tensor([1, 2], dtype=torch.long).unsqueeze(1)
    solo = m.generate(prompt, steps=6, eos_token=eos)[:, 0].tolist()
    assert solo[-1] == eos and len(solo) < 8   # it really stopped early
    ra = eng.submit([1, 2], 6, eos_token=eos)
    rb = eng.submit([5, 6], 5)                 # no eos, full length
    eng.drain()
    assert eng.result(ra, timeout=0) == solo
    assert eng.result(rb, timeout=0) == _solo(m, [5, 6], 5)


def test_sampled_requests_match_solo_seeded():
    """A temperature request in the batch samples with its own seeded
    generator and reproduces the solo seeded decode exactly -- even while
    sharing steps with greedy neighbors."""
    m = _model(seed=18)
    eng = ContinuousBatcher(m)
    prompt = torch.tensor([2, 3], dtype=torch.long).unsqueeze(1)
    g = torch.Generator().manual_seed(77)
    solo = m.generate(prompt, steps=6, temperature=0.8, top_k=8,
                      generator=g)[:, 0].tolist()
    ra = eng.submit([2, 3], 6, temperature=0.8, top_k=8, seed=77)
    rb = eng.submit([4], 4)                      # greedy neighbor
    eng.drain()
    assert eng.result(ra, timeout=0) == solo
    assert eng.result(rb, timeout=0) == _solo(m, [4], 4)


def test_streaming_endpoint_tokens_match_solo():
    """SSE /generate_stream: concatenated streamed tokens == the solo
    greedy continuation, with a [DONE] terminator."""
    import json
    from fastapi.testclient import TestClient
    from examples.serve import build_app

    m = _model(seed=19)
    eng = ContinuousBatcher(m)
    eng.start()
    try:
        app = build_app(m, torch.device("cpu"), batcher=eng)
        client = TestClient(app)
        got, done = [], False
        with client.stream("POST", "/generate_stream",
                           json={"tokens": [3, 1], "steps": 6}) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[len("data: "):]
                if payload == "[DONE]":
                    done = True
                    break
                got.extend(json.loads(payload))
        assert done
        assert got == _solo(m, [3, 1], 6)[2:]
    finally:
        eng.stop()


def test_batcher_random_schedules_property():
    """Randomized schedules (hypothesis): arbitrary prompts, lengths and
    join times -- every request must still match its solo greedy decode."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    m = _model(seed=20)

    @st.composite
    def schedule(draw):
        n = draw(st.integers(1, 6))
        reqs = []
        for _ in range(n):
            plen = draw(st.integers(1, 6))
            prompt = [draw(st.integers(0, 47)) for _ in range(plen)]
            steps = draw(st.integers(1, 8))
            join_after = draw(st.integers(0, 5))   # engine steps before submit
            reqs.append((prompt, steps, join_after))
        return reqs

    @given(schedule())
    @settings(max_examples=20, deadline=None)
    def run(reqs):
        eng = ContinuousBatcher(m, max_batch=3)
        pending = sorted(enumerate(reqs), key=lambda kv: kv[1][2])
        rids = {}
        t = 0
        while pending or eng.stats()["active"] or eng.stats()["queued"]:
            while pending and pending[0][1][2] <= t:
                i, (prompt, steps, _) = pending.pop(0)
                rids[i] = eng.submit(prompt, steps)
            eng.step()
            t += 1
        for i, (prompt, steps, _) in enumerate(reqs):
            assert eng.result(rids[i], timeout=0) == _solo(m, prompt, steps)

    run()


def test_batcher_bench_tool_degenerate():
    """tools/batcher_bench.py runs on CPU (one GPU lease away from the r3
    multi-tenant throughput measurement)."""
    from tools.batcher_bench import main
    r = main(["--hidden", "32", "--vocab", "64", "--layers", "1",
              "--clients", "3", "--steps", "3", "--prompt", "2",
              "--requests", "6", "--solo-baseline", "1"])
    assert r["requests"] == 6 and r["req_per_s"] > 0
    assert r["p50_ms"] > 0 and r["solo_ms"] > 0


def test_cancel_queued_live_and_finished():
    """cancel(): dequeues a waiting request, retires a live one at the
    next step, drops an unclaimed result -- neighbors unaffected."""
    m = _model(seed=22)
    eng = ContinuousBatcher(m, max_batch=2)
    ra = eng.submit([1, 2], 8)
    rb = eng.submit([3], 6)
    rq = eng.submit([5], 4)            # queued (max_batch=2)
    eng.step()
    assert eng.cancel(rq)              # still queued: dequeued
    eng.step()
    assert eng.cancel(ra)              # live: retires next boundary
    eng.drain()
    assert eng.try_result(ra) is None  # cancelled -> no result stored
    assert eng.try_result(rq) is None
    assert eng.result(rb, timeout=0) == _solo(m, [3], 6)  # unaffected
    rc = eng.submit([7], 2)
    eng.drain()
    assert eng.cancel(rc)              # finished: unclaimed result dropped
    assert eng.try_result(rc) is None
    assert not eng.cancel(12345)       # unknown rid
