"""Property-based fuzzing (hypothesis): planner robustness over random
model/world shapes, paged-cache invariants under random lease/release
traffic, and chunked-CE equality at arbitrary shapes."""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from tensorlink_amd.models.configs import ModelConfig
from tensorlink_amd.parallel.planner import AssignmentError, plan_for_world


@settings(max_examples=40, deadline=None)
@given(layers=st.integers(2, 48), world=st.integers(1, 8),
       hidden=st.sampled_from([256, 512, 1024, 4096]),
       heads=st.sampled_from([4, 8, 16, 32]))
def test_planner_fuzz_valid_or_clean_error(layers, world, hidden, heads):
    """Any config either plans cleanly (full cover, ordered stages,
    embed-first/head-last) or raises AssignmentError — never crashes or
    returns a malformed plan."""
    cfg = ModelConfig(
        name=f"fuzz-{layers}-{hidden}", vocab_size=1024,
        hidden_size=hidden, intermediate_size=hidden * 3,
        num_hidden_layers=layers, num_attention_heads=heads,
        num_key_value_heads=max(1, heads // 2), head_dim=hidden // heads,
        max_position_embeddings=4096)
    try:
        plan = plan_for_world(cfg, world)
    except AssignmentError:
        return
    assert plan.num_stages == world
    covered = []
    for i in range(world):
        spec = plan.stage_for_rank(i)
        covered.extend(range(spec.layer_start, spec.layer_end))
        assert spec.has_embedding == (i == 0)
        assert spec.has_head == (i == world - 1)
    assert covered == list(range(layers))


@settings(max_examples=25, deadline=None)
@given(st.data())
def test_paged_allocator_fuzz_invariants(data):
    """Random ensure/release traffic never leaks or double-frees pages,
    and table entries always point at leased pages."""
    from tensorlink_amd.models.paged import PAGE, DynamicPagedKVCache
    from tensorlink_amd.models.configs import get_config
    slots, pool = 4, 12
    cache = DynamicPagedKVCache(1, slots, pool, PAGE * 6,
                                get_config("tiny"), torch.device("cpu"),
                                torch.float32)
    lens = [0] * slots
    for _ in range(data.draw(st.integers(5, 40))):
        slot = data.draw(st.integers(0, slots - 1))
        if data.draw(st.booleans()):
            want = data.draw(st.integers(1, PAGE * 6))
            need = (want + PAGE - 1) // PAGE - len(cache._slot_pages[slot])
            if need <= cache.allocator.n_free:
                cache.ensure(slot, want)
                lens[slot] = max(lens[slot], want)
        else:
            cache.release_slot(slot)
            lens[slot] = 0
        leased = sum(len(p) for p in cache._slot_pages)
        assert leased + cache.allocator.n_free == pool
        all_pages = [p for sp in cache._slot_pages for p in sp]
        assert len(set(all_pages)) == len(all_pages)      # no double-lease
    for s in range(slots):
        cache.release_slot(s)
    assert cache.allocator.n_free == pool


@settings(max_examples=20, deadline=None)
@given(B=st.integers(1, 3), S=st.integers(2, 20),
       V=st.integers(16, 200), chunk=st.integers(1, 64),
       pad=st.booleans())
def test_chunked_ce_fuzz(B, S, V, chunk, pad):
    """Chunked CE == full CE for arbitrary shapes/chunk sizes/padding."""
    from tensorlink_amd.ops import reference as ref
    H = 16
    hidden = torch.randn(B, S, H, requires_grad=True)
    w = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (B, S))
    if pad:
        labels[:, -1] = -100
    loss_c = ref.chunked_causal_lm_loss(hidden, w, labels, chunk=chunk)
    h2 = hidden.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    loss_f = ref.causal_lm_loss(h2 @ w2.t(), labels)
    if torch.isnan(loss_f):
        return
    loss_c.backward()
    loss_f.backward()
    assert torch.allclose(loss_c, loss_f, atol=1e-5)
    assert torch.allclose(hidden.grad, h2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)


@settings(max_examples=10, deadline=None)
@given(lens=st.lists(st.integers(3, 200), min_size=1, max_size=4),
       chunk=st.sampled_from([None, 16, 64]),
       prefix=st.booleans(), new=st.integers(1, 8),
       pool=st.sampled_from([None, 8, 12]))
def test_batcher_fuzz_matches_serial(lens, chunk, prefix, new, pool):
    """Random prompt mixes through the batcher (any chunking / prefix
    caching / pool pressure incl. preemption+eviction) always reproduce
    serial greedy outputs and drain the pool."""
    from tensorlink_amd.engine.batcher import ContinuousBatcher
    from tensorlink_amd.parallel.pipeline import (PipelineRunner,
                                                  SamplingParams)
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                       device=torch.device("cpu"))
    b = ContinuousBatcher(r, max_slots=4, max_ctx=512,
                          prefill_chunk=chunk, pool_pages=pool,
                          prefix_caching=prefix).start()
    try:
        g = torch.Generator().manual_seed(sum(lens) + new)
        prompts = [torch.randint(0, 1024, (n,), generator=g)
                   for n in lens]
        reqs = [b.submit(p.clone(), max_new_tokens=new) for p in prompts]
        outs = [rq.result(timeout=120) for rq in reqs]
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0),
                             SamplingParams(max_new_tokens=new))
            assert o == ref[0].tolist()
    finally:
        b.stop()
    free = b.cache.allocator.n_free
    if prefix:
        free += len(b.cache.lru)
    assert free == b.cache.allocator.n_pages


@settings(max_examples=30, deadline=None)
@given(N=st.integers(1, 8), groups=st.integers(1, 6),
       scale=st.floats(1e-3, 1e3))
def test_mxfp4_fuzz_bounds(N, groups, scale):
    """MXFP4 quantization error stays within the e2m1 grid bound for
    arbitrary magnitudes, and packing is exactly invertible on its own
    dequantized output."""
    from tensorlink_amd.models.quant import (dequantize_mxfp4,
                                             quantize_mxfp4)
    K = groups * 32
    w = torch.randn(N, K) * scale
    p, e = quantize_mxfp4(w)
    wq = dequantize_mxfp4(p, e)
    grp = w.reshape(N, groups, 32)
    err = (grp - wq.reshape(N, groups, 32)).abs().amax(-1)
    assert bool((err <= grp.abs().amax(-1) * 0.26 + 1e-9).all())
    p2, e2 = quantize_mxfp4(wq)
    torch.testing.assert_close(dequantize_mxfp4(p2, e2), wq)


@settings(max_examples=10, deadline=None)
@given(lens=st.lists(st.integers(3, 160), min_size=1, max_size=4),
       chunk=st.sampled_from([None, 16, 64]),
       prefix=st.booleans(), new=st.integers(1, 16),
       mix=st.booleans())
def test_batcher_fuzz_speculative_ragged(lens, chunk, prefix, new, mix):
    """The ragged speculative decode batch (greedy slots contribute 1+k
    verify rows, optionally mixed with concurrent SAMPLED requests in
    the same batch) always reproduces serial greedy for the greedy
    requests, for any prompt mix / chunking / prefix caching."""
    from tensorlink_amd.engine.batcher import ContinuousBatcher
    from tensorlink_amd.parallel.pipeline import (PipelineRunner,
                                                  SamplingParams)
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                       device=torch.device("cpu"), seed=10)
    b = ContinuousBatcher(r, max_slots=4, max_ctx=512,
                          prefill_chunk=chunk, prefix_caching=prefix,
                          speculative=True).start()
    try:
        g = torch.Generator().manual_seed(sum(lens) * 31 + new)
        prompts = [torch.randint(0, 1024, (n,), generator=g)
                   for n in lens]
        reqs = [b.submit(p.clone(), max_new_tokens=new) for p in prompts]
        noise = []
        if mix:
            # concurrent sampled + penalized requests share the ragged
            # batch but must not perturb the greedy rows
            noise = [b.submit(torch.randint(0, 1024, (24,), generator=g),
                              max_new_tokens=new, temperature=0.9,
                              seed=7, presence_penalty=0.3)
                     for _ in range(2)]
        outs = [rq.result(timeout=120) for rq in reqs]
        for rq in noise:
            assert len(rq.result(timeout=120)) == new
        for p, o in zip(prompts, outs):
            ref = r.generate(p.unsqueeze(0),
                             SamplingParams(max_new_tokens=new))
            assert o == ref[0].tolist()
    finally:
        b.stop()
