"""HIP kernel numerics tests (MI355X): each kernel vs a plain PyTorch fp32
reference of the same op (the kernel-unit tier the reference
lacks; cf. its gradient-hash idea ml/proofs.py:6)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _ext():
    from tensorlink_amd import ops
    assert ops.extension_loaded(), "HIP extension must be built on GPU boxes"
    return ops


def test_rmsnorm_fwd():
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(0)
    for N, H in [(64, 256), (33, 3584), (1024, 4096), (2, 8192)]:
        x = torch.randn(N, H, device=DEV, dtype=torch.bfloat16)
        w = torch.randn(H, device=DEV, dtype=torch.bfloat16)
        y = ops.rmsnorm(x, w, 1e-6)
        y_ref = ref.rmsnorm(x.float(), w.float(), 1e-6)
        torch.testing.assert_close(y.float(), y_ref, atol=2e-2, rtol=2e-2)


def test_rmsnorm_residual_fwd():
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    x = torch.randn(128, 3584, device=DEV, dtype=torch.bfloat16)
    r = torch.randn(128, 3584, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(3584, device=DEV, dtype=torch.bfloat16)
    y, rs = ops.rmsnorm_residual(x, r, w, 1e-6)
    y_ref, rs_ref = ref.rmsnorm_residual(x.float(), r.float(), w.float(), 1e-6)
    torch.testing.assert_close(rs.float(), rs_ref, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(y.float(), y_ref, atol=2e-2, rtol=2e-2)


def test_rmsnorm_backward():
    ops = _ext()
    torch.manual_seed(1)
    N, H = 64, 1024
    x = torch.randn(N, H, device=DEV, dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(H, device=DEV, dtype=torch.bfloat16, requires_grad=True)
    y = ops.rmsnorm(x, w, 1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)

    x32 = x.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    var = x32.pow(2).mean(-1, keepdim=True)
    y32 = x32 * torch.rsqrt(var + 1e-6) * w32
    y32.backward(dy.float())
    torch.testing.assert_close(x.grad.float(), x32.grad, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(w.grad.float(), w32.grad, atol=5e-2, rtol=5e-2)


def test_rope():
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(2)
    T, Hq, Hkv, D = 33, 8, 2, 128
    q = torch.randn(T, Hq, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(T, Hkv, D, device=DEV, dtype=torch.bfloat16)
    pos = torch.randint(0, 1000, (T,), device=DEV, dtype=torch.int32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2, device=DEV).float() / D))
    q2, k2 = q.clone(), k.clone()
    ops.apply_rope_(q2, k2, pos, inv)
    cos, sin = ref.rope_cos_sin(D, pos, device=DEV)
    qr, kr = ref.apply_rope(q.float().unsqueeze(0), k.float().unsqueeze(0),
                            cos, sin)
    torch.testing.assert_close(q2.float(), qr[0], atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(k2.float(), kr[0], atol=2e-2, rtol=2e-2)
    # inverse rotation restores
    ops.apply_rope_(q2, k2, pos, inv, sign=-1.0)
    torch.testing.assert_close(q2.float(), q.float(), atol=3e-2, rtol=3e-2)


def test_swiglu_fwd_bwd():
    ops = _ext()
    torch.manual_seed(3)
    g = torch.randn(64, 18944, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    u = torch.randn(64, 18944, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    out = ops.swiglu(g, u)
    g32 = g.detach().float().requires_grad_(True)
    u32 = u.detach().float().requires_grad_(True)
    ref32 = torch.nn.functional.silu(g32) * u32
    torch.testing.assert_close(out.float(), ref32.detach(), atol=2e-2,
                               rtol=2e-2)
    dout = torch.randn_like(out)
    out.backward(dout)
    ref32.backward(dout.float())
    torch.testing.assert_close(g.grad.float(), g32.grad, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(u.grad.float(), u32.grad, atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("B,Hq,Hkv,D,L", [
    (2, 28, 4, 128, 300),   # Qwen2.5-7B shape (G=7)
    (4, 32, 8, 128, 77),    # Llama/Qwen3 shape (G=4)
    (1, 4, 2, 64, 1500),    # small-D long-L
    (3, 8, 8, 128, 64),     # MHA (G=1)
])
def test_decode_attn(B, Hq, Hkv, D, L):
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(4)
    Smax = L + 37
    q = torch.randn(B, Hq, D, device=DEV, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    vc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    lens = torch.randint(1, L + 1, (B,), device=DEV, dtype=torch.int32)
    lens[0] = L
    out = ops.attention_decode(q, kc, vc, lens)
    out_ref = ref.attention_decode(
        q.unsqueeze(1).float(), kc.permute(0, 2, 1, 3).float(),
        vc.permute(0, 2, 1, 3).float(), lens)
    torch.testing.assert_close(out.float(), out_ref[:, 0], atol=2e-2,
                               rtol=2e-2)


@pytest.mark.parametrize("B,S,Hq,Hkv,D", [
    (2, 128, 28, 4, 128),
    (1, 500, 8, 8, 128),    # non-multiple of 64
    (2, 64, 4, 2, 64),
    (1, 2048, 8, 2, 128),
])
def test_prefill_attn(B, S, Hq, Hkv, D):
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(5)
    q = torch.randn(B, S, Hq, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=DEV, dtype=torch.bfloat16)
    out = ops.attention_prefill(q, k, v, causal=True)
    out_ref = ref.attention_prefill(q.float(), k.float(), v.float(),
                                    causal=True)
    torch.testing.assert_close(out.float(), out_ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("B,Sq,Skv,q_off,Hq,Hkv,D", [
    (1, 64, 192, 128, 8, 2, 128),   # aligned chunk
    (2, 50, 127, 77, 4, 2, 128),    # ragged chunk + ragged history
    (1, 33, 99, 66, 4, 4, 64),
])
def test_prefill_attn_q_off(B, Sq, Skv, q_off, Hq, Hkv, D):
    """Chunked prefill: queries at global rows q_off..q_off+Sq-1 attend to
    all Skv cached keys with the causal mask applied at global positions."""
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(15)
    q = torch.randn(B, Sq, Hq, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(B, Skv, Hkv, D, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(B, Skv, Hkv, D, device=DEV, dtype=torch.bfloat16)
    out = ops.attention_prefill(q, k, v, causal=True, q_off=q_off)
    out_ref = ref.attention_prefill(q.float(), k.float(), v.float(),
                                    causal=True, q_off=q_off)
    torch.testing.assert_close(out.float(), out_ref, atol=2e-2, rtol=2e-2)


def test_adamw_matches_torch():
    ops = _ext()
    torch.manual_seed(6)
    n = 100003
    p_ref = torch.randn(n, device=DEV, dtype=torch.float32)
    p_mine = p_ref.clone().to(torch.bfloat16)
    p_ref = p_ref.to(torch.bfloat16).float()   # match bf16 starting point
    g = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    pr = p_ref.clone().requires_grad_(True)
    opt = torch.optim.AdamW([pr], lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=0.01)
    for step in range(1, 5):
        pr.grad = g.clone()
        opt.step()
        ops.adamw_(p_mine, g.to(torch.bfloat16), m, v, lr=1e-2, beta1=0.9,
                   beta2=0.999, eps=1e-8, weight_decay=0.01, step=step)
    # bf16 params accumulate rounding each step; tolerance reflects that
    torch.testing.assert_close(p_mine.float(), pr.detach(), atol=2e-2,
                               rtol=2e-2)


def test_decode_attn_softmax_stability():
    """Large score magnitudes must not overflow (online softmax)."""
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    B, Hq, Hkv, D, L = 1, 4, 2, 128, 200
    q = (torch.randn(B, Hq, D, device=DEV) * 10).to(torch.bfloat16)
    kc = (torch.randn(B, Hkv, L, D, device=DEV) * 10).to(torch.bfloat16)
    vc = torch.randn(B, Hkv, L, D, device=DEV).to(torch.bfloat16)
    lens = torch.full((B,), L, device=DEV, dtype=torch.int32)
    out = ops.attention_decode(q, kc, vc, lens)
    assert torch.isfinite(out.float()).all()
    out_ref = ref.attention_decode(q.unsqueeze(1).float(),
                                   kc.permute(0, 2, 1, 3).float(),
                                   vc.permute(0, 2, 1, 3).float(), lens)
    torch.testing.assert_close(out.float(), out_ref[:, 0], atol=3e-2,
                               rtol=3e-2)


def test_rope_append_fused():
    """Fused-QKV rope+append vs reference rotate-then-place."""
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(7)
    B, S, Hq, Hkv, D, Smax = 3, 5, 8, 2, 128, 64
    T = B * S
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, device=DEV,
                      dtype=torch.bfloat16)
    q = qkv[:, :Hq * D].reshape(T, Hq, D).contiguous()
    k = qkv[:, Hq * D:(Hq + Hkv) * D].reshape(T, Hkv, D).contiguous()
    v = qkv[:, (Hq + Hkv) * D:].reshape(T, Hkv, D).contiguous()
    kc = torch.zeros(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    pos = torch.arange(S, device=DEV, dtype=torch.int32).repeat(B) + 3
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2, device=DEV).float() / D))
    q_out = ops.rope_append_(qkv, kc, vc, pos, inv, S, Hq, Hkv)
    cos, sin = ref.rope_cos_sin(D, pos, device=DEV)
    qr, kr = ref.apply_rope(q.float().unsqueeze(0), k.float().unsqueeze(0),
                            cos, sin)
    torch.testing.assert_close(q_out.float(), qr[0], atol=2e-2, rtol=2e-2)
    krB = kr[0].view(B, S, Hkv, D)
    vB = v.view(B, S, Hkv, D).float()
    for b in range(B):
        for s in range(S):
            p = int(pos[b * S + s])
            torch.testing.assert_close(kc[b, :, p].float(), krB[b, s],
                                       atol=2e-2, rtol=2e-2)
            torch.testing.assert_close(vc[b, :, p].float(), vB[b, s],
                                       atol=1e-3, rtol=1e-3)


def test_swiglu_fused_gpu():
    ops = _ext()
    torch.manual_seed(8)
    N, I = 33, 512
    gu = torch.randn(N, 2 * I, device=DEV, dtype=torch.bfloat16)
    out = ops.swiglu_fused(gu)
    ref32 = torch.nn.functional.silu(gu[:, :I].float()) * gu[:, I:].float()
    torch.testing.assert_close(out.float(), ref32, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("B,Hq,Hkv,D,L,ns", [
    (2, 28, 4, 128, 300, 1),    # Qwen2.5-7B, no split
    (2, 28, 4, 128, 300, 4),    # with flash-decode split
    (1, 32, 8, 128, 1000, 8),   # Llama/Qwen3 G=4, long L, split
    (4, 32, 8, 128, 77, 1),
    (1, 4, 2, 64, 333, 2),      # D=64
    (3, 8, 8, 128, 64, 1),      # MHA G=1
    (2, 64, 8, 128, 129, 2),    # G=8 (Llama-3-70B shape)
])
def test_decode_attn_mfma(B, Hq, Hkv, D, L, ns):
    ops = _ext()
    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(11)
    Smax = L + 19
    q = torch.randn(B, Hq, D, device=DEV, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    vc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    lens = torch.randint(max(1, L - 40), L + 1, (B,), device=DEV,
                         dtype=torch.int32)
    lens[0] = L
    out = ops.attention_decode(q, kc, vc, lens, n_split=ns)
    out_ref = ref.attention_decode(
        q.unsqueeze(1).float(), kc.permute(0, 2, 1, 3).float(),
        vc.permute(0, 2, 1, 3).float(), lens)
    torch.testing.assert_close(out.float(), out_ref[:, 0], atol=2e-2,
                               rtol=2e-2)


@pytest.mark.parametrize("M,N,K,bias", [
    (256, 3584, 3584, False),     # o_proj (tiled kernel)
    (256, 4608, 3584, True),      # fused-qkv shape w/ bias (tiled)
    (256, 18944, 3584, False),    # gate/up (tiled)
    (256, 3584, 18944, False),    # down (tiled)
    (150, 1024, 256, False),      # mid-M, MT=16 partial tail
    (96, 1024, 256, True),        # MT=8 template
    (300, 1024, 256, False),      # blockIdx.z m-tiling (2 blocks)
    (512, 4608, 3584, True),      # dispatch ceiling, real shape
    (64, 1024, 256, True),        # streaming-kernel boundary
    (16, 3584, 3584, False),      # small-batch decode (streaming)
    (7, 1024, 256, True),         # tiny M, M tail
    (33, 512, 96, False),         # K%64==32 tail path
])
@torch.no_grad()
def test_tl_gemm(M, N, K, bias):
    ops = _ext()
    torch.manual_seed(12)
    x = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=DEV, dtype=torch.bfloat16) / (K ** 0.5)
    b = torch.randn(N, device=DEV, dtype=torch.bfloat16) if bias else None
    before = ops.gemm_dispatch_count
    out = ops.linear(x, w, b)
    # every family-routed parametrization must hit the hand-written
    # kernels (classes defaulted to the library are numerics-only here)
    if ops._gemm_class(N, K) not in ops._LIB_CLASSES:
        assert ops.gemm_dispatch_count == before + 1
    ref = torch.nn.functional.linear(x.float(), w.float(),
                                     b.float() if bias else None)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_decode_attn_batch_independence():
    """A row's decode attention is bitwise identical whatever batch it is
    in (per-row flash-decode split from its own length — common.hpp
    tl_split_for_len). Mixed short/long rows vs each row alone."""
    ops = _ext()
    torch.manual_seed(14)
    B, Hq, Hkv, D, Smax = 5, 8, 4, 128, 1536
    q = torch.randn(B, Hq, D, device=DEV, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    vc = torch.randn(B, Hkv, Smax, D, device=DEV, dtype=torch.bfloat16)
    # lengths straddling the 512/1024 split boundaries
    lens = torch.tensor([100, 511, 513, 1024, 1500], device=DEV,
                        dtype=torch.int32)
    batched = ops.attention_decode(q, kc, vc, lens)
    for b in range(B):
        solo = ops.attention_decode(q[b:b + 1], kc[b:b + 1], vc[b:b + 1],
                                    lens[b:b + 1])
        assert torch.equal(batched[b], solo[0]), f"row {b} diverged"


@torch.no_grad()
def test_tl_gemm_row_m_independence():
    """The determinism contract: a row's GEMM result is bitwise identical
    whatever batch it is computed in (streaming M<=64 kernel, tiled
    kernel, any M, with or without the rows around it). Chunked prefill /
    speculative verify / ragged decode equality all rest on this."""
    ops = _ext()
    torch.manual_seed(13)
    K, N = 3584, 4608
    x = torch.randn(300, K, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=DEV, dtype=torch.bfloat16) / (K ** 0.5)
    full = ops.linear(x, w)                        # tiled, 2 m-blocks
    one = ops.linear(x[7:8], w)                    # streaming M=1
    chunk = ops.linear(x[:64], w)                  # streaming M=64
    mid = ops.linear(x[:150], w)                   # tiled MT=16
    assert torch.equal(full[7], one[0])
    assert torch.equal(full[:64], chunk)
    assert torch.equal(full[:150], mid)


# ---------------------------------------------------------------------------
# Fused sampling kernel (ops/csrc/sampling.hip)
# ---------------------------------------------------------------------------
def _sample_params(B, dev, temp=1.0, top_p=1.0, top_k=0):
    return dict(
        temps=torch.full((B,), float(temp), device=dev),
        top_ps=torch.full((B,), float(top_p), device=dev),
        top_ks=torch.full((B,), int(top_k), device=dev, dtype=torch.int32),
        pres=torch.zeros(B, device=dev),
        freqs=torch.zeros(B, device=dev))


def test_sample_greedy_matches_argmax():
    ops = _ext()
    torch.manual_seed(40)
    logits = torch.randn(5, 4096, device=DEV, dtype=torch.bfloat16)
    out = ops.sample_tokens(logits, **_sample_params(5, DEV, temp=0.0))
    assert torch.equal(out, logits.argmax(-1))


def test_sample_penalties_match_reference():
    ops = _ext()
    torch.manual_seed(41)
    B, V = 3, 2048
    logits = torch.randn(B, V, device=DEV, dtype=torch.bfloat16)
    counts = torch.zeros(B, V, device=DEV, dtype=torch.int32)
    counts[0, 7] = 3
    counts[1, :50] = 1
    counts[2, 100] = 10
    p = _sample_params(B, DEV, temp=0.0)
    p["pres"] = torch.full((B,), 0.8, device=DEV)
    p["freqs"] = torch.full((B,), 0.5, device=DEV)
    before = counts.clone()
    out = ops.sample_tokens(logits, counts=counts, **p)
    for b in range(B):
        nz = torch.nonzero(before[b]).flatten()
        d = {int(t): int(before[b, t]) for t in nz}
        ref = ops.sample_token(logits[b:b + 1], temperature=0.0,
                               token_counts=d, presence_penalty=0.8,
                               frequency_penalty=0.5)
        assert int(out[b]) == int(ref[0])
        # chosen token's count was incremented on device
        assert int(counts[b, out[b]]) == int(before[b, out[b]]) + 1


def test_sample_topk_topp_membership():
    ops = _ext()
    torch.manual_seed(42)
    V = 8192
    logits = torch.randn(1, V, device=DEV, dtype=torch.bfloat16) * 3
    lf = logits[0].float()
    # top-k: every draw's logit must be >= the k-th largest value
    kth = lf.topk(16).values[-1]
    seeds = torch.arange(100, device=DEV, dtype=torch.int64)
    rows = logits.expand(100, V).contiguous()
    out = ops.sample_tokens(rows, seeds=seeds,
                            **_sample_params(100, DEV, temp=1.0, top_k=16))
    assert (lf[out] >= kth).all(), lf[out].min()
    # top-p: every draw's logit must be >= the crossing token's value
    p = 0.6
    sl, _ = lf.sort(descending=True)
    probs = sl.softmax(-1)
    cross = int((probs.cumsum(-1) > p).nonzero()[0])
    out = ops.sample_tokens(rows, seeds=seeds,
                            **_sample_params(100, DEV, temp=1.0, top_p=p))
    assert (lf[out] >= sl[cross]).all()


def test_sample_seeded_reproducible():
    ops = _ext()
    torch.manual_seed(43)
    logits = torch.randn(4, 4096, device=DEV, dtype=torch.bfloat16)
    seeds = torch.tensor([ops.request_seed(9, s) for s in range(4)],
                         device=DEV, dtype=torch.int64)
    a = ops.sample_tokens(logits, seeds=seeds,
                          **_sample_params(4, DEV, temp=0.9, top_p=0.95))
    b = ops.sample_tokens(logits, seeds=seeds,
                          **_sample_params(4, DEV, temp=0.9, top_p=0.95))
    assert torch.equal(a, b)
    seeds2 = seeds + 1
    c = ops.sample_tokens(logits, seeds=seeds2,
                          **_sample_params(4, DEV, temp=0.9, top_p=0.95))
    assert not torch.equal(a, c)


def test_sample_distribution():
    """Draw frequencies track softmax probabilities on a small vocab."""
    ops = _ext()
    torch.manual_seed(44)
    V, N = 32, 8192
    logits = (torch.randn(V) * 2).to(DEV, torch.bfloat16)
    probs = logits.float().softmax(-1).cpu()
    rows = logits.unsqueeze(0).expand(N, V).contiguous()
    seeds = torch.arange(N, device=DEV, dtype=torch.int64) * 7919
    out = ops.sample_tokens(rows, seeds=seeds,
                            **_sample_params(N, DEV, temp=1.0)).cpu()
    freq = torch.bincount(out, minlength=V).float() / N
    assert (freq - probs).abs().max() < 0.03, (freq - probs).abs().max()


def test_graph_sampled_decode():
    """Sampled decode under hipGraph capture: seeded reproducibility and
    divergence from greedy (the fused kernel's counter-based RNG)."""
    from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
    from tensorlink_amd.parallel.planner import plan_for_world
    r = PipelineRunner(plan_for_world("tiny", 1), 0, 1, device=DEV,
                       dtype=torch.bfloat16, seed=11)
    torch.manual_seed(45)
    ids = torch.randint(0, 1024, (2, 16))
    sp = SamplingParams(temperature=0.8, top_p=0.9, max_new_tokens=12,
                        seed=5)
    o1 = r.generate(ids, sp)
    o2 = r.generate(ids, sp)
    assert torch.equal(o1, o2)
    assert r._decode_graph is not None      # captured, not eager fallback
    greedy = r.generate(ids, SamplingParams(max_new_tokens=12))
    assert not torch.equal(o1, greedy)


# ---------------------------------------------------------------------------
# Grouped MoE expert GEMM (ops/csrc/moe_gemm.hip)
# ---------------------------------------------------------------------------
@torch.no_grad()
@pytest.mark.parametrize("fp8", [False, True])
def test_moe_gemm_grouped_vs_loop(fp8):
    """One grouped launch over expert-sorted pairs equals the per-expert
    GEMM loop (bf16 exact vs the family kernel; fp8 vs dequant mm)."""
    ops = _ext()
    C = ops._require_ext()
    torch.manual_seed(50)
    E, K, N, T, k = 4, 256, 512, 33, 2
    x = torch.randn(T, K, device=DEV, dtype=torch.bfloat16)
    ws = [torch.randn(N, K, device=DEV, dtype=torch.bfloat16) / K ** 0.5
          for _ in range(E)]
    idx = torch.randint(0, E, (T, k), device=DEV)
    fi = idx.reshape(-1)
    order = fi.argsort(stable=True)
    seg = torch.zeros(E + 1, device=DEV, dtype=torch.int32)
    seg[1:] = torch.bincount(fi, minlength=E).cumsum(0)
    pair_tok = (order // k).to(torch.int32)
    if fp8:
        from tensorlink_amd.models.quant import quantize_fp8_per_channel
        q = [quantize_fp8_per_channel(w) for w in ws]
        wp = torch.tensor([a.data_ptr() for a, _ in q], dtype=torch.int64,
                          device=DEV)
        sp = torch.tensor([s.data_ptr() for _, s in q], dtype=torch.int64,
                          device=DEV)
        out = C.moe_gemm(x, pair_tok, seg, wp, sp, N, True)
        for p in range(T * k):
            e = int(fi[order[p]])
            wq, sc = q[e]
            ref = (x[pair_tok[p].long()].float()
                   @ (wq.float() * sc[:, None]).t())
            torch.testing.assert_close(out[p].float(), ref, atol=5e-2,
                                       rtol=5e-2)
    else:
        wp = torch.tensor([w.data_ptr() for w in ws], dtype=torch.int64,
                          device=DEV)
        out = C.moe_gemm(x, pair_tok, seg, wp, None, N, False)
        for p in range(T * k):
            e = int(fi[order[p]])
            ref = x[pair_tok[p].long()].float() @ ws[e].float().t()
            torch.testing.assert_close(out[p].float(), ref, atol=3e-2,
                                       rtol=3e-2)
        # determinism: a pair's row is identical whatever else is in
        # the batch (drop half the pairs, recompute, compare shared)
        half = (T * k) // 2
        seg2 = torch.clamp(seg, max=half)
        out2 = C.moe_gemm(x, pair_tok[:half], seg2, wp, None, N, False)
        n_cmp = int(seg2[-1])
        assert torch.equal(out2[:n_cmp], out[:n_cmp])


@torch.no_grad()
def test_moe_fused_block_matches_loop():
    """MoEMLP's fused grouped path equals the per-expert loop."""
    from tensorlink_amd.models.configs import get_config
    from tensorlink_amd.models.dense import MoEMLP
    _ext()
    torch.manual_seed(51)
    cfg = get_config("tiny-moe")
    mlp = MoEMLP(cfg).to(DEV, torch.bfloat16).eval()
    x = torch.randn(2, 9, cfg.hidden_size, device=DEV,
                    dtype=torch.bfloat16)
    fused = mlp(x)
    assert mlp._fused_kind() == "bf16"
    # reference: the python loop (force by pretending no uniform experts)
    import tensorlink_amd.models.dense as dense_mod
    orig = MoEMLP._fused_kind
    MoEMLP._fused_kind = lambda self: None
    try:
        loop = mlp(x)
    finally:
        MoEMLP._fused_kind = orig
    torch.testing.assert_close(fused.float(), loop.float(), atol=3e-2,
                               rtol=3e-2)


def test_moe_fp8_graph_decode():
    """Quantized-MoE decode is hipGraph-captured with the grouped kernel
    (round 1 disabled graphs for fp8 because of _scaled_mm)."""
    from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
    from tensorlink_amd.parallel.planner import plan_for_world
    r = PipelineRunner(plan_for_world("tiny-moe", 1), 0, 1, device=DEV,
                       dtype=torch.bfloat16, seed=6, quantize="fp8")
    torch.manual_seed(52)
    ids = torch.randint(0, 1024, (2, 12))
    out = r.generate(ids, SamplingParams(max_new_tokens=8))
    assert out.shape == (2, 8)
    assert r._decode_graph is not None, "fp8 MoE decode fell back to eager"


# ---------------------------------------------------------------------------
# Flash-attention backward (ops/csrc/flash_bwd.hip)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("B,S,Hq,Hkv,D,causal", [
    (2, 128, 8, 2, 128, True),    # GQA G=4
    (1, 200, 4, 4, 64, True),     # MHA, ragged S, D=64
    (2, 64, 28, 4, 128, True),    # Qwen2.5-7B head shape
    (1, 96, 8, 8, 128, False),    # non-causal
])
def test_flash_attn_backward(B, S, Hq, Hkv, D, causal):
    """Hand-written flash fwd+bwd vs fp32 autograd reference."""
    ops = _ext()
    torch.manual_seed(60)
    mk = lambda *s: (torch.randn(*s, device=DEV, dtype=torch.bfloat16)
                     .requires_grad_(True))
    q, k, v = mk(B, S, Hq, D), mk(B, S, Hkv, D), mk(B, S, Hkv, D)
    out = ops.attention_train(q, k, v, causal=causal)
    # the custom path must actually be active
    assert out.grad_fn is not None \
        and "FlashAttn" in type(out.grad_fn).__name__
    dout = torch.randn_like(out)
    out.backward(dout)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    rep = Hq // Hkv
    ref = torch.nn.functional.scaled_dot_product_attention(
        q32.transpose(1, 2),
        k32.transpose(1, 2).repeat_interleave(rep, dim=1),
        v32.transpose(1, 2).repeat_interleave(rep, dim=1),
        is_causal=causal, scale=1.0 / math.sqrt(D)).transpose(1, 2)
    torch.testing.assert_close(out.float(), ref.detach(), atol=2e-2,
                               rtol=2e-2)
    ref.backward(dout.float())
    torch.testing.assert_close(q.grad.float(), q32.grad, atol=7e-2,
                               rtol=7e-2)
    torch.testing.assert_close(k.grad.float(), k32.grad, atol=7e-2,
                               rtol=7e-2)
    torch.testing.assert_close(v.grad.float(), v32.grad, atol=7e-2,
                               rtol=7e-2)


def test_training_step_uses_flash_bwd():
    """A PipelineTrainer step runs with the flash kernels on the path
    and the loss decreases."""
    from tensorlink_amd.parallel.pipeline import PipelineTrainer
    from tensorlink_amd.parallel.planner import plan_for_world
    t = PipelineTrainer(plan_for_world("tiny", 1, training=True), 0, 1,
                        device=DEV, seed=1, lr=5e-3)
    torch.manual_seed(61)
    ids = torch.randint(0, 1024, (2, 64))
    losses = [t.train_step(ids, labels=ids) for _ in range(4)]
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses
