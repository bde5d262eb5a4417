"""Reference-op sanity tests (CPU): these define the numerics contract the
HIP kernels are validated against in tests/test_ops_gpu.py."""

import math

import pytest
import torch

from tensorlink_amd.ops import reference as ref


def test_rmsnorm_matches_manual():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = ref.rmsnorm(x, w, 1e-6)
    expected = x / (x.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w
    assert torch.allclose(y, expected, atol=1e-5)


def test_rmsnorm_residual_consistency():
    x, r = torch.randn(4, 64), torch.randn(4, 64)
    w = torch.randn(64)
    y, rsum = ref.rmsnorm_residual(x, r, w)
    assert torch.allclose(rsum, x + r, atol=1e-6)
    assert torch.allclose(y, ref.rmsnorm(x + r, w), atol=1e-6)


def test_rope_preserves_norm_and_inverts():
    B, S, H, D = 2, 8, 4, 64
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, 2, D)
    pos = torch.arange(S)
    cos, sin = ref.rope_cos_sin(D, pos)
    q2, k2 = ref.apply_rope(q, k, cos, sin)
    # rotation preserves pairwise norms
    n1 = q[..., :32].pow(2) + q[..., 32:].pow(2)
    n2 = q2[..., :32].pow(2) + q2[..., 32:].pow(2)
    assert torch.allclose(n1, n2, atol=1e-4)
    # position 0 is identity
    assert torch.allclose(q2[:, 0], q[:, 0], atol=1e-6)
    # inverse rotation restores input
    q3, k3 = ref.apply_rope(q2, k2, cos, -sin)
    assert torch.allclose(q3, q, atol=1e-5)


def test_rope_matches_hf_convention():
    # independent re-derivation of HF rotate_half
    D, S = 8, 4
    q = torch.randn(1, S, 1, D)
    pos = torch.arange(S)
    cos, sin = ref.rope_cos_sin(D, pos, theta=10000.0)
    q2, _ = ref.apply_rope(q, q.clone(), cos, sin)
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
    for s in range(S):
        ang = s * inv
        c, sn = ang.cos(), ang.sin()
        x1, x2 = q[0, s, 0, :4], q[0, s, 0, 4:]
        torch.testing.assert_close(q2[0, s, 0, :4], x1 * c - x2 * sn,
                                   atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(q2[0, s, 0, 4:], x2 * c + x1 * sn,
                                   atol=1e-5, rtol=1e-4)


def test_attention_prefill_vs_manual_softmax():
    B, S, Hq, Hkv, D = 1, 16, 4, 2, 32
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    out = ref.attention_prefill(q, k, v, causal=True)
    # manual per-position check for head 3 (kv head 1), position 5
    h, p = 3, 5
    kk = k[0, :p + 1, h // 2]
    vv = v[0, :p + 1, h // 2]
    sc = (q[0, p, h] @ kk.t()) / math.sqrt(D)
    expected = sc.softmax(-1) @ vv
    torch.testing.assert_close(out[0, p, h], expected, atol=1e-5, rtol=1e-4)


def test_decode_matches_prefill_last_position():
    B, S, Hq, Hkv, D = 2, 9, 4, 2, 32
    q_all = torch.randn(B, S, Hq, D)
    k_all = torch.randn(B, S, Hkv, D)
    v_all = torch.randn(B, S, Hkv, D)
    full = ref.attention_prefill(q_all, k_all, v_all, causal=True)
    # decode: query = last position, cache = all positions
    # (ref.attention_decode takes the cache as [B,Smax,Hkv,D])
    out2 = ref.attention_decode(q_all[:, -1:], k_all, v_all,
                                torch.full((B,), S))
    torch.testing.assert_close(out2[:, 0], full[:, -1], atol=1e-5, rtol=1e-4)


def test_sampling_greedy_and_topp():
    logits = torch.tensor([[1.0, 5.0, 2.0, 0.1]])
    assert ref.sample_token(logits, temperature=0.0).item() == 1
    g = torch.Generator().manual_seed(0)
    # top_p tiny -> only the argmax survives
    for _ in range(10):
        t = ref.sample_token(logits, temperature=1.0, top_p=0.01, generator=g)
        assert t.item() == 1
    # top_k = 2 -> only tokens 1 and 2
    for _ in range(20):
        t = ref.sample_token(logits, temperature=2.0, top_k=2, generator=g)
        assert t.item() in (1, 2)


def test_adamw_matches_torch():
    torch.manual_seed(0)
    p_ref = torch.randn(100)
    g = torch.randn(100)
    p_mine = p_ref.clone()
    m = torch.zeros(100)
    v = torch.zeros(100)
    opt = torch.optim.AdamW([p_ref.requires_grad_()], lr=1e-2, betas=(0.9, 0.999),
                            eps=1e-8, weight_decay=0.01)
    for step in range(1, 4):
        p_ref.grad = g.clone()
        opt.step()
        ref.adamw_step(p_mine, g, m, v, lr=1e-2, beta1=0.9, beta2=0.999,
                       eps=1e-8, weight_decay=0.01, step=step)
    torch.testing.assert_close(p_mine, p_ref.detach(), atol=1e-6, rtol=1e-5)


def test_moe_router():
    logits = torch.randn(10, 8)
    w, idx = ref.moe_topk_router(logits, 2)
    assert w.shape == (10, 2) and idx.shape == (10, 2)
    torch.testing.assert_close(w.sum(-1), torch.ones(10))
    # indices are the top-2 of softmax == top-2 of logits
    expected_idx = logits.topk(2, dim=-1).indices
    assert torch.equal(idx, expected_idx)


def test_causal_lm_loss_shift():
    V = 11
    logits = torch.zeros(1, 3, V)
    logits[0, 0, 5] = 100.0  # predicts token at position 1
    logits[0, 1, 7] = 100.0  # predicts token at position 2
    labels = torch.tensor([[9, 5, 7]])
    loss = ref.causal_lm_loss(logits, labels)
    assert loss.item() < 1e-4


def test_chunked_ce_matches_full():
    """Chunked CE (loss + d_hidden + d_head) == full-logits CE exactly,
    incl. ignore_index padding and chunk boundaries not dividing N."""
    import torch

    from tensorlink_amd.ops import reference as ref
    torch.manual_seed(4)
    B, S, H, V = 2, 13, 32, 97
    hidden = torch.randn(B, S, H, requires_grad=True)
    w = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (B, S))
    labels[0, 5:8] = -100

    loss_c = ref.chunked_causal_lm_loss(hidden, w, labels, chunk=7)
    loss_c.backward()
    gh_c, gw_c = hidden.grad.clone(), w.grad.clone()

    hidden2 = hidden.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    logits = hidden2 @ w2.t()
    loss_f = ref.causal_lm_loss(logits, labels)
    loss_f.backward()

    assert torch.allclose(loss_c, loss_f, atol=1e-6)
    assert torch.allclose(gh_c, hidden2.grad, atol=1e-6)
    assert torch.allclose(gw_c, w2.grad, atol=1e-5)
