"""CPU tests of the reference op implementations and their gradients."""

import math

import pytest
import torch
import torch.nn.functional as F

from hypha_amd.ops import reference as R


def test_rmsnorm_matches_torch():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    got = R.rmsnorm(x, w, eps=1e-5)
    want = F.rms_norm(x, (64,), weight=w, eps=1e-5)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-5)


def test_rope_rotation_preserves_norm():
    cos, sin = R.rope_cos_sin(32, 16, base=10000.0)
    x = torch.randn(2, 4, 32, 16)
    y = R.apply_rope(x, cos, sin)
    torch.testing.assert_close(
        x.norm(dim=-1), y.norm(dim=-1), rtol=1e-5, atol=1e-5
    )
    # position 0 is identity
    torch.testing.assert_close(y[:, :, 0], x[:, :, 0], rtol=1e-6, atol=1e-6)


def test_rope_inverse():
    cos, sin = R.rope_cos_sin(32, 16, base=10000.0)
    x = torch.randn(1, 2, 32, 16)
    y = R.apply_rope(x, cos, sin)
    back = R.apply_rope(y, cos, -sin)
    torch.testing.assert_close(back, x, rtol=1e-5, atol=1e-5)


def test_attention_matches_sdpa():
    torch.manual_seed(0)
    q = torch.randn(2, 4, 16, 8)
    k = torch.randn(2, 2, 16, 8)
    v = torch.randn(2, 2, 16, 8)
    got = R.attention(q, k, v, causal=True)
    want = F.scaled_dot_product_attention(
        q, k.repeat_interleave(2, 1), v.repeat_interleave(2, 1), is_causal=True
    )
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


def test_swiglu():
    g = torch.randn(8, 16)
    u = torch.randn(8, 16)
    torch.testing.assert_close(R.swiglu(g, u), F.silu(g) * u, rtol=1e-5, atol=1e-6)


def test_cross_entropy_ignore_index():
    logits = torch.randn(6, 11)
    t = torch.tensor([1, 2, -100, 4, 5, -100])
    torch.testing.assert_close(
        R.cross_entropy(logits, t), F.cross_entropy(logits, t, ignore_index=-100)
    )


def test_adamw_matches_torch_optim():
    torch.manual_seed(1)
    n = 257
    master = torch.randn(n)
    p_ref = master.clone().requires_grad_(True)
    opt = torch.optim.AdamW(
        [p_ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.1
    )
    param = master.clone()
    m = torch.zeros(n)
    v = torch.zeros(n)
    for step in range(1, 6):
        g = torch.randn(n)
        p_ref.grad = g.clone()
        opt.step()
        R.adamw_step(
            master, param, g, m, v,
            lr=1e-2, beta1=0.9, beta2=0.95, eps=1e-8, weight_decay=0.1, step=step,
        )
    torch.testing.assert_close(master, p_ref.detach(), rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(param, master)


def test_nesterov_matches_torch_sgd():
    """The golden-value pattern from the reference parameter server
    (parameter_server.rs:448-525): our outer step with pseudo-gradient
    delta must equal torch SGD(nesterov=True) fed gradient -delta."""
    torch.manual_seed(2)
    n = 97
    theta = torch.randn(n)
    p_ref = theta.clone().requires_grad_(True)
    opt = torch.optim.SGD([p_ref], lr=0.7, momentum=0.9, nesterov=True)
    momentum = torch.zeros(n)
    for _ in range(4):
        delta = torch.randn(n)
        p_ref.grad = -delta
        opt.step()
        R.nesterov_outer_step(theta, delta, momentum, lr=0.7, mu=0.9)
    torch.testing.assert_close(theta, p_ref.detach(), rtol=1e-5, atol=1e-6)


def test_attention_reference_causal_masking():
    q = torch.randn(1, 1, 8, 4)
    k = torch.randn(1, 1, 8, 4)
    v = torch.randn(1, 1, 8, 4)
    out_full = R.attention(q, k, v, causal=True)
    # first position can only attend to itself
    want0 = v[0, 0, 0]
    torch.testing.assert_close(out_full[0, 0, 0], want0, rtol=1e-5, atol=1e-6)
