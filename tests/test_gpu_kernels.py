"""GPU (MI355X) numerics tests: every HIP kernel vs a plain PyTorch fp32
reference of the same op (the reference repo's golden-value pattern,
parameter_server.rs:448-525)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from hypha_amd import _C, ops
    from hypha_amd.ops import reference as R

DEV = "cuda:0"


def rand_bf16(*shape, seed=0, scale=1.0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).bfloat16().to(DEV)


# ---------------------------------------------------------------------------
# MFMA fragment-layout probes (transpose-detecting: asymmetric B — guide G9)
# ---------------------------------------------------------------------------

def test_mfma_layout_32x32x16():
    a = rand_bf16(32, 16, seed=1)
    b = rand_bf16(16, 32, seed=2)
    got = _C.mfma_probe_32x32x16(a, b)
    want = a.float() @ b.float()
    torch.testing.assert_close(got, want, rtol=2e-2, atol=2e-2)


def test_mfma_layout_16x16x32():
    a = rand_bf16(16, 32, seed=3)
    b = rand_bf16(32, 16, seed=4)
    got = _C.mfma_probe_16x16x32(a, b)
    want = a.float() @ b.float()
    torch.testing.assert_close(got, want, rtol=2e-2, atol=2e-2)


# ---------------------------------------------------------------------------
# Fused optimizers
# ---------------------------------------------------------------------------

def test_adamw_matches_reference():
    n = 12345  # not a multiple of 8: exercises the tail path
    torch.manual_seed(0)
    master = torch.randn(n, device=DEV)
    m = torch.rand(n, device=DEV) * 0.1
    v = torch.rand(n, device=DEV) * 0.01
    param = master.bfloat16()
    cm, cmm, cv = master.cpu().clone(), m.cpu().clone(), v.cpu().clone()
    cparam = cm.clone()
    for step in (1, 2, 3):
        g = torch.randn(n).bfloat16()
        _C.adamw_step_(master, param, g.to(DEV), m, v, 1e-2, 0.9, 0.95, 1e-8, 0.1, step)
        R.adamw_step(cm, cparam, g.float(), cmm, cv,
                     lr=1e-2, beta1=0.9, beta2=0.95, eps=1e-8, weight_decay=0.1, step=step)
    torch.testing.assert_close(master.cpu(), cm, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(m.cpu(), cmm, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(v.cpu(), cv, rtol=1e-5, atol=1e-7)
    torch.testing.assert_close(param.cpu().float(), cm, rtol=1e-2, atol=1e-2)


def test_nesterov_matches_torch_sgd():
    n = 1001
    torch.manual_seed(1)
    theta = torch.randn(n, device=DEV)
    mom = torch.zeros(n, device=DEV)
    p_ref = theta.cpu().clone().requires_grad_(True)
    opt = torch.optim.SGD([p_ref], lr=0.7, momentum=0.9, nesterov=True)
    for i in range(4):
        delta = torch.randn(n).bfloat16()
        _C.nesterov_step_(theta, delta.to(DEV), mom, 0.7, 0.9)
        p_ref.grad = -delta.float()
        opt.step()
    torch.testing.assert_close(theta.cpu(), p_ref.detach(), rtol=1e-4, atol=1e-4)


def test_extract_delta():
    n = 999
    a = torch.randn(n, device=DEV)
    b = torch.randn(n, device=DEV)
    out = torch.empty(n, dtype=torch.bfloat16, device=DEV)
    _C.extract_delta(a, b, out)
    torch.testing.assert_close(out.cpu().float(), (a - b).cpu().bfloat16().float())


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------

def test_swiglu_fwd_bwd():
    g = rand_bf16(64, 256, seed=5, scale=2.0).requires_grad_(True)
    u = rand_bf16(64, 256, seed=6, scale=2.0).requires_grad_(True)
    out = ops.swiglu(g, u)
    want = R.swiglu(g.detach().float(), u.detach().float())
    torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2)
    dout = rand_bf16(64, 256, seed=7)
    out.backward(dout)
    gf = g.detach().float().requires_grad_(True)
    uf = u.detach().float().requires_grad_(True)
    R.swiglu(gf, uf).backward(dout.float())
    torch.testing.assert_close(g.grad.float(), gf.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(u.grad.float(), uf.grad, rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------

def test_rmsnorm_fwd_bwd():
    x = rand_bf16(128, 512, seed=8).requires_grad_(True)
    w = (torch.randn(512) * 0.1 + 1.0).bfloat16().to(DEV).requires_grad_(True)
    y = ops.rmsnorm(x, w, 1e-5)
    want = R.rmsnorm(x.detach().float(), w.detach().float(), 1e-5)
    torch.testing.assert_close(y.float(), want, rtol=2e-2, atol=2e-2)
    dy = rand_bf16(128, 512, seed=9)
    y.backward(dy)
    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    R.rmsnorm(xf, wf, 1e-5).backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=3e-2, atol=3e-1)


# ---------------------------------------------------------------------------
# LayerNorm + GELU (GPT-2 block ops, ops/hip/layernorm.hip)
# ---------------------------------------------------------------------------

def test_layernorm_fwd_bwd():
    import torch.nn.functional as F

    x = rand_bf16(128, 512, seed=18).requires_grad_(True)
    w = (torch.randn(512) * 0.1 + 1.0).bfloat16().to(DEV).requires_grad_(True)
    b = (torch.randn(512) * 0.1).bfloat16().to(DEV).requires_grad_(True)
    y = ops.layernorm(x, w, b, 1e-5)
    want = F.layer_norm(x.detach().float(), (512,), w.detach().float(),
                        b.detach().float(), 1e-5)
    torch.testing.assert_close(y.float(), want, rtol=2e-2, atol=2e-2)
    dy = rand_bf16(128, 512, seed=19)
    y.backward(dy)
    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    F.layer_norm(xf, (512,), wf, bf, 1e-5).backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=3e-2, atol=3e-1)
    torch.testing.assert_close(b.grad.float(), bf.grad, rtol=3e-2, atol=3e-1)


def test_gelu_fwd_bwd():
    import torch.nn.functional as F

    x = rand_bf16(1000, 256, seed=20).requires_grad_(True)
    y = ops.gelu(x)
    want = F.gelu(x.detach().float(), approximate="tanh")
    torch.testing.assert_close(y.float(), want, rtol=2e-2, atol=2e-2)
    dy = rand_bf16(1000, 256, seed=21)
    y.backward(dy)
    xf = x.detach().float().requires_grad_(True)
    F.gelu(xf, approximate="tanh").backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=3e-2)


def test_gpt2_block_native_matches_cpu():
    """The GPT-2 model forward+backward on GPU (native LayerNorm/GELU/attention)
    vs the same weights on CPU (torch reference path)."""
    from hypha_amd import models

    torch.manual_seed(5)
    # gpt2-small: head_dim 64 and seq >= 128 satisfy the attention kernel
    m_cpu = models.build("gpt2-small")
    m_gpu = models.build("gpt2-small")
    m_gpu.load_state_dict(m_cpu.state_dict())
    m_gpu = m_gpu.to(DEV, torch.bfloat16)
    ids = torch.randint(0, m_cpu.cfg.vocab_size, (2, 128))
    loss_cpu = m_cpu(ids, labels=ids)
    loss_gpu = m_gpu(ids.to(DEV), labels=ids.to(DEV))
    assert abs(float(loss_cpu) - float(loss_gpu)) < 0.05 * max(1.0, float(loss_cpu))
    loss_gpu.backward()  # exercises the native backward path end-to-end
    assert all(p.grad is not None for p in m_gpu.parameters() if p.requires_grad)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

def test_rope_fwd_bwd():
    cos, sin = R.rope_cos_sin(128, 64, base=10000.0)
    cos, sin = cos.to(DEV), sin.to(DEV)
    q = rand_bf16(2, 4, 128, 64, seed=10).requires_grad_(True)
    k = rand_bf16(2, 2, 128, 64, seed=11).requires_grad_(True)
    qo, ko = ops.apply_rope_qk(q, k, cos, sin)
    qw = R.apply_rope(q.detach().float().cpu(), cos.cpu(), sin.cpu())
    kw = R.apply_rope(k.detach().float().cpu(), cos.cpu(), sin.cpu())
    torch.testing.assert_close(qo.float().cpu(), qw, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(ko.float().cpu(), kw, rtol=2e-2, atol=2e-2)
    (qo.float().pow(2).sum() + ko.float().pow(2).sum()).backward()
    # gradient of sum of squares through a rotation: dq = 2 * R^-1 R q = 2q
    torch.testing.assert_close(q.grad.float(), 2 * q.detach().float(), rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# Cross-entropy
# ---------------------------------------------------------------------------

def test_ce_fwd_bwd():
    torch.manual_seed(2)
    N, V = 256, 1000
    logits = rand_bf16(N, V, seed=12, scale=3.0).requires_grad_(True)
    t = torch.randint(0, V, (N,))
    t[5] = -100
    t[77] = -100
    tg = t.to(DEV)
    loss = ops.cross_entropy_loss(logits, tg)
    lf = logits.detach().float().requires_grad_(True)
    want = torch.nn.functional.cross_entropy(lf, tg, ignore_index=-100)
    assert abs(float(loss.detach()) - float(want.detach())) < 2e-2 * max(1.0, abs(float(want)))
    loss.backward()
    want.backward()
    torch.testing.assert_close(logits.grad.float(), lf.grad, rtol=5e-2, atol=1e-4)


# ---------------------------------------------------------------------------
# Flash attention
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("hd,hq,hkv", [(128, 8, 2), (64, 4, 4)])
def test_attention_fwd(hd, hq, hkv):
    B, S = 2, 256
    q = rand_bf16(B, hq, S, hd, seed=13)
    k = rand_bf16(B, hkv, S, hd, seed=14)
    v = rand_bf16(B, hkv, S, hd, seed=15)
    o = ops.flash_attention(q, k, v, causal=True)
    want = R.attention(q.float(), k.float(), v.float(), causal=True)
    torch.testing.assert_close(o.float(), want, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("hd,hq,hkv", [(128, 8, 2), (64, 4, 4)])
def test_attention_bwd(hd, hq, hkv):
    B, S = 2, 256
    q = rand_bf16(B, hq, S, hd, seed=16).requires_grad_(True)
    k = rand_bf16(B, hkv, S, hd, seed=17).requires_grad_(True)
    v = rand_bf16(B, hkv, S, hd, seed=18).requires_grad_(True)
    o = ops.flash_attention(q, k, v, causal=True)
    do = rand_bf16(B, hq, S, hd, seed=19)
    o.backward(do)
    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    R.attention(qf, kf, vf, causal=True).backward(do.float())
    torch.testing.assert_close(q.grad.float(), qf.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), kf.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), vf.grad, rtol=5e-2, atol=5e-2)


def test_attention_noncausal():
    q = rand_bf16(1, 2, 128, 128, seed=20)
    k = rand_bf16(1, 2, 128, 128, seed=21)
    v = rand_bf16(1, 2, 128, 128, seed=22)
    o = ops.flash_attention(q, k, v, causal=False)
    want = R.attention(q.float(), k.float(), v.float(), causal=False)
    torch.testing.assert_close(o.float(), want, rtol=2e-2, atol=2e-2)


# ---------------------------------------------------------------------------
# End-to-end: tiny model steps on GPU with native ops
# ---------------------------------------------------------------------------

def test_model_step_gpu():
    from hypha_amd import models
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

    assert ops.has_native()
    torch.manual_seed(0)
    model = models.build("llama-tiny")  # head_dim 64
    w = DiLoCoWorker(model, DiLoCoConfig(h=2, inner=InnerOptConfig(warmup_steps=0)),
                     comm=Comm(), device=torch.device(DEV))
    data = SyntheticTokens(512, 128, 2, seed=30)
    ids, labels = data.next_batch()
    first = w.train_step(ids, labels)
    losses = [first]
    for _ in range(8):
        losses.append(w.train_step(ids.clone(), labels.clone()))
        w.maybe_outer_sync()
    assert all(math.isfinite(l) for l in losses), losses
    assert losses[-1] < losses[0], losses


def test_attention_bshd_matches_bhsd():
    """The stride-aware BSHD path must equal the BHSD path (fwd + bwd)."""
    B, S, Hq, Hkv, D = 2, 256, 8, 2, 128
    qs = rand_bf16(B, S, Hq, D, seed=40).requires_grad_(True)
    ks = rand_bf16(B, S, Hkv, D, seed=41).requires_grad_(True)
    vs = rand_bf16(B, S, Hkv, D, seed=42).requires_grad_(True)
    o_bshd = ops.flash_attention(qs, ks, vs, causal=True, layout="bshd")

    qb = qs.detach().transpose(1, 2).contiguous().requires_grad_(True)
    kb = ks.detach().transpose(1, 2).contiguous().requires_grad_(True)
    vb = vs.detach().transpose(1, 2).contiguous().requires_grad_(True)
    o_bhsd = ops.flash_attention(qb, kb, vb, causal=True)
    torch.testing.assert_close(
        o_bshd.transpose(1, 2).contiguous(), o_bhsd, rtol=1e-3, atol=1e-3
    )
    do = rand_bf16(B, S, Hq, D, seed=43)
    o_bshd.backward(do)
    o_bhsd.backward(do.transpose(1, 2).contiguous())
    torch.testing.assert_close(
        qs.grad.transpose(1, 2).contiguous(), qb.grad, rtol=1e-3, atol=1e-3
    )
    torch.testing.assert_close(
        ks.grad.transpose(1, 2).contiguous(), kb.grad, rtol=1e-3, atol=1e-3
    )
    torch.testing.assert_close(
        vs.grad.transpose(1, 2).contiguous(), vb.grad, rtol=1e-3, atol=1e-3
    )


def test_rope_bshd_matches_bhsd():
    cos, sin = R.rope_cos_sin(256, 128, base=10000.0)
    cos, sin = cos.to(DEV), sin.to(DEV)
    q = rand_bf16(2, 256, 4, 128, seed=44)
    k = rand_bf16(2, 256, 2, 128, seed=45)
    qo, ko = ops.apply_rope_qk(q, k, cos, sin, layout="bshd")
    qb, kb = ops.apply_rope_qk(
        q.transpose(1, 2).contiguous(), k.transpose(1, 2).contiguous(), cos, sin
    )
    torch.testing.assert_close(qo.transpose(1, 2).contiguous(), qb)
    torch.testing.assert_close(ko.transpose(1, 2).contiguous(), kb)


def test_moe_model_step_gpu():
    from hypha_amd import models
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

    torch.manual_seed(0)
    model = models.build("moe-tiny")
    w = DiLoCoWorker(model, DiLoCoConfig(h=2, inner=InnerOptConfig(warmup_steps=0)),
                     comm=Comm(), device=torch.device(DEV))
    data = SyntheticTokens(512, 128, 2, seed=31)
    ids, labels = data.next_batch()
    first = w.train_step(ids, labels)
    for _ in range(6):
        last = w.train_step(ids.clone(), labels.clone())
    assert math.isfinite(last) and last < first, (first, last)


def test_adamw8_tracks_fp32_adamw():
    """8-bit-state AdamW must track full-precision AdamW closely over steps
    (config 5 building block: m/v blockwise-quantized to uint8)."""
    n = 5000
    torch.manual_seed(3)
    master = torch.randn(n, device=DEV)
    param = master.bfloat16()
    m8 = torch.full((n,), 127, dtype=torch.uint8, device=DEV)
    v8 = torch.zeros(n, dtype=torch.uint8, device=DEV)
    nblocks = (n + 2047) // 2048
    m_scale = torch.full((nblocks,), 1e-12, device=DEV)
    v_scale = torch.full((nblocks,), 1e-12, device=DEV)
    ref_master = master.cpu().clone()
    ref_m = torch.zeros(n)
    ref_v = torch.zeros(n)
    ref_param = ref_master.clone()
    for step in range(1, 20):
        g = torch.randn(n).bfloat16()
        _C.adamw8_step_(master, param, g.to(DEV), m8, v8, m_scale, v_scale,
                        1e-2, 0.9, 0.95, 1e-8, 0.0, step)
        R.adamw_step(ref_master, ref_param, g.float(), ref_m, ref_v,
                     lr=1e-2, beta1=0.9, beta2=0.95, eps=1e-8, weight_decay=0.0,
                     step=step)
    err = (master.cpu() - ref_master).abs().max()
    drift = (master.cpu() - ref_master).norm() / ref_master.norm()
    assert float(err) < 0.05 and float(drift) < 0.01, (float(err), float(drift))


def test_fp8_linear_fwd_bwd():
    """fp8 (e4m3) scaled-GEMM linear vs bf16 reference (config 5 path)."""
    from hypha_amd.ops.fp8 import Fp8Linear

    torch.manual_seed(4)
    lin = Fp8Linear(2048, 1024).to(DEV)
    lin.weight.data = lin.weight.data.bfloat16().float()  # freeze a bf16-able init
    lin = lin.bfloat16().to(DEV)
    x = rand_bf16(64, 2048, seed=50).requires_grad_(True)
    out = lin(x)
    ref = torch.nn.functional.linear(x.detach().float(), lin.weight.detach().float())
    rel = (out.float() - ref).norm() / ref.norm()
    assert float(rel) < 0.05, float(rel)
    out.float().pow(2).sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert lin.weight.grad is not None and torch.isfinite(lin.weight.grad).all()
    # gradient direction sanity vs autograd bf16 reference
    xf = x.detach().float().requires_grad_(True)
    wf = lin.weight.detach().float().requires_grad_(True)
    torch.nn.functional.linear(xf, wf).pow(2).sum().backward()
    cos = torch.nn.functional.cosine_similarity(
        x.grad.float().flatten(), xf.grad.flatten(), dim=0
    )
    assert float(cos) > 0.98, float(cos)


def test_fp8_convert_model_step():
    from hypha_amd import models
    from hypha_amd.ops.fp8 import convert_linears_to_fp8
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

    torch.manual_seed(0)
    model = models.build("llama-tiny", hidden_size=2048, n_heads=16, n_kv_heads=16,
                         ffn_hidden=4096, n_layers=1)
    n = convert_linears_to_fp8(model)
    assert n >= 4  # qkvo + mlp projections
    w = DiLoCoWorker(model, DiLoCoConfig(h=4, inner=InnerOptConfig(warmup_steps=0)),
                     comm=Comm(), device=torch.device(DEV))
    data = SyntheticTokens(512, 128, 2, seed=32)
    ids, labels = data.next_batch()
    first = w.train_step(ids, labels)
    for _ in range(6):
        last = w.train_step(ids.clone(), labels.clone())
    assert math.isfinite(last) and last < first, (first, last)


def test_grad_norm_sq():
    x = rand_bf16(123457, seed=60)
    got = _C.grad_norm_sq(x).sqrt()
    want = torch.linalg.vector_norm(x, dtype=torch.float32)
    torch.testing.assert_close(got, want, rtol=1e-3, atol=1e-3)


def test_grouped_gemm_vs_per_expert_matmul():
    """Native MoE grouped GEMM vs per-expert torch matmul (ragged groups,
    incl. an empty group and a non-tile-multiple group)."""
    torch.manual_seed(7)
    K, N, E = 256, 512, 4
    counts = [200, 0, 128, 37]
    T = sum(counts)
    x = rand_bf16(T, K, seed=70)
    w = rand_bf16(E, N, K, seed=71, scale=0.1)
    off = torch.tensor([0] + list(torch.tensor(counts).cumsum(0)), dtype=torch.int32)
    got = _C.grouped_gemm(x, w, off)
    want = torch.empty(T, N, dtype=torch.bfloat16, device=DEV)
    s = 0
    for g, c in enumerate(counts):
        if c:
            want[s : s + c] = x[s : s + c].float() @ w[g].float().t()
        s += c
    torch.testing.assert_close(got.float(), want.float(), rtol=3e-2, atol=3e-2)


def test_moe_inference_native_grouped_path(monkeypatch):
    """MoE eval forward through the native grouped-GEMM kernel must agree
    with the torch per-expert path (opt-in route)."""
    from hypha_amd import models

    monkeypatch.setenv("HYPHA_NATIVE_GROUPED", "1")
    torch.manual_seed(8)
    m = models.build("moe-tiny", hidden_size=128, ffn_hidden=256)
    m = m.to(device=DEV, dtype=torch.bfloat16)
    for buf in m.buffers():
        if buf.dtype in (torch.bfloat16,):
            buf.data = buf.data.float()
    m.eval()
    ids = torch.randint(0, 512, (1, 128), device=DEV)
    with torch.no_grad():
        native = m(ids)
        with torch.enable_grad():  # forces the torch per-expert path
            ref = m(ids)
    torch.testing.assert_close(native.float(), ref.float(), rtol=3e-2, atol=3e-2)


def test_adamw8_lean_sr_tracks_fp32():
    """Master-free stochastically-rounded AdamW must stay unbiased: after
    many steps the bf16-SR weights track the fp32 reference in expectation
    (cosine of trajectories; also the small-update accumulation property
    that plain RNE bf16 fails)."""
    n = 4096
    torch.manual_seed(9)
    param = torch.randn(n, device=DEV).bfloat16()
    ref = param.float().cpu().clone()
    m8 = torch.full((n,), 127, dtype=torch.uint8, device=DEV)
    v8 = torch.zeros(n, dtype=torch.uint8, device=DEV)
    nb = (n + 2047) // 2048
    msc = torch.full((nb,), 1e-12, device=DEV)
    vsc = torch.full((nb,), 1e-12, device=DEV)
    rm = torch.zeros(n)
    rv = torch.zeros(n)
    rp = ref.clone()
    for step in range(1, 60):
        g = (torch.randn(n) * 0.1).bfloat16()
        _C.adamw8_lean_(param, g.to(DEV), m8, v8, msc, vsc,
                        1e-3, 0.9, 0.95, 1e-8, 0.0, step, step * 7919)
        R.adamw_step(ref, rp, g.float(), rm, rv,
                     lr=1e-3, beta1=0.9, beta2=0.95, eps=1e-8, weight_decay=0.0,
                     step=step)
    moved = (ref - rp).abs().mean()  # sanity: the reference moved
    drift = (param.cpu().float() - ref).norm() / ref.norm()
    assert float(drift) < 0.02, float(drift)


def test_lean_worker_trains_and_syncs():
    from hypha_amd import models
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, InnerOptConfig, LeanDiLoCoWorker

    torch.manual_seed(0)
    model = models.build("llama-tiny")
    w = LeanDiLoCoWorker(
        model, DiLoCoConfig(h=3, inner=InnerOptConfig(lr=1e-3, warmup_steps=0,
                                                      schedule="constant")),
        comm=Comm(), device=torch.device(DEV))
    data = SyntheticTokens(512, 128, 2, seed=33)
    ids, labels = data.next_batch()
    first = w.train_step(ids, labels)
    for _ in range(8):
        last = w.train_step(ids.clone(), labels.clone())
        w.maybe_outer_sync()
    assert math.isfinite(last) and last < first, (first, last)
    assert w.round >= 2
    # grads are released: no parameter should hold one after a step
    assert all(p.grad is None for p in w.params)


def test_lean_checkpoint_roundtrip(tmp_path):
    from hypha_amd import checkpoint, models
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, InnerOptConfig, LeanDiLoCoWorker

    def make():
        torch.manual_seed(0)
        return LeanDiLoCoWorker(
            models.build("llama-tiny"),
            DiLoCoConfig(h=2, inner=InnerOptConfig(lr=1e-3, warmup_steps=0,
                                                   schedule="constant")),
            comm=Comm(), device=torch.device(DEV))

    w = make()
    data = SyntheticTokens(512, 128, 2, seed=34)
    for _ in range(3):
        ids, labels = data.next_batch()
        w.train_step(ids, labels)
        w.maybe_outer_sync()
    checkpoint.save_checkpoint(w, str(tmp_path))
    w2 = make()
    manifest = checkpoint.load_checkpoint(w2, str(tmp_path))
    assert manifest["format"].startswith("hypha_amd.checkpoint.lean")
    torch.testing.assert_close(w2.flat, w.flat)
    torch.testing.assert_close(w2.theta0_host, w.theta0_host)
    assert torch.equal(w2.m8, w.m8) and torch.equal(w2.v8, w.v8)
    assert w2.round == w.round


def test_attention_bwd_noncausal():
    """Non-causal backward (v4 all-tiles-active path) vs the fp32 oracle."""
    B, S, hq, hkv, hd = 1, 256, 4, 2, 128
    q = rand_bf16(B, hq, S, hd, seed=50).requires_grad_(True)
    k = rand_bf16(B, hkv, S, hd, seed=51).requires_grad_(True)
    v = rand_bf16(B, hkv, S, hd, seed=52).requires_grad_(True)
    o = ops.flash_attention(q, k, v, causal=False)
    do = rand_bf16(B, hq, S, hd, seed=53)
    o.backward(do)
    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    R.attention(qf, kf, vf, causal=False).backward(do.float())
    torch.testing.assert_close(q.grad.float(), qf.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), kf.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(v.grad.float(), vf.grad, rtol=5e-2, atol=5e-2)


def test_fp8_cast_transpose():
    """Fused cast+transpose+amax kernel vs plain PyTorch quantization."""
    from hypha_amd import _C

    torch.manual_seed(7)
    for R, C in ((128, 256), (100, 72), (4096, 4096)):
        x = (torch.randn(R, C, device=DEV) * 3).bfloat16()
        scale = torch.tensor([0.05], dtype=torch.float32, device=DEV)
        n = _C.fp8_cast_grid_size(R, C)
        amax = torch.zeros(n, dtype=torch.float32, device=DEV)
        x8, x8t = _C.fp8_cast_transpose(x, scale, amax)
        assert x8.shape == (R, C) and x8t.shape == (C, R)
        ref = (x.float() / 0.05).clamp(-448, 448).to(torch.float8_e4m3fn)
        torch.testing.assert_close(x8.float(), ref.float(), rtol=0, atol=0)
        torch.testing.assert_close(x8t.float(), ref.t().float(), rtol=0, atol=0)
        torch.testing.assert_close(amax.max().item(), x.float().abs().max().item(),
                                   rtol=1e-3, atol=0)


def test_fp8_delayed_scale_reuse():
    """Weight casts are cached within one fp8 epoch and refresh after fp8_step."""
    from hypha_amd.ops import fp8 as f8

    lin = f8.Fp8Linear(256, 128).bfloat16().to(DEV)
    x = rand_bf16(64, 256, seed=60)
    lin(x)
    st = lin._state
    c1 = st._wcache
    lin(x)
    assert lin._state._wcache is c1  # same epoch: cache reused
    f8.fp8_step()
    lin(x)
    assert lin._state._wcache is not c1


def test_fp8_sr_requant_unbiased():
    """fp8_requant_ is stochastic-rounding-unbiased: the mean of many
    quantize->dequant draws converges to the source values."""
    from hypha_amd import _C

    torch.manual_seed(9)
    n = 4096
    src = (torch.randn(n, device=DEV) * 0.02).bfloat16()
    w8 = torch.empty(n, dtype=torch.uint8, device=DEV)
    ws = torch.empty(n // 2048, dtype=torch.float32, device=DEV)
    zero = torch.zeros(n, dtype=torch.bfloat16, device=DEV)
    acc = torch.zeros(n, dtype=torch.float64, device=DEV)
    deq = torch.empty(n, dtype=torch.bfloat16, device=DEV)
    draws = 64
    for s in range(draws):
        _C.fp8_requant_(src, w8, ws, 1234 + s * 7919)
        _C.fp8_extract_delta(w8, ws, zero, deq)
        acc += deq.double()
        # every draw is within one e4m3 quantum of the source
        q = ws.repeat_interleave(2048) * (2.0 ** -3) * 2  # generous bound
        assert ((deq.float() - src.float()).abs() <= 448 * ws.max() * 0.25).all()
    mean = (acc / draws).float()
    err = (mean - src.float()).abs().max()
    # SR mean error shrinks ~1/sqrt(draws) below the deterministic quantum
    det_err = 448 * float(ws.max()) / 16  # one quantum at max magnitude
    assert float(err) < det_err, (float(err), det_err)


def test_adamw8_fp8_lean_matches_fp32_reference():
    """Fused fp8-storage AdamW vs an fp32-master reference over 30 steps."""
    from hypha_amd import _C

    torch.manual_seed(11)
    n = 8192
    w0 = (torch.randn(n, device=DEV) * 0.05).float()
    ref_w = w0.clone()
    ref_m = torch.zeros(n, device=DEV)
    ref_v = torch.zeros(n, device=DEV)

    w8 = torch.empty(n, dtype=torch.uint8, device=DEV)
    ws = torch.empty(n // 2048, dtype=torch.float32, device=DEV)
    _C.fp8_requant_(w0.bfloat16(), w8, ws, 3)
    m8 = torch.full((n,), 127, dtype=torch.uint8, device=DEV)
    v8 = torch.zeros(n, dtype=torch.uint8, device=DEV)
    ms = torch.full((n // 2048,), 1e-12, device=DEV)
    vs = torch.full((n // 2048,), 1e-12, device=DEV)

    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.95, 1e-8, 0.0
    deq = torch.empty(n, dtype=torch.bfloat16, device=DEV)
    zero = torch.zeros(n, dtype=torch.bfloat16, device=DEV)
    for step in range(1, 31):
        g = torch.randn(n, device=DEV) * 0.01
        ref_m = b1 * ref_m + (1 - b1) * g
        ref_v = b2 * ref_v + (1 - b2) * g * g
        mh = ref_m / (1 - b1 ** step)
        vh = ref_v / (1 - b2 ** step)
        ref_w = ref_w - lr * mh / (vh.sqrt() + eps)
        _C.adamw8_fp8_lean_(w8, ws, g.bfloat16(), m8, v8, ms, vs,
                            lr, b1, b2, eps, wd, step, step * 104729)
    _C.fp8_extract_delta(w8, ws, zero, deq)
    drift = (deq.float() - ref_w).norm() / ref_w.norm()
    # e4m3 SR noise per element is ~quantum/sqrt(12) per step (quantum up
    # to 12.5% of the block amax), accumulating as a random walk: after 30
    # steps the expected relative drift is ~0.2-0.3. The assertion checks
    # the MATH is right (unbiased, well-directed), not bf16-grade precision
    # — that noise floor is the documented cost of 1-byte weights.
    assert float(drift) < 0.45, float(drift)
    cos = torch.nn.functional.cosine_similarity(deq.float(), ref_w, dim=0)
    assert float(cos) > 0.92, float(cos)
    # and the net update direction from init agrees with the reference
    upd_k = deq.float() - w0
    upd_r = ref_w - w0
    cos_u = torch.nn.functional.cosine_similarity(upd_k, upd_r, dim=0)
    assert float(cos_u) > 0.5, float(cos_u)


def test_fp8_weight_cast_transpose_matches_reference():
    """Block-scaled storage -> per-tensor GEMM operands vs a plain torch
    dequant+quantize reference."""
    from hypha_amd import _C

    torch.manual_seed(12)
    R, C = 512, 2048
    w = (torch.randn(R, C, device=DEV) * 0.03).bfloat16()
    w8s = torch.empty(R * C, dtype=torch.uint8, device=DEV)
    ws = torch.empty(R * C // 2048, dtype=torch.float32, device=DEV)
    _C.fp8_requant_(w.reshape(-1), w8s, ws, 5)
    # exact fp32 dequant of the STORED values (uint8 bytes ARE e4m3)
    deq2d = (w8s.view(torch.float8_e4m3fn).float().view(R, C)
             * ws.repeat_interleave(2048).view(R, C))

    scale = torch.tensor([float(deq2d.abs().max()) / 448.0],
                         dtype=torch.float32, device=DEV)
    n = _C.fp8_cast_grid_size(R, C)
    partials = torch.zeros(n, dtype=torch.float32, device=DEV)
    w8, w8t = _C.fp8_weight_cast_transpose(w8s.view(R, C), ws, scale, partials)
    ref = (deq2d / scale).clamp(-448, 448).to(torch.float8_e4m3fn).float()
    # identical up to RNE boundary ties from the fused multiply order:
    # allow one quantum on a tiny fraction of elements
    diff = (w8.float() - ref).abs()
    qmax = ref.abs().max() / 16
    assert (diff <= qmax).all() and (diff > 0).float().mean() < 1e-3
    torch.testing.assert_close(w8t.float(), w8.t().contiguous().float(),
                               rtol=0, atol=0)
    assert abs(partials.max().item() - deq2d.abs().max().item()) < 1e-3


def test_lean_fp8_worker_trains():
    """Config-5 path end to end at tiny scale: fp8 weight storage + fp8
    GEMMs + 8-bit optimizer + outer sync; loss decreases and a checkpoint
    round-trips."""
    from hypha_amd import models
    from hypha_amd.checkpoint import load_checkpoint, save_checkpoint
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, InnerOptConfig
    from hypha_amd.parallel.lean import LeanDiLoCoWorker

    torch.manual_seed(21)
    model = models.build("llama-tiny", hidden_size=2048, n_heads=16,
                         n_kv_heads=16, ffn_hidden=4096, n_layers=2)
    w = LeanDiLoCoWorker(model,
                         DiLoCoConfig(h=12, inner=InnerOptConfig(
                             lr=5e-3, warmup_steps=0, schedule="constant")),
                         comm=Comm(), device=torch.device(DEV),
                         fp8_weights=True)
    assert w.fp8_numel > 0 and w.fp8_numel % 2048 == 0
    # SR quantization noise per write has std ~sqrt(delta*quantum): the
    # update only beats the noise when delta ~>= quantum (docs/memory.md).
    # Memorize ONE batch at lr >= the e4m3 quantum so the trend is signal.
    data = SyntheticTokens(512, 128, 2, seed=33)
    ids, labels = data.next_batch()
    losses = []
    for _ in range(24):
        losses.append(w.train_step(ids.clone(), labels.clone()))
        w.maybe_outer_sync()
    assert w.round == 2
    first4 = sum(losses[:4]) / 4
    last4 = sum(losses[-4:]) / 4
    assert last4 < first4 - 0.3, losses
    assert all(torch.isfinite(torch.tensor(losses)))

    import tempfile

    with tempfile.TemporaryDirectory() as d:
        ckpt = d + "/ck"
        save_checkpoint(w, ckpt)
        w8_before = w.flat_w8.clone()
        w.flat_w8.zero_()
        w.theta0_fp8_host.zero_()
        load_checkpoint(w, ckpt)
        assert torch.equal(w.flat_w8, w8_before)


def test_attn_decode_kernel_vs_reference():
    """Flash-decoding kernel vs plain fp32 softmax attention over the
    valid cache prefix, across GQA ratios, head dims, and ragged lengths."""
    from hypha_amd import _C

    torch.manual_seed(31)
    for B, Hq, Hkv, D, T_alloc, t in (
        (4, 32, 8, 128, 2048, 2048),   # llama3-8b shape, full cache
        (2, 32, 8, 128, 1024, 1000),   # unaligned t
        (3, 8, 8, 64, 512, 3),         # MHA (gpt2-like), tiny prefix
        (2, 64, 8, 128, 768, 700),     # G=8 (llama3-70b grouping)
        (1, 16, 16, 64, 257, 257),     # odd alloc
    ):
        q = rand_bf16(B, Hq, D, seed=40 + t)
        kc = rand_bf16(B, T_alloc, Hkv, D, seed=41 + t)
        vc = rand_bf16(B, T_alloc, Hkv, D, seed=42 + t)
        o = _C.attn_decode(q, kc, vc, t)
        rep = Hq // Hkv
        kh = kc[:, :t].float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
        vh = vc[:, :t].float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
        scores = torch.einsum("bhd,bhtd->bht", q.float(), kh) / D ** 0.5
        p = torch.softmax(scores, dim=-1)
        ref = torch.einsum("bht,bhtd->bhd", p, vh)
        torch.testing.assert_close(o.float(), ref, rtol=2e-2, atol=2e-2)


def test_generate_kv_cache_matches_recompute():
    """Greedy KV-cache generation equals full-recompute decoding for every
    registry family (decode kernel + pre-allocated cache vs plain forward)."""
    from hypha_amd import models

    torch.manual_seed(17)
    # gpt2-small rather than gpt2-tiny: head_dim 16 has no native kernel
    for name, kw in (("llama-tiny", {}), ("gpt2-small", {}), ("moe-tiny", {})):
        model = models.build(name, **kw).to(DEV).bfloat16().eval()
        for buf in model.buffers():  # rope tables stay fp32
            if buf.dtype is torch.bfloat16:
                buf.data = buf.data.float()
        ids = torch.randint(0, 256, (2, 12), device=DEV)
        out = model.generate(ids, max_new_tokens=12)
        # full-recompute greedy reference (right-pad to the flash kernel's
        # 128-multiple; causal masking makes padding invisible to position t)
        toks = ids.clone()
        with torch.no_grad():
            for _ in range(12):
                t_len = toks.shape[1]
                pad = (-t_len) % 128
                padded = torch.nn.functional.pad(toks, (0, pad))
                logits = model(padded)[:, t_len - 1]
                toks = torch.cat([toks, logits.argmax(-1, keepdim=True)], dim=1)
        assert out.shape == toks.shape
        agree = (out == toks).float().mean().item()
        assert agree >= 0.95, (name, agree, out[:, -12:], toks[:, -12:])


def test_add_rmsnorm_fused_vs_reference():
    """Fused residual-add+RMSNorm fwd/bwd vs the fp32 composition."""
    from hypha_amd import ops

    torch.manual_seed(23)
    N, D = 512, 4096
    a = rand_bf16(N, D, seed=70).requires_grad_(True)
    b = rand_bf16(N, D, seed=71).requires_grad_(True)
    w = (torch.randn(D) * 0.1 + 1).bfloat16().to(DEV).requires_grad_(True)
    h, y = ops.add_rmsnorm(a, b, w, 1e-5)
    # downstream uses BOTH h (residual) and y (norm) like a transformer block
    loss = (h.float() * 0.3).sum() + (y.float() * 0.7).sum()
    loss.backward()

    af = a.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    hf = af + bf
    yf = hf * torch.rsqrt(hf.pow(2).mean(-1, keepdim=True) + 1e-5) * wf
    (hf * 0.3).sum().add((yf * 0.7).sum()).backward()
    torch.testing.assert_close(h.float(), hf, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float(), yf, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(a.grad.float(), af.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(b.grad.float(), bf.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=3e-2, atol=3e-1)


def test_graphed_decode_matches_eager_generate():
    """hipGraph-captured greedy decode equals the eager KV-cache decode."""
    from hypha_amd import models
    from hypha_amd.runtime.graphed_decode import GraphedDecoder

    torch.manual_seed(29)
    model = models.build("llama-tiny", hidden_size=2048, n_heads=16,
                         n_kv_heads=4, ffn_hidden=4096, n_layers=2)
    model = model.to(DEV).bfloat16().eval()
    for buf in model.buffers():
        if buf.dtype is torch.bfloat16:
            buf.data = buf.data.float()
    ids = torch.randint(0, 500, (4, 128), device=DEV)
    eager = model.generate(ids, max_new_tokens=24)
    dec = GraphedDecoder(model, batch=4, prompt_len=128, max_new=32)
    graphed = dec.generate(ids, max_new_tokens=24)
    assert graphed.shape == eager.shape
    agree = (graphed == eager).float().mean().item()
    assert agree >= 0.97, (agree, graphed[:, -24:], eager[:, -24:])
    # second call reuses the captured graph
    graphed2 = dec.generate(ids, max_new_tokens=24)
    assert torch.equal(graphed2, graphed)


def test_skinny_gemm_matches_matmul():
    """Decode GEMV kernel vs plain matmul across decode shapes."""
    from hypha_amd import _C

    torch.manual_seed(37)
    for M, N, K in ((8, 4096, 4096), (8, 14336, 4096), (8, 4096, 14336),
                    (1, 1024, 4096), (4, 128256, 4096), (8, 4096, 4100)):
        x = rand_bf16(M, K, seed=80 + N % 97, scale=0.5)
        w = rand_bf16(N, K, seed=81 + N % 97, scale=0.5)
        o = _C.skinny_gemm(x, w)
        ref = (x.float() @ w.float().t())
        torch.testing.assert_close(o.float(), ref, rtol=2e-2, atol=K ** 0.5 * 2e-2)


def test_fp8_kv_cache_decode_matches_bf16():
    """fp8 (e4m3 + per-row scales) KV cache: kernel parity vs the bf16
    cache within quantization tolerance, and greedy generation agreement."""
    from hypha_amd import _C, models
    from hypha_amd.models.kv_cache import KVCache

    torch.manual_seed(41)
    B, Hq, Hkv, D, T, t = 4, 32, 8, 128, 1024, 700
    q = rand_bf16(B, Hq, D, seed=90)
    kraw = rand_bf16(B, t, Hkv, D, seed=91)
    vraw = rand_bf16(B, t, Hkv, D, seed=92)
    c16 = KVCache(B, T, Hkv, D, DEV)
    c8 = KVCache(B, T, Hkv, D, DEV, quant="fp8")
    c16.append(kraw, vraw)
    c8.append(kraw, vraw)
    o16 = _C.attn_decode(q, c16.k, c16.v, t)
    o8 = _C.attn_decode_fp8(q, c8.k, c8.v, c8.k_scale, c8.v_scale, t)
    rel = (o8.float() - o16.float()).norm() / o16.float().norm()
    # e4m3 rows carry ~3% RMS quantization noise; on uniform-random KV the
    # attention output inherits it (~4%). Generation agreement below is the
    # functional bar.
    assert float(rel) < 0.06, float(rel)

    # end-to-end greedy generation with fp8 cache tracks the bf16 cache
    model = models.build("llama-tiny", hidden_size=2048, n_heads=16,
                         n_kv_heads=4, ffn_hidden=4096, n_layers=2)
    model = model.to(DEV).bfloat16().eval()
    for buf in model.buffers():
        if buf.dtype is torch.bfloat16:
            buf.data = buf.data.float()
    ids = torch.randint(0, 500, (2, 128), device=DEV)
    out16 = model.generate(ids, max_new_tokens=16)
    out8 = model.generate(ids, max_new_tokens=16, kv_quant="fp8")
    agree = (out16 == out8).float().mean().item()
    assert agree >= 0.9, agree

    # graphed decode with fp8 cache
    from hypha_amd.runtime.graphed_decode import GraphedDecoder

    dec = GraphedDecoder(model, batch=2, prompt_len=128, max_new=24,
                         kv_quant="fp8")
    g8 = dec.generate(ids, max_new_tokens=16)
    agree_g = (g8 == out8).float().mean().item()
    assert agree_g >= 0.9, agree_g
