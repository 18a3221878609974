import torch

from hypha_amd import models


def test_llama_tiny_forward_backward():
    torch.manual_seed(0)
    m = models.build("llama-tiny")
    ids = torch.randint(0, 512, (2, 32))
    loss = m(ids, labels=ids)
    assert loss.dim() == 0 and torch.isfinite(loss)
    loss.backward()
    grads = [p.grad for p in m.parameters() if p.requires_grad]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)


def test_gpt2_tiny_forward_backward():
    torch.manual_seed(0)
    m = models.build("gpt2-tiny")
    ids = torch.randint(0, 512, (2, 32))
    loss = m(ids, labels=ids)
    assert torch.isfinite(loss)
    loss.backward()


def test_llama_param_count_8b():
    cfg = models.llama.PRESETS["llama3-8b"]
    n = cfg.num_params()
    assert 7.5e9 < n < 8.5e9, n


def test_loss_near_uniform_at_init():
    import math

    torch.manual_seed(0)
    m = models.build("llama-tiny")
    ids = torch.randint(0, 512, (2, 64))
    loss = float(m(ids, labels=ids))
    assert abs(loss - math.log(512)) < 1.0, loss


def test_registry_lists_flagships():
    av = models.available()
    for name in ("llama3-8b", "llama3-70b", "gpt2-small", "llama-tiny"):
        assert name in av


def test_moe_tiny_forward_backward():
    torch.manual_seed(0)
    m = models.build("moe-tiny")
    ids = torch.randint(0, 512, (2, 32))
    loss = m(ids, labels=ids)
    assert torch.isfinite(loss)
    loss.backward()
    # every expert and the router must receive gradient
    blk = m.blocks[0].mlp
    assert blk.router.weight.grad is not None
    assert blk.w_gate.grad is not None and torch.isfinite(blk.w_gate.grad).all()


def test_moe_routing_weights_normalized():
    torch.manual_seed(1)
    m = models.build("moe-tiny")
    m.eval()
    ids = torch.randint(0, 512, (1, 16))
    out = m(ids)
    assert out.shape == (1, 16, 512)


def test_mixtral_param_count():
    cfg = models.moe.PRESETS["mixtral-8x7b"]
    n = cfg.num_params()
    assert 45e9 < n < 48e9, n  # Mixtral-8x7B ~46.7B params


def test_generate_with_kv_cache_matches_recompute():
    """Cached decode must produce the same greedy tokens as full recompute."""
    torch.manual_seed(0)
    m = models.build("llama-tiny")
    m.eval()
    ids = torch.randint(0, 512, (1, 8))
    out = m.generate(ids, max_new_tokens=6)
    assert out.shape == (1, 14)
    # full-recompute greedy reference
    cur = ids.clone()
    for _ in range(6):
        logits = m(cur)
        cur = torch.cat([cur, logits[:, -1].argmax(-1, keepdim=True)], dim=1)
    torch.testing.assert_close(out, cur)


def test_generate_sampling_reproducible():
    torch.manual_seed(0)
    m = models.build("llama-tiny")
    ids = torch.randint(0, 512, (1, 4))
    a = m.generate(ids, max_new_tokens=5, temperature=0.8, top_k=20, seed=7)
    b = m.generate(ids, max_new_tokens=5, temperature=0.8, top_k=20, seed=7)
    torch.testing.assert_close(a, b)


def test_gpt2_generate():
    """Every registry family exposes the same generate surface (inference
    jobs may name any model); GPT-2 decodes by full recompute."""
    torch.manual_seed(4)
    m = models.build("gpt2-tiny")
    ids = torch.randint(0, 512, (2, 8))
    out = m.generate(ids, max_new_tokens=5)
    assert out.shape == (2, 13)
    assert torch.equal(out[:, :8], ids)
    a = m.generate(ids, max_new_tokens=4, temperature=0.9, top_k=10, seed=3)
    b = m.generate(ids, max_new_tokens=4, temperature=0.9, top_k=10, seed=3)
    assert torch.equal(a, b)  # seeded sampling is reproducible


def test_moe_generate():
    torch.manual_seed(5)
    m = models.build("moe-tiny")
    ids = torch.randint(0, 512, (1, 8))
    out = m.generate(ids, max_new_tokens=4)
    assert out.shape == (1, 12) and torch.equal(out[:, :8], ids)
