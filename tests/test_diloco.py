"""Single-process DiLoCo engine tests (CPU)."""

import torch

from hypha_amd import checkpoint, models
from hypha_amd.data.synthetic import SyntheticTokens
from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig


def make_worker(h=3, lr=1e-3):
    torch.manual_seed(0)
    model = models.build("llama-tiny")
    cfg = DiLoCoConfig(h=h, inner=InnerOptConfig(lr=lr, warmup_steps=0, schedule="constant"))
    return DiLoCoWorker(model, cfg, comm=Comm(), device=torch.device("cpu"))


def test_inner_steps_reduce_loss():
    w = make_worker(h=100)
    data = SyntheticTokens(512, 32, 4, seed=3)
    # overfit one batch: loss must drop
    ids, labels = data.next_batch()
    first = w.train_step(ids, labels)
    for _ in range(10):
        last = w.train_step(ids, labels)
    assert last < first, (first, last)


def test_outer_sync_runs_and_counts():
    w = make_worker(h=2)
    data = SyntheticTokens(512, 32, 2, seed=4)
    synced = 0
    for _ in range(4):
        ids, labels = data.next_batch()
        w.train_step(ids, labels)
        synced += int(w.maybe_outer_sync())
    assert synced == 2
    stats = w.comm_stats()
    assert stats["outer_rounds"] == 2
    assert stats["outer_sync_payload_bytes"] == 2 * w.fp.numel * 2  # bf16 comm dtype


def test_outer_sync_world1_moves_toward_theta_t():
    """With world=1, delta = theta_t - theta0 and mu=0, lr=1: theta_new = theta_t."""
    w = make_worker(h=1)
    w.cfg.outer.lr = 1.0
    w.cfg.outer.momentum = 0.0
    data = SyntheticTokens(512, 32, 2, seed=5)
    ids, labels = data.next_batch()
    w.train_step(ids, labels)
    theta_t = w.fp.master.clone()
    w.outer_sync()
    torch.testing.assert_close(w.fp.theta0, theta_t, rtol=1e-5, atol=1e-5)


def test_flat_param_views_alias_model():
    w = make_worker()
    p0 = next(w.model.parameters())
    w.fp.flat.add_(1.0)
    assert torch.allclose(p0.data, w.fp.flat[: p0.numel()].view(p0.shape))


def test_checkpoint_roundtrip(tmp_path):
    w = make_worker(h=2)
    data = SyntheticTokens(512, 32, 2, seed=6)
    for _ in range(3):
        ids, labels = data.next_batch()
        w.train_step(ids, labels)
        w.maybe_outer_sync()
    checkpoint.save_checkpoint(w, str(tmp_path))

    w2 = make_worker(h=2)
    manifest = checkpoint.load_checkpoint(w2, str(tmp_path))
    assert manifest["inner_step_count"] == 3
    torch.testing.assert_close(w2.fp.master, w.fp.master)
    torch.testing.assert_close(w2.fp.outer_momentum, w.fp.outer_momentum)
    torch.testing.assert_close(w2.fp.exp_avg, w.fp.exp_avg)
    assert w2.inner_step_count == w.inner_step_count
    assert w2.round == w.round

    # both continue identically
    ids, labels = data.next_batch()
    l1 = w.train_step(ids.clone(), labels.clone())
    l2 = w2.train_step(ids, labels)
    assert abs(l1 - l2) < 1e-6


def test_checkpoint_crash_safety(tmp_path):
    """Atomic checkpoint swap: a crash mid-swap leaves `.bak`, which
    load_checkpoint falls back to; a second save cleans it up."""
    import os

    from hypha_amd.parallel import DiLoCoConfig, DiLoCoWorker, InnerOptConfig

    torch.manual_seed(3)
    m = models.build("llama-tiny")
    cfg = DiLoCoConfig(h=4, inner=InnerOptConfig(warmup_steps=0, schedule="constant"))
    w = DiLoCoWorker(m, cfg, device=torch.device("cpu"))
    w.round = 7
    ck = str(tmp_path / "ckpt")
    checkpoint.save_checkpoint(w, ck)
    assert not os.path.exists(ck + ".tmp") and not os.path.exists(ck + ".bak")

    # simulate a crash after parking the old dir but before the new rename
    os.rename(ck, ck + ".bak")
    m2 = models.build("llama-tiny")
    w2 = DiLoCoWorker(m2, cfg, device=torch.device("cpu"))
    manifest = checkpoint.load_checkpoint(w2, ck)
    assert manifest["round"] == 7  # restored from .bak

    # a fresh save replaces everything and removes the bak
    checkpoint.save_checkpoint(w, ck)
    assert os.path.exists(os.path.join(ck, "manifest.json"))
    assert not os.path.exists(ck + ".bak")
