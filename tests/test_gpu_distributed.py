"""Real-RCCL distributed test: two ranks on two GPUs with backend "nccl"
(= RCCL on ROCm) running DiLoCo outer syncs — the exact collective path the
8-GPU scaling bench uses. NCCL/RCCL forbids two ranks sharing one device, so
this skips on single-GPU boxes (the gloo world-2 tests cover the protocol
there; HSA_ENABLE_IPC_MODE_LEGACY=0 must stay in the env for RCCL IPC)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(
        not torch.cuda.is_available() or torch.cuda.device_count() < 2,
        reason="needs >= 2 GPUs (RCCL forbids rank-sharing one device)",
    ),
]


def _rank_main(rank, world, port, q):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    try:
        from hypha_amd import models
        from hypha_amd.data.synthetic import SyntheticTokens
        from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

        torch.manual_seed(100 + rank)  # init broadcast must reconcile this
        model = models.build("llama-tiny")
        comm = Comm(backend="nccl")
        w = DiLoCoWorker(
            model,
            DiLoCoConfig(h=2, inner=InnerOptConfig(lr=1e-3, warmup_steps=0,
                                                   schedule="constant")),
            comm=comm,
            device=torch.device("cuda", rank),
        )
        start = float(w.fp.master.sum())
        data = SyntheticTokens(512, 128, 2, seed=21, rank=rank)
        for _ in range(4):  # two outer rounds over real RCCL
            ids, labels = data.next_batch()
            w.train_step(ids, labels)
            w.maybe_outer_sync()
        torch.cuda.synchronize()
        end = float(w.fp.master.sum())
        norm = float(w.fp.master.norm())
        q.put(("ok", rank, start, end, norm, w.round))
        comm.shutdown()
    except Exception:
        import traceback

        q.put(("err", rank, traceback.format_exc(), "", 0.0, 0))


@pytest.mark.timeout(300)
def test_two_rank_rccl_diloco_on_one_gpu():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, 29771, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
    for r in results:
        assert r[0] == "ok", r
    results.sort(key=lambda r: r[1])
    # identical init (broadcast) and identical post-sync global weights
    assert results[0][2] == pytest.approx(results[1][2], rel=1e-5)
    assert results[0][3] == pytest.approx(results[1][3], rel=1e-5)
    assert results[0][5] == 2
