"""Multi-process (gloo, world_size=2) DiLoCo tests — BASELINE config 1
('GPT-2-small DiLoCo H=10, 2 CPU worker processes on gloo') at tiny scale."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        from hypha_amd import models
        from hypha_amd.data.synthetic import SyntheticTokens
        from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

        torch.manual_seed(100 + rank)  # intentionally different init; broadcast fixes it
        model = models.build("llama-tiny")
        cfg = DiLoCoConfig(h=2, inner=InnerOptConfig(lr=1e-3, warmup_steps=0, schedule="constant"))
        comm = Comm(backend="gloo")
        w = DiLoCoWorker(model, cfg, comm=comm, device=torch.device("cpu"))

        # after init broadcast all ranks hold identical weights
        digest0 = float(w.fp.master.sum())

        data = SyntheticTokens(512, 32, 2, seed=11, rank=rank)
        for _ in range(4):  # 2 rounds of h=2
            ids, labels = data.next_batch()
            w.train_step(ids, labels)
            w.maybe_outer_sync()

        digest1 = float(w.fp.master.sum())
        l2 = float(w.fp.master.norm())
        q.put(("ok", rank, digest0, digest1, l2, w.round))
        comm.shutdown()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put(("err", rank, traceback.format_exc(), str(e), 0.0, 0))


@pytest.mark.timeout(180)
def test_two_process_gloo_diloco():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29541
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
    for r in results:
        assert r[0] == "ok", r
    results.sort(key=lambda r: r[1])
    # identical start (broadcast) and identical post-sync global weights
    assert results[0][2] == pytest.approx(results[1][2], rel=1e-6)
    assert results[0][3] == pytest.approx(results[1][3], rel=1e-6)
    assert results[0][5] == 2  # two outer rounds completed


@pytest.mark.timeout(300)
def test_bench_contract_world2_gloo(tmp_path):
    """The driver's scale run (torch.distributed.run, one rank per GPU)
    exercised end-to-end on CPU/gloo: rendezvous on 127.0.0.1, bucketed
    all-reduce outer sync inside the timed window, max-over-ranks timing,
    and exactly ONE JSON line from rank 0 honoring the bench contract."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613", str(repo / "bench.py"), "--gpus", "2",
         "--steps", "4", "--warmup", "1", "--model", "llama-tiny",
         "--batch", "2", "--seq-len", "128", "--h", "3"],
        cwd=repo, capture_output=True, text=True, timeout=280,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, r.stdout  # rank 0 only
    out = json.loads(json_lines[0])
    assert out["n_gpus"] == 2 and out["steps"] == 4
    assert out["value"] > 0 and out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    cfg = out["config"]
    assert cfg["parallelism"] == "diloco-dp2"
    assert cfg["global_batch"] == 4  # whole-job aggregate
    assert cfg["outer_syncs_in_timed_window"] >= 1  # phase-shifted cadence
    assert cfg["outer_sync_wire_bytes_per_rank"] > 0  # ring wire accounting


def _elastic_worker(rank, port, q):
    """World starts at 3; rank 2 dies after round 1; survivors reform to
    world 2 on a fresh rendezvous and complete another round."""
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = "3"
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        from hypha_amd import models
        from hypha_amd.data.synthetic import SyntheticTokens
        from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

        torch.manual_seed(200 + rank)
        model = models.build("llama-tiny")
        cfg = DiLoCoConfig(h=2, inner=InnerOptConfig(lr=1e-3, warmup_steps=0,
                                                     schedule="constant"))
        comm = Comm(backend="gloo")
        w = DiLoCoWorker(model, cfg, comm=comm, device=torch.device("cpu"))
        data = SyntheticTokens(512, 32, 2, seed=13, rank=rank)
        for _ in range(2):  # round 1
            ids, labels = data.next_batch()
            w.train_step(ids, labels)
            w.maybe_outer_sync()

        if rank == 2:  # this worker "dies" after round 1
            q.put(("dead", rank, 0.0, 0))
            comm.shutdown()
            return

        # survivors: scheduler would assign new ranks {0,1} + a fresh port
        comm.reform(rank=rank, world_size=2, master_port=port + 1)
        for _ in range(2):  # round 2 with the shrunken world
            ids, labels = data.next_batch()
            w.train_step(ids, labels)
            w.maybe_outer_sync()
        q.put(("ok", rank, float(w.fp.master.sum()), w.round))
        comm.shutdown()
    except Exception:  # pragma: no cover
        import traceback

        q.put(("err", rank, traceback.format_exc(), 0))


@pytest.mark.timeout(180)
def test_elastic_reform_after_worker_death():
    """Config-3 semantics on the collective path: the communicator is
    re-formed on membership change at an outer-sync boundary (Comm.reform —
    RCCL groups cannot shrink in place; re-bootstrap is amortized over H)."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29551
    procs = [ctx.Process(target=_elastic_worker, args=(r, port, q)) for r in range(3)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(3)]
    for p in procs:
        p.join(timeout=120)
    ok = sorted(r for r in results if r[0] == "ok")
    assert len(ok) == 2 and any(r[0] == "dead" for r in results), results
    # both survivors completed 2 rounds and hold identical global weights
    assert ok[0][3] == 2 and ok[1][3] == 2
    assert ok[0][2] == pytest.approx(ok[1][2], rel=1e-6)


@pytest.mark.timeout(300)
def test_bench_expert_parallel_world2_gloo():
    """bench.py --expert-parallel under torch.distributed.run (gloo, 2
    ranks): experts shard, tokens exchange, the JSON contract holds and
    parallelism reports ep2+diloco."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29617", str(repo / "bench.py"), "--gpus", "2",
         "--steps", "3", "--warmup", "1", "--model", "moe-tiny",
         "--batch", "2", "--seq-len", "128", "--h", "2",
         "--expert-parallel"],
        cwd=repo, capture_output=True, text=True, timeout=280,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, r.stdout
    out = json.loads(json_lines[0])
    assert out["config"]["parallelism"] == "ep2+diloco"
    assert out["value"] > 0
    assert out["config"]["outer_syncs_in_timed_window"] >= 1


def _weighted_worker(rank, port, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK=str(rank),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    try:
        from hypha_amd.parallel import Comm

        comm = Comm(backend="gloo")
        # rank 0 processed 3 samples with delta=1; rank 1 processed 1 with
        # delta=5 -> weighted mean = (3*1 + 1*5)/4 = 2.0
        d = torch.full((1000,), 1.0 if rank == 0 else 5.0)
        w = 3.0 if rank == 0 else 1.0
        total = comm.weighted_all_reduce_flat(d, w)
        # zero-weight edge: both ranks weight 0 -> plain mean, no nan
        z = torch.full((10,), float(rank))
        comm.weighted_all_reduce_flat(z, 0.0)
        q.put(("ok", rank, float(d[0]), float(d[-1]), total,
               bool(torch.isfinite(z).all())))
        comm.shutdown()
    except Exception:
        import traceback

        q.put(("err", rank, traceback.format_exc(), 0, 0, False))


@pytest.mark.timeout(120)
def test_weighted_all_reduce_flat():
    """Sample-weighted outer aggregation (PS weighting semantics on the
    RCCL path): sum(w_i d_i)/sum(w_i), zero-weight degenerates to mean."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_weighted_worker, args=(r, 29821, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=90)
    for r in res:
        assert r[0] == "ok", r
        assert r[2] == pytest.approx(2.0, rel=1e-5)
        assert r[3] == pytest.approx(2.0, rel=1e-5)
        assert r[4] == pytest.approx(4.0)
        assert r[5]
