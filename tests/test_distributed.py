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
    assert cfg["outer_syncs_in_timed_window"] == 1  # h=3 < steps+warmup
    assert cfg["outer_sync_wire_bytes_per_rank"] > 0  # ring wire accounting
