"""DiLoCo training executor: the subprocess a worker daemon spawns for a
Train job. Speaks only the job-bridge API ({SOCKET_PATH}/{WORK_DIR}/{JOB_JSON}
contract, executor/process.rs:201-205).

Semantics mirror the reference executor
(/root/reference/executors/accelerate/.../training.py): train batches and
send a Status per batch until the scheduler's ScheduleUpdate counter runs
out; then extract the pseudo-gradient (theta_t - theta_0, utils.py:118-123),
push it to the parameter server, wait for the aggregated Nesterov update via
SSE, merge (theta <- theta_0 + U, utils.py:105-115), and continue or stop on
Done. Compute runs on this framework's engine (fused kernels on GPU, the
reference op path on CPU) instead of HF Accelerate.
"""

from __future__ import annotations

import argparse
import json
import os
import queue
import sys
import threading

import torch


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--socket", required=True)
    p.add_argument("--work-dir", required=True)
    p.add_argument("--job", required=True)
    args = p.parse_args()

    from safetensors.torch import load_file, save_file

    from hypha_amd import models
    from hypha_amd.data.synthetic import load_slice
    from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig
    from hypha_amd.runtime.session import Session

    with open(args.job) as f:
        cfg = json.load(f)

    session = Session(args.socket)
    torch.manual_seed(0)  # all workers start from the same init
    model = models.build(cfg["model"])
    device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
    inner = InnerOptConfig(
        lr=cfg.get("optimizer", {}).get("adam", {}).get("learning_rate", 4e-4),
        warmup_steps=0, schedule="constant",
    )
    worker = DiLoCoWorker(
        model, DiLoCoConfig(h=1 << 30, inner=inner), comm=Comm(), device=device
    )
    batch_size = int(cfg.get("batch_size", 4))
    seq_len = int(cfg.get("seq_len", 128))
    data_ref = cfg["data"]
    updates_ref = cfg["updates"]
    ckpt_dir = cfg.get("checkpoint_dir") or os.path.join(args.work_dir, "checkpoint")
    ckpt_every = int(cfg.get("checkpoint_every_rounds", 0))

    # resume from a previous run's checkpoint if one exists
    resume_round = 0
    if os.path.exists(os.path.join(ckpt_dir, "manifest.json")) or os.path.exists(
        os.path.join(ckpt_dir + ".bak", "manifest.json")
    ):
        from hypha_amd import checkpoint as ckpt_mod

        manifest = ckpt_mod.load_checkpoint(worker, ckpt_dir)
        resume_round = int(manifest["round"])
        print(f"[executor] resumed from round {resume_round}", flush=True)

    # background SSE listener for aggregated updates
    updates_q: "queue.Queue[dict]" = queue.Queue()

    def listen():
        try:
            for ev in session.receive():
                updates_q.put(ev)
        except Exception as e:  # bridge gone = worker daemon died: abort
            # instead of training on as an orphan (grad-release of the whole
            # job happens via the daemon's lease machinery, not us)
            print(f"# sse listener ended: {type(e).__name__}: {e}", file=sys.stderr)
        updates_q.put({"__bridge_closed__": True})

    threading.Thread(target=listen, daemon=True).start()

    def watchdog():
        # the worker daemon owns this job: if its bridge stops answering,
        # exit rather than train on as an orphan (daemon SIGKILL leaves no
        # one to SIGTERM us; the reference has the same subprocess model)
        import httpx as _httpx

        misses = 0
        while True:
            import time as _time

            _time.sleep(15)
            try:
                _httpx.Client(transport=_httpx.HTTPTransport(uds=args.socket),
                              timeout=5.0).get("http://bridge/openapi.json")
                misses = 0
            except Exception:
                misses += 1
                if misses >= 2:
                    print("# bridge unreachable: exiting orphaned executor",
                          file=sys.stderr)
                    os._exit(3)

    threading.Thread(target=watchdog, daemon=True).start()

    # infinite batch stream over scheduler-assigned slices (utils.py fetch_data)
    def batches():
        while True:
            got = session.fetch(data_ref)
            for path in got["files"]:
                ids = load_slice(path)
                for i in range(0, ids.shape[0] - batch_size + 1, batch_size):
                    b = ids[i : i + batch_size, :seq_len]
                    yield b, b.clone()

    batch_iter = batches()
    print(f"[executor] model={cfg['model']} device={device} bs={batch_size}", flush=True)

    if cfg.get("join"):
        # replacement worker (kill/rejoin path): catch up to the current
        # global weights via the PS's cumulative offset before training
        print("[executor] joining: waiting for global offset", flush=True)
        ev = updates_q.get(timeout=600)
        if ev.get("__bridge_closed__"):
            raise RuntimeError("bridge connection lost (worker daemon gone)")
        offset = load_file(ev["path"])
        if "delta" in offset:
            worker.fp.theta0.add_(offset["delta"].to(worker.fp.theta0.device))
            worker.fp.master.copy_(worker.fp.theta0)
            worker.fp.flat.copy_(worker.fp.master)
        print("[executor] joined at current global weights", flush=True)

    # round-start global weights on disk (training.py:61-63 parity:
    # `0_global_weights`; SafeTensors as the reference's checkpoint format)
    save_file({"flat": worker.fp.theta0.cpu()},
              os.path.join(args.work_dir, "0_global_weights.safetensors"))

    done = False
    round_idx = resume_round
    while not done:
        # ---- inner loop: train until the scheduler's counter is exhausted ----
        import time as _time

        round_t0 = _time.perf_counter()
        remaining = None
        round_samples = 0
        while remaining is None or remaining > 0:
            ids, labels = next(batch_iter)
            loss = worker.train_step(ids, labels)
            round_samples += batch_size
            if remaining is not None:
                remaining -= 1
            resp = session.send_status({"kind": "status", "batch_size": batch_size})
            if resp.get("kind") == "schedule-update" and remaining is None:
                remaining = int(resp.get("counter", 0))
        round_s = max(1e-9, _time.perf_counter() - round_t0)
        session.send_status(
            {"kind": "metrics", "round": round_idx,
             "metrics": {"loss": loss,
                         "tokens_per_sec": round_samples * seq_len / round_s,
                         "samples": round_samples}}
        )

        # ---- extract and push the pseudo-gradient ----
        session.send_status({"kind": "update"})
        delta = (worker.fp.master - worker.fp.theta0).cpu()
        fname = f"{round_idx}_local_gradients.safetensors"
        save_file({"delta": delta}, os.path.join(args.work_dir, fname))
        # sample count rides with the push for optional weighted aggregation
        session.send_resource({**updates_ref, "samples": round_samples}, fname)

        # ---- wait for the aggregated Nesterov update, merge ----
        ev = updates_q.get(timeout=600)
        if ev.get("__bridge_closed__"):
            raise RuntimeError("bridge connection lost (worker daemon gone)")
        u = load_file(ev["path"])["delta"].to(worker.fp.theta0.device)
        worker.fp.theta0.add_(u)
        worker.fp.master.copy_(worker.fp.theta0)
        worker.fp.flat.copy_(worker.fp.master)

        resp = session.send_status({"kind": "update-received"})
        done = resp.get("kind") == "done"
        round_idx += 1
        worker.round = round_idx
        if ckpt_every and (round_idx % ckpt_every == 0 or done):
            from hypha_amd import checkpoint as ckpt_mod

            ckpt_mod.save_checkpoint(worker, ckpt_dir)
            print(f"[executor] checkpoint saved at round {round_idx}", flush=True)
        print(f"[executor] round {round_idx} merged, loss={loss:.4f} done={done}",
              flush=True)

    session.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
