"""Scheduler-driven throughput bench: the flagship DiLoCo config running
THROUGH the control plane (gateway -> auction -> lease -> dispatch -> bridge)
with the outer sync on the RCCL data plane (`sync: "rccl"`), instead of the
bare bench.py loop. VERDICT r1 item 1's acceptance check: this should land
within ~2% of bare `bench.py` at the same N/batch/seq/H, because the only
control-plane cost in steady state is one loopback status RPC per batch.

Usage (one MI355X):
  python tools/cluster_bench.py --gpus 1 --rounds 2 --h 30
Spawns 1 gateway + 1 data node + N worker daemons (each pinned to one GPU via
--gpu-ids) + a scheduler, runs `rounds` outer rounds of llama3-8b DiLoCo, and
reports per-round tokens/s (sum over workers) parsed from the scheduler's
metrics stream. Round 0 is discarded (its window contains compile/cache
warmup).
"""

from __future__ import annotations

import argparse
import json
import os
import re
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "bin"
sys.path.insert(0, str(REPO))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--batch", type=int, default=6)
    p.add_argument("--seq-len", type=int, default=2048)
    p.add_argument("--h", type=int, default=100, help="inner steps per round")
    p.add_argument("--rounds", type=int, default=2)
    p.add_argument("--work", default="/tmp/hypha-cluster-bench")
    args = p.parse_args()

    from hypha_amd.data.synthetic import write_slice_files

    work = Path(args.work)
    work.mkdir(parents=True, exist_ok=True)
    data_dir = work / "slices"
    if not (data_dir / "synth-00000.safetensors").exists():
        vocab = 128256 if "llama3" in args.model else 512
        write_slice_files(str(data_dir), "synth", num_slices=max(4, 2 * args.gpus),
                          samples_per_slice=4 * args.batch, vocab_size=vocab,
                          seq_len=args.seq_len)

    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(work / f"{name}.log", "w")
        proc = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                                start_new_session=True)
        procs.append(proc)
        return proc

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(args.gpus):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1",
                                 "--gateway-port", str(gw_port),
                                 "--gpu-ids", str(i),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(work / f"work{i}")])
        time.sleep(0.5)
        cfg = work / "job.json"
        cfg.write_text(json.dumps({
            "model": args.model, "dataset": "synth", "num_workers": args.gpus,
            "update_rounds": args.rounds,
            # FSM counts samples: H steps/worker/round at this batch size
            "avg_samples_between_updates": args.h * args.batch,
            "batch_size": args.batch, "seq_len": args.seq_len,
            "inner_lr": 4e-4, "sync": "rccl", "rccl_timeout_s": 600,
        }))
        sched_log = open(work / "sched.log", "w")
        t0 = time.time()
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE, stderr=sched_log, text=True)
        procs.append(sched)
        out, _ = sched.communicate()
        wall = time.time() - t0
        ok = "Job is completed." in (out or "")
        # parse per-worker, per-round tokens/s from the scheduler metrics log
        metrics = []  # (worker, round, tokens_per_sec)
        pat = re.compile(r"\[metrics\] (\S+) (\{.*\})")
        for line in (work / "sched.log").read_text().splitlines():
            m = pat.search(line)
            if not m:
                continue
            try:
                d = json.loads(m.group(2))
            except json.JSONDecodeError:
                continue
            if "tokens_per_sec" in d:
                metrics.append((m.group(1), d["tokens_per_sec"], d.get("samples", 0)))
        # steady-state = drop each worker's first round (warmup/compile)
        per_worker: dict[str, list[float]] = {}
        for w, tps, _ in metrics:
            per_worker.setdefault(w, []).append(tps)
        steady = [tps for rounds in per_worker.values() for tps in rounds[1:]] or [
            tps for rounds in per_worker.values() for tps in rounds]
        node_tps = 0.0
        for w, rounds in per_worker.items():
            vals = rounds[1:] or rounds
            node_tps += sum(vals) / len(vals)
        result = {
            "metric": "tokens/sec (node) via control plane (auction+lease+dispatch+RCCL)",
            "value": node_tps,
            "unit": "tokens/s",
            "n_gpus": args.gpus,
            "rounds": args.rounds,
            "h": args.h,
            "wall_s": wall,
            "completed": ok,
            "per_worker_round_tps": {w: [round(v, 1) for v in r]
                                     for w, r in per_worker.items()},
            "config": {"model": args.model, "global_batch": args.batch * args.gpus,
                       "seq_len": args.seq_len, "sync": "rccl"},
        }
        print(json.dumps(result), flush=True)
        return 0 if ok else 1
    finally:
        for proc in procs:
            try:
                os.killpg(proc.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if proc.poll() is None:
                    proc.send_signal(signal.SIGKILL)
        for proc in procs:
            try:
                proc.wait(timeout=10)
            except Exception:
                pass


if __name__ == "__main__":
    sys.exit(main())
