"""End-to-end cluster test — BASELINE config 1 at tiny scale: gateway +
data node + 3 worker daemons (2 train + 1 parameter server) + scheduler,
all real processes on loopback, training llama-tiny DiLoCo for 2 outer
rounds on CPU. The reference's quickstart (docs/quickstart.md) is the
manual version of this; here it is automated."""

import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "bin"


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def binaries():
    if not (BIN / "hypha-gateway").exists():
        subprocess.run(
            [sys.executable, "setup.py", "build_ext", "--inplace"], cwd=REPO, check=True
        )
    return BIN


@pytest.mark.timeout(300)
def test_cluster_diloco_round_trip(binaries, tmp_path):
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)

    gw_port = free_port()
    trace_file = tmp_path / "traces.jsonl"
    env = dict(os.environ, PYTHONPATH=str(REPO),
               HYPHA_TRACE_FILE=str(trace_file))
    procs = []
    logs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        logs[name] = tmp_path / f"{name}.log"
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)

        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 2, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE, stderr=open(tmp_path / "sched.log", "w"),
            text=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (
            out,
            *[f"--- {n}: {p.read_text()[-2000:]}" for n, p in logs.items()],
            (tmp_path / "sched.log").read_text()[-3000:],
        )
        assert sched.returncode == 0
        # OTLP spans from the executors: dispatch + rounds + outer syncs
        import json as _json

        spans = []
        if trace_file.exists():
            for line in trace_file.read_text().splitlines():
                batch = _json.loads(line)
                for rs in batch["resourceSpans"]:
                    for ss in rs["scopeSpans"]:
                        spans.extend(sp["name"] for sp in ss["spans"])
        assert "job.dispatch" in spans, spans
        assert "diloco.round" in spans and "diloco.outer_sync" in spans, spans
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_worker_kill_and_rejoin(binaries, tmp_path):
    """BASELINE config 3 at tiny scale: kill a train worker's daemon mid-run;
    the scheduler must detect the lease failure, shrink the round, auction a
    replacement, catch it up via the PS's cumulative offset, and finish."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []
    worker_procs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(4):  # 2 train + 1 PS + 1 spare for the replacement
            worker_procs[f"worker-{i}"] = spawn(
                f"worker{i}",
                [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                 "--exec-cmd", exec_cmd, "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        # enough rounds that the 10 s lease TTL always expires (and the
        # replacement joins) well before training can complete on its own
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 12, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
            start_new_session=True,
        )
        procs.append(sched)

        # wait until some train worker finishes round 1, then kill its daemon
        victim = None
        deadline = time.time() + 120
        while victim is None and time.time() < deadline:
            time.sleep(1)
            for i in range(4):
                log = (tmp_path / f"worker{i}.log")
                if log.exists() and "round 1 merged" in log.read_text():
                    victim = f"worker-{i}"
                    break
        assert victim is not None, "no worker reached round 1"
        worker_procs[victim].send_signal(signal.SIGKILL)

        out, _ = sched.communicate(timeout=200)
        sched_log = (tmp_path / "sched.log").read_text()
        assert "Job is completed." in out, (out, sched_log[-3000:])
        assert "lost" in sched_log and "joined" in sched_log, sched_log[-3000:]
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_checkpointing(binaries, tmp_path):
    """Train jobs write periodic checkpoints when the job config asks for
    them (checkpoint_every_rounds), in the SafeTensors+manifest format."""
    import json as pyjson

    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", 4, 16, 512, 128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(2):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        ckpt_root = tmp_path / "ckpt"
        cfg = tmp_path / "job.json"
        cfg.write_text(pyjson.dumps({
            "model": "llama-tiny", "dataset": "synth", "num_workers": 1,
            "update_rounds": 2, "avg_samples_between_updates": 8,
            "batch_size": 2, "seq_len": 128, "inner_lr": 0.001,
            "checkpoint_every_rounds": 1, "checkpoint_dir": str(ckpt_root),
        }))
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True)
        procs.append(sched)
        out, _ = sched.communicate(timeout=200)
        assert "Job is completed." in out, (tmp_path / "sched.log").read_text()[-2000:]
        manifest = pyjson.loads((ckpt_root / "manifest.json").read_text())
        assert manifest["round"] == 2
        assert (ckpt_root / "0_global_weights.safetensors").exists()
        assert (ckpt_root / "optimizer_state.safetensors").exists()
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_diloco_over_mtls(binaries, tmp_path):
    """The full cluster flow with every control-plane connection mutually
    authenticated (certutil PKI -> --tls-* flags on all daemons)."""
    from hypha_amd.data.synthetic import write_slice_files

    pki = tmp_path / "pki"
    tool = REPO / "tools" / "hypha_certutil.py"
    run = lambda *a: subprocess.run([sys.executable, str(tool), *a], check=True)
    run("root", "--out", str(pki))
    run("org", "--out", str(pki), "--name", "org1")
    for n in ("gw", "data-node", "scheduler", "worker-0", "worker-1", "worker-2"):
        run("node", "--out", str(pki), "--org", "org1", "--name", n)

    def tls(n):
        return ["--tls-cert", str(pki / f"{n}.chain.pem"),
                "--tls-key", str(pki / f"{n}.key"),
                "--tls-ca", str(pki / "root.crt")]

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", 4, 16, 512, 128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port), *tls("gw")])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir),
                       *tls("data-node")])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(tmp_path / f"work{i}"),
                                 *tls(f"worker-{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 2, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg), *tls("scheduler")],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True)
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (tmp_path / "sched.log").read_text()[-3000:]
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_scheduler_death_reclaims_workers(binaries, tmp_path):
    """SURVEY §5 failure-detection parity: scheduler death means no renewals,
    so workers' 250 ms pruner expires the 10 s leases and cancels the orphan
    executor subprocesses (arbiter.rs:98-141 semantics — no zombie jobs)."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1",
                                 "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 50, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = spawn("sched", [str(BIN / "hypha-scheduler"), "--name", "scheduler",
                                "--gateway-host", "127.0.0.1",
                                "--gateway-port", str(gw_port),
                                "--config", str(cfg)])

        # wait until training actually started (an executor is running)
        deadline = time.time() + 90
        started = False
        while time.time() < deadline and not started:
            time.sleep(1)
            for i in range(3):
                log = tmp_path / f"worker{i}.log"
                if log.exists() and "[executor] model=" in log.read_text():
                    started = True
        assert started, "no job was ever dispatched"

        sched.send_signal(signal.SIGKILL)  # scheduler dies mid-run
        sched.wait(timeout=10)

        # train executors exit when their status posts fail (the reference's
        # training.py would equally crash on a dead scheduler); the PS job has
        # no scheduler dependency mid-round, so the LEASE machinery must
        # cancel it: renewals stop -> 10 s TTL expires -> pruner kills it
        deadline = time.time() + 40
        cancelled = 0
        while time.time() < deadline and cancelled < 1:
            time.sleep(1)
            cancelled = sum(
                "cancelled job" in (tmp_path / f"worker{i}.log").read_text()
                for i in range(3)
                if (tmp_path / f"worker{i}.log").exists()
            )
        assert cancelled >= 1, [
            (tmp_path / f"worker{i}.log").read_text()[-1500:] for i in range(3)
        ]

        # the definitive reclamation check: a SECOND scheduler can buy the
        # same workers again and run a fresh job to completion
        cfg2 = tmp_path / "job2.json"
        cfg2.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 2, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        time.sleep(12)  # let every stale lease age out
        sched2 = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler2",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg2)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched2.log", "w"), text=True,
        )
        procs.append(sched2)
        out, _ = sched2.communicate(timeout=180)
        assert "Job is completed." in out, (
            out, (tmp_path / "sched2.log").read_text()[-3000:])
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_heterogeneous_batch_sizes(binaries, tmp_path):
    """Performance-aware scheduling parity (hypha-scheduler.rs:320-322 +
    rfc/2025-10-16): a worker offering 2 GPUs (--offer-strategy whole) gets
    2x the per-worker batch of a 1-GPU worker; the FSM still closes rounds."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        # big: 2 GPUs, sold whole; small: 1 GPU; third: PS
        spawn("workerbig", [str(BIN / "hypha-worker"), "--name", "worker-big",
                            "--gateway-host", "127.0.0.1",
                            "--gateway-port", str(gw_port), "--gpu", "2",
                            "--offer-strategy", "whole",
                            "--executors", "diloco-transformer",
                            "--exec-cmd", exec_cmd,
                            "--work-root", str(tmp_path / "workbig")])
        for nm, execs in (("small", "diloco-transformer"),
                          ("ps", "parameter-server")):
            spawn(f"worker{nm}", [str(BIN / "hypha-worker"), "--name", f"worker-{nm}",
                                  "--gateway-host", "127.0.0.1",
                                  "--gateway-port", str(gw_port), "--gpu", "1",
                                  "--executors", execs,
                                  "--exec-cmd", exec_cmd,
                                  "--work-root", str(tmp_path / f"work{nm}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 2, "avg_samples_between_updates": 12,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001,'
            ' "weighted_aggregation": true}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
            start_new_session=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (
            out, (tmp_path / "sched.log").read_text()[-3000:])

        logs = "".join((tmp_path / f"worker{n}.log").read_text()
                       for n in ("big", "small", "ps")
                       if (tmp_path / f"worker{n}.log").exists())
        big_log = (tmp_path / "workerbig.log").read_text()
        # the 2-GPU whole-offer worker trained with bs=4, some worker with bs=2
        assert "bs=4" in big_log, big_log[-1500:]
        assert "bs=2" in logs, logs[-1500:]
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_inference_jobs(binaries, tmp_path):
    """Inference through the full cluster: the scheduler's `generate` mode
    auctions inference-capable workers, dispatches generation jobs over the
    bridge contract, slices flow through the data scheduler, and completions
    land as SafeTensors in the job work dirs."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=8,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        infer_cmd = (f"{sys.executable} -m hypha_amd.runtime.infer_executor "
                     "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(2):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1",
                                 "--gateway-port", str(gw_port),
                                 "--exec-cmd", "true", "--infer-cmd", infer_cmd,
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"job_type": "generate", "model": "llama-tiny", "dataset": "synth",'
            ' "num_workers": 2, "update_rounds": 1,'
            ' "avg_samples_between_updates": 1, "batch_size": 2, "seq_len": 128,'
            ' "max_new_tokens": 8, "num_batches": 2}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
            start_new_session=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (
            out, (tmp_path / "sched.log").read_text()[-3000:],
            *[(tmp_path / f"worker{i}.log").read_text()[-1500:] for i in range(2)],
        )
        # completions exist and extend the prompts by max_new_tokens
        from safetensors.torch import load_file

        comps = list(tmp_path.glob("work*/hypha-*/completion-*.safetensors"))
        assert len(comps) >= 4, comps  # 2 workers x 2 batches
        toks = load_file(str(comps[0]))["tokens"]
        assert toks.shape[-1] == 128 + 8
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_parameter_server_loss_fails_fast(binaries, tmp_path):
    """The aggregate executor is a single point per job (as in the
    reference): killing its worker mid-run must abort the job with a clear
    error within ~a lease period, not hang until executor timeouts."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []
    worker_procs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            worker_procs[i] = spawn(
                f"worker{i}",
                [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                 "--exec-cmd", exec_cmd, "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 50, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
            start_new_session=True,
        )
        procs.append(sched)

        # find the PS worker (its log shows "PS start"), then kill it
        ps_idx = None
        deadline = time.time() + 90
        while ps_idx is None and time.time() < deadline:
            time.sleep(1)
            for i in range(3):
                log = tmp_path / f"worker{i}.log"
                if log.exists() and "PS start" in log.read_text():
                    ps_idx = i
                    break
        assert ps_idx is not None, "no PS job started"
        worker_procs[ps_idx].send_signal(signal.SIGKILL)

        out, _ = sched.communicate(timeout=120)
        assert sched.returncode == 1, (out, sched.returncode)
        sched_log = (tmp_path / "sched.log").read_text()
        assert "parameter server lost" in sched_log, sched_log[-2000:]
    finally:
        for p in procs:
            try:  # kill the whole group: daemons AND their executors —
                # even if the daemon itself already died (orphan children)
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_rccl_sync_round_trip(binaries, tmp_path):
    """Control-plane-orchestrated collective data plane:
    with `sync: "rccl"` the scheduler assigns {rank, world_size, rendezvous}
    in each dispatched job, there is NO parameter-server job, and the outer
    sync is a bucketed all-reduce (gloo here, RCCL on GPU) + replicated
    Nesterov. Both workers must finish with bitwise-identical global weights."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(2):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)

        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 2, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001,'
            ' "sync": "rccl", "rccl_timeout_s": 60,'
            ' "checkpoint_every_rounds": 1}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        logs = "".join((tmp_path / f"worker{i}.log").read_text() for i in range(2))
        assert "Job is completed." in out, (
            out, (tmp_path / "sched.log").read_text()[-3000:], logs[-3000:])
        assert sched.returncode == 0
        # the RCCL path actually ran (not the PS file path)
        assert "rccl round" in logs, logs[-2000:]
        assert "PS start" not in logs
        # bitwise-identical global weights on both workers after the final sync
        from safetensors.torch import load_file

        ckpts = sorted(tmp_path.glob("work*/hypha-job-train-*/checkpoint/"
                                     "0_global_weights.safetensors"))
        assert len(ckpts) == 2, ckpts
        t0 = load_file(str(ckpts[0]))["theta_global"]
        t1 = load_file(str(ckpts[1]))["theta_global"]
        assert (t0 == t1).all(), float((t0 - t1).abs().max())
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_rccl_kill_and_rejoin(binaries, tmp_path):
    """BASELINE config 3 on the collective data plane: kill a train worker
    mid-run under `sync: "rccl"`. The surviving worker's collective fails, the
    scheduler auctions a replacement and re-forms the communicator on a fresh
    rendezvous (new ranks, new port), the joiner catches up via the rank-0
    state broadcast, and training completes."""
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=4, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []
    worker_procs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        spawn("data", [str(BIN / "hypha-data"), "--name", "data-node",
                       "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                       "--dataset", "synth", "--dataset-path", str(data_dir)])
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):  # 2 train + 1 spare for the replacement
            worker_procs[f"worker-{i}"] = spawn(
                f"worker{i}",
                [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                 "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
                 "--exec-cmd", exec_cmd, "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)
        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny", "dataset": "synth", "num_workers": 2,'
            ' "update_rounds": 12, "avg_samples_between_updates": 8,'
            ' "batch_size": 2, "seq_len": 128, "inner_lr": 0.001,'
            ' "sync": "rccl", "rccl_timeout_s": 30}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
            start_new_session=True,
        )
        procs.append(sched)

        # wait until some train worker finishes round 1, then kill its daemon
        victim = None
        deadline = time.time() + 120
        while victim is None and time.time() < deadline:
            time.sleep(1)
            for i in range(3):
                log = (tmp_path / f"worker{i}.log")
                if log.exists() and "rccl round 1 merged" in log.read_text():
                    victim = f"worker-{i}"
                    break
        assert victim is not None, "no worker reached round 1"
        os.killpg(worker_procs[victim].pid, signal.SIGKILL)

        out, _ = sched.communicate(timeout=220)
        sched_log = (tmp_path / "sched.log").read_text()
        logs = "".join((tmp_path / f"worker{i}.log").read_text()
                       for i in range(3) if (tmp_path / f"worker{i}.log").exists())
        assert "Job is completed." in out, (out, sched_log[-3000:], logs[-3000:])
        assert "re-formed RCCL group" in sched_log, sched_log[-3000:]
        assert "joins as rank" in sched_log, sched_log[-3000:]
        # the survivors actually re-formed (executor-side evidence)
        assert "reform -> rank" in logs, logs[-3000:]
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass
