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


def _state_sync(worker, comm, round_idx: int) -> int:
    """Broadcast {theta_global, outer momentum, round} from rank 0 after a
    communicator (re)formation. For an unchanged membership this is a cheap
    no-op semantically (all ranks identical); for a joiner it IS the catch-up
    path — it replaces the reference PS's cumulative-offset file
    (parameter_server.rs sync_to) with one xGMI broadcast."""
    import torch as _torch

    if not comm.is_distributed:
        return round_idx
    fp = worker.fp
    comm.broadcast_flat(fp.theta0, src=0)
    comm.broadcast_flat(fp.outer_momentum, src=0)
    meta = _torch.tensor([round_idx, worker.inner_step_count], dtype=_torch.float64)
    if fp.theta0.is_cuda:
        meta = meta.to(fp.theta0.device)
    comm.broadcast_flat(meta, src=0)
    round_idx = int(meta[0].item())
    # adopt: master/params <- global weights (joiner starts the round clean;
    # survivors are at a round boundary where master==theta0 already)
    fp.master.copy_(fp.theta0)
    fp.flat.copy_(fp.master)
    worker.round = round_idx
    return round_idx


def run_rccl(worker, comm, session, batch_iter, reform_q, rccl, batch_size,
             seq_len, resume_round, ckpt_dir, ckpt_every, work_dir) -> int:
    """Control-plane-orchestrated RCCL DiLoCo loop: the scheduler assigned
    {rank, world_size, rendezvous}; the outer sync is a bucketed all-reduce of
    the pseudo-gradient over xGMI plus a replicated fused Nesterov step on
    every rank (no parameter-server star — SURVEY.md §2.10 C1/C2). Membership
    changes (worker kill/rejoin, BASELINE config 3) arrive as `reform` events;
    they are applied at the outer-sync boundary, where a collective timeout is
    also recovered by re-forming on the scheduler's newest rendezvous and
    retrying the sync.
    """
    import time as _time

    from hypha_amd import ops
    from safetensors.torch import save_file as _save_file

    fp = worker.fp
    # pseudo-gradient staging buffer for the full model: the elastic path
    # cannot mutate theta0 until the WHOLE all-reduce succeeded (a retry after
    # a membership change recomputes delta from the untouched master/theta0)
    delta = torch.empty(fp.numel, dtype=worker.cfg.comm_dtype, device=fp.master.device)

    def apply_reform(ev) -> None:
        print(f"[executor] reform -> rank {ev['rank']}/{ev['world_size']} "
              f"@ {ev.get('master_addr', '127.0.0.1')}:{ev['master_port']}", flush=True)
        comm.reform(int(ev["rank"]), int(ev["world_size"]),
                    ev.get("master_addr", "127.0.0.1"), int(ev["master_port"]),
                    timeout_s=float(ev.get("timeout_s", comm.timeout_s)))

    def drain_reforms(block_s: float = 0.0):
        """Apply the newest pending reform order, if any."""
        newest = None
        try:
            newest = reform_q.get(timeout=block_s) if block_s > 0 else reform_q.get_nowait()
            while True:
                newest = reform_q.get_nowait()
        except queue.Empty:
            pass
        if newest is not None:
            if newest.get("__bridge_closed__"):
                raise RuntimeError("bridge connection lost (worker daemon gone)")
            apply_reform(newest)
            return True
        return False

    # form the scheduler-assigned group (initial members and joiners alike;
    # a joiner blocks here until the survivors reach their sync boundary and
    # re-form onto the same rendezvous)
    comm.reform(int(rccl["rank"]), int(rccl["world_size"]),
                rccl.get("master_addr", "127.0.0.1"),
                int(rccl.get("master_port", 29531)), timeout_s=comm.timeout_s)
    round_idx = resume_round
    # initial state sync: no-op for a fresh group (identical seeds), the
    # catch-up path for a joiner dispatched into a running job
    round_idx = _state_sync(worker, comm, round_idx)
    _save_file({"flat": fp.theta0.cpu()},
               os.path.join(work_dir, "0_global_weights.safetensors"))

    from hypha_amd.telemetry import get_tracer

    tracer = get_tracer()
    job_span = tracer.start_span("job.execute",
                                 job_id=os.environ.get("HYPHA_JOB_ID", ""),
                                 rank=comm.rank, world_size=comm.world_size)
    done = False
    while not done:
        round_span = tracer.start_span("diloco.round", parent=job_span,
                                       round=round_idx, rank=comm.rank)
        round_t0 = _time.perf_counter()
        remaining = None
        round_samples = 0
        loss = float("nan")
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
        round_span.set_attribute("samples", round_samples)
        round_span.set_attribute("loss", float(loss))
        round_span.end()
        session.send_status(
            {"kind": "metrics", "round": round_idx,
             "metrics": {"loss": loss,
                         "tokens_per_sec": round_samples * seq_len / round_s,
                         "samples": round_samples}})
        session.send_status({"kind": "update"})

        # ---- outer sync boundary: apply membership changes, then sync ----
        sync_span = tracer.start_span("diloco.outer_sync", parent=job_span,
                                      round=round_idx, rank=comm.rank)
        if drain_reforms():
            round_idx = _state_sync(worker, comm, round_idx)
        for attempt in range(4):
            ops.interface.extract_delta(fp.master, fp.theta0, delta)
            try:
                # sample-weighted aggregation (PS-path parity on RCCL): a
                # rank that ran fewer batches this round — heterogeneous
                # FSM schedules, a joiner with 0 — counts proportionally
                comm.weighted_all_reduce_flat(delta, float(round_samples))
                break
            except Exception as e:
                # a member died mid-round: the collective timed out. Wait for
                # the scheduler's re-formation order (lease expiry detection
                # is ~10 s, well inside the comm timeout), re-form, retry.
                print(f"[executor] outer sync failed ({type(e).__name__}); "
                      f"waiting for reform (attempt {attempt + 1})", flush=True)
                if attempt == 3:
                    raise
                if not drain_reforms(block_s=comm.timeout_s + 30.0):
                    raise RuntimeError("no reform order after sync failure")
                round_idx = _state_sync(worker, comm, round_idx)
        ops.fused_nesterov(fp.theta0, delta, fp.outer_momentum,
                           lr=worker.cfg.outer.lr, mu=worker.cfg.outer.momentum)
        fp.master.copy_(fp.theta0)
        fp.flat.copy_(fp.master)
        worker.outer_sync_payload_bytes += delta.numel() * delta.element_size()
        sync_span.set_attribute("payload_bytes",
                                delta.numel() * delta.element_size())
        sync_span.end()

        # FSM ordering parity with the PS path: `updated` (advances the round)
        # must reach the scheduler before any `update-received` — rank 0 sends
        # it, then the barrier releases the other ranks
        if comm.rank == 0:
            session.send_status({"kind": "updated"})
        comm.barrier()
        resp = session.send_status({"kind": "update-received"})
        done = resp.get("kind") == "done"
        round_idx += 1
        worker.round = round_idx
        worker.steps_in_round = 0
        if ckpt_every and (round_idx % ckpt_every == 0 or done):
            from hypha_amd import checkpoint as ckpt_mod

            ckpt_mod.save_checkpoint(worker, ckpt_dir)
            print(f"[executor] checkpoint saved at round {round_idx}", flush=True)
        print(f"[executor] rccl round {round_idx} merged, loss={loss:.4f} "
              f"done={done}", flush=True)

    job_span.end()
    tracer.flush()
    comm.shutdown()
    session.close()
    return 0


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
    # dispatch span: covers worker-side spawn -> executor ready (start
    # timestamp injected by the worker daemon across the process boundary)
    from hypha_amd.telemetry import get_tracer as _get_tracer

    _tr = _get_tracer()
    _disp_ns = os.environ.get("HYPHA_DISPATCH_TS_NS")
    if _disp_ns:
        _d = _tr.start_span("job.dispatch",
                            job_id=os.environ.get("HYPHA_JOB_ID", ""))
        _d.start_ns = int(_disp_ns)
        _d.end()
        _tr.flush()
    torch.manual_seed(0)  # all workers start from the same init
    model = models.build(cfg["model"])
    device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
    inner = InnerOptConfig(
        lr=cfg.get("optimizer", {}).get("adam", {}).get("learning_rate", 4e-4),
        warmup_steps=0, schedule="constant",
    )
    # sync mode: "rccl" = scheduler-assigned rendezvous, outer sync is a
    # bucketed RCCL all-reduce + replicated Nesterov (the on-node data plane,
    # VERDICT r1 item 1); absent = parameter-server star via push streams
    # (the WAN path, parameter_server.rs parity).
    rccl = cfg.get("rccl")
    # The group is always formed via comm.reform() inside run_rccl, AFTER the
    # model/flat-buffer build: initial members and mid-run joiners then follow
    # the exact same collective sequence (reform -> state broadcast), which a
    # constructor-time init broadcast would desynchronize.
    comm = Comm(timeout_s=float(rccl.get("timeout_s", 600.0)) if rccl else 600.0)
    worker = DiLoCoWorker(
        model, DiLoCoConfig(h=1 << 30, inner=inner), comm=comm, device=device
    )
    batch_size = int(cfg.get("batch_size", 4))
    seq_len = int(cfg.get("seq_len", 128))
    data_ref = cfg["data"]
    updates_ref = cfg.get("updates")  # absent in rccl sync mode (no PS star)
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

    # background SSE listener: aggregated-update file pointers go to
    # updates_q; communicator re-formation orders (scheduler-driven elastic
    # membership) go to reform_q
    updates_q: "queue.Queue[dict]" = queue.Queue()
    reform_q: "queue.Queue[dict]" = queue.Queue()

    def listen():
        try:
            for ev in session.receive():
                if ev.get("kind") == "reform":
                    reform_q.put(ev)
                else:
                    updates_q.put(ev)
        except Exception as e:  # bridge gone = worker daemon died: abort
            # instead of training on as an orphan (grad-release of the whole
            # job happens via the daemon's lease machinery, not us)
            print(f"# sse listener ended: {type(e).__name__}: {e}", file=sys.stderr)
        updates_q.put({"__bridge_closed__": True})
        reform_q.put({"__bridge_closed__": True})

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

    # optional preprocessor (TrainExecutorConfig.preprocessor,
    # messages lib.rs:483-489): fetch its artifact through the connector,
    # resolve the PreprocessorType, and run raw slice columns through it
    preprocessor = None
    pre_inputs: list = []
    pre_cfg = cfg.get("preprocessor")
    if pre_cfg:
        from hypha_amd.data.stream import build_preprocessor

        got = session.fetch(pre_cfg["artifact"])
        art = got["files"][0] if got.get("files") else None
        if art and os.path.isfile(art):
            art = os.path.dirname(art)  # processors load from a directory
        preprocessor = build_preprocessor(pre_cfg.get("task", "tokenizer"), art)
        pre_inputs = list(pre_cfg.get("input_names", []))
        pre_out_key = pre_cfg.get("output_key", "input_ids")
        print(f"[executor] preprocessor={pre_cfg.get('task')} artifact={art}",
              flush=True)

    # infinite batch stream over scheduler-assigned slices (utils.py fetch_data)
    def batches():
        while True:
            got = session.fetch(data_ref)
            for path in got["files"]:
                if preprocessor is not None:
                    # reference dataset.py:26-30: pop the processor inputs
                    # from the slice tensors, run them through, train on
                    # the processed column
                    from safetensors.torch import load_file as _lf

                    data = _lf(path)
                    fed = {k: data.pop(k) for k in pre_inputs if k in data}
                    out = preprocessor(**fed)
                    ids = out[pre_out_key].long()
                else:
                    ids = load_slice(path)
                for i in range(0, ids.shape[0] - batch_size + 1, batch_size):
                    b = ids[i : i + batch_size, :seq_len]
                    yield b, b.clone()

    batch_iter = batches()
    print(f"[executor] model={cfg['model']} device={device} bs={batch_size}", flush=True)

    if rccl:
        return run_rccl(worker, comm, session, batch_iter, reform_q, rccl,
                        batch_size, seq_len, resume_round, ckpt_dir, ckpt_every,
                        args.work_dir)

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
    _ps_job_span = _tr.start_span("job.execute",
                                  job_id=os.environ.get("HYPHA_JOB_ID", ""))
    while not done:
        # ---- inner loop: train until the scheduler's counter is exhausted ----
        import time as _time

        _round_span = _tr.start_span("diloco.round", parent=_ps_job_span,
                                     round=round_idx)
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

        _round_span.set_attribute("samples", round_samples)
        _round_span.set_attribute("loss", float(loss))
        _round_span.end()

        # ---- extract and push the pseudo-gradient ----
        _sync_span = _tr.start_span("diloco.outer_sync", parent=_ps_job_span,
                                    round=round_idx)
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
        _sync_span.end()

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

    _ps_job_span.end()
    _tr.flush()
    session.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
