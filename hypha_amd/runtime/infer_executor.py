"""Inference executor: serves generation jobs through the same job-bridge
contract as training ({SOCKET_PATH}/{WORK_DIR}/{JOB_JSON}).

The reference positions itself as orchestration "for AI training AND
inference" with inference delegated to the same executor mechanism; this
executor is the native counterpart: it fetches prompt slices via the bridge
(scheduler-tracked data slices), runs KV-cache generation on the model
registry, writes completions as SafeTensors to the work dir, pushes them to
the configured result peers, and reports per-batch Status/Metrics progress
so the scheduler's FSM and lease machinery apply unchanged.

Job config: {"model", "data": <fetch ref>, "results": <send ref>?,
             "max_new_tokens", "batch_size", "seq_len", "temperature",
             "top_k", "num_batches", "kv_cache": "fp8"?}

Greedy Llama-family jobs on GPU decode through the hipGraph-captured step
(runtime/graphed_decode.py); "kv_cache": "fp8" halves KV memory.
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--socket", required=True)
    p.add_argument("--work-dir", required=True)
    p.add_argument("--job", required=True)
    args = p.parse_args()

    from safetensors.torch import save_file

    from hypha_amd import models
    from hypha_amd.data.synthetic import load_slice
    from hypha_amd.runtime.session import Session

    with open(args.job) as f:
        cfg = json.load(f)

    session = Session(args.socket)
    from hypha_amd.telemetry import get_tracer

    tracer = get_tracer()
    disp_ns = os.environ.get("HYPHA_DISPATCH_TS_NS")
    if disp_ns:
        d = tracer.start_span("job.dispatch",
                              job_id=os.environ.get("HYPHA_JOB_ID", ""))
        d.start_ns = int(disp_ns)
        d.end()
    job_span = tracer.start_span("job.execute", kind_attr="generate",
                                 job_id=os.environ.get("HYPHA_JOB_ID", ""))
    torch.manual_seed(int(cfg.get("seed", 0)))
    device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
    model = models.build(cfg["model"])
    model.to(device=device,
             dtype=torch.bfloat16 if device.type == "cuda" else torch.float32)
    for buf in model.buffers():
        if buf.dtype in (torch.bfloat16, torch.float16):
            buf.data = buf.data.float()
    model.eval()

    batch_size = int(cfg.get("batch_size", 1))
    seq_len = int(cfg.get("seq_len", 64))
    max_new = int(cfg.get("max_new_tokens", 32))
    num_batches = int(cfg.get("num_batches", 1))
    temperature = float(cfg.get("temperature", 0.0))
    top_k = int(cfg.get("top_k", 0))

    # greedy Llama-family serving on GPU: hipGraph-captured decode step
    # (runtime/graphed_decode.py; eager decode is launch-bound)
    graphed = None
    if (temperature <= 0 and device.type == "cuda"
            and type(model).__name__ == "LlamaForCausalLM"
            and model.cfg.head_dim in (64, 128)
            and model.cfg.n_heads // model.cfg.n_kv_heads <= 8):
        from hypha_amd.runtime.graphed_decode import GraphedDecoder

        graphed = GraphedDecoder(model, batch_size, seq_len, max_new,
                                 kv_quant=cfg.get("kv_cache"))

    done_batches = 0
    out_idx = 0
    while done_batches < num_batches:
        got = session.fetch(cfg["data"])
        for path in got["files"]:
            ids = load_slice(path)
            for i in range(0, ids.shape[0] - batch_size + 1, batch_size):
                if done_batches >= num_batches:
                    break
                prompts = ids[i : i + batch_size, :seq_len].to(device)
                if graphed is not None and prompts.shape[0] == batch_size:
                    out = graphed.generate(prompts, max_new_tokens=max_new)
                else:
                    out = model.generate(prompts, max_new_tokens=max_new,
                                         temperature=temperature, top_k=top_k,
                                         seed=0)
                fname = f"completion-{out_idx:05d}.safetensors"
                save_file({"tokens": out.cpu().contiguous()},
                          os.path.join(args.work_dir, fname))
                out_idx += 1
                if cfg.get("results"):
                    session.send_resource(cfg["results"], fname)
                done_batches += 1
                session.send_status({"kind": "status", "batch_size": batch_size})
            if done_batches >= num_batches:
                break
    session.send_status(
        {"kind": "metrics", "round": 0,
         "metrics": {"completions": float(out_idx)}}
    )
    print(f"[infer] wrote {out_idx} completion batches", flush=True)
    job_span.set_attribute("completions", out_idx)
    job_span.end()
    tracer.flush()
    session.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
