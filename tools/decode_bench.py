"""Decode (serving) throughput microbench: prefill once, then timed
single-token KV-cache decode steps. Prints decode tokens/s."""

import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

# pre-tuned hipBLASLt algos (shared table with bench.py; includes the
# skinny decode shapes)
_TUNED_BASE = str(Path(__file__).resolve().parent.parent /
                  "tunableop_gfx950_llama8b.csv")
if (os.path.exists(_TUNED_BASE.replace(".csv", "0.csv"))
        and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ):
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = _TUNED_BASE

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--prefill", type=int, default=512)
    p.add_argument("--decode", type=int, default=128)
    p.add_argument("--warmup", type=int, default=16)
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-captured decode step (greedy)")
    p.add_argument("--kv-fp8", action="store_true",
                   help="fp8 (e4m3 + per-row scales) KV cache")
    args = p.parse_args()

    from hypha_amd import models

    dev = "cuda:0"
    torch.manual_seed(0)
    with torch.device(dev):
        torch.set_default_dtype(torch.bfloat16)
        model = models.build(args.model)
        torch.set_default_dtype(torch.float32)
    model = model.to(dev).bfloat16().eval()
    for buf in model.buffers():  # rope tables must stay fp32 for the kernels
        if buf.dtype in (torch.bfloat16, torch.float16):
            buf.data = buf.data.float()
    ids = torch.randint(0, model.cfg.vocab_size - 1, (args.batch, args.prefill),
                        device=dev)

    kvq = "fp8" if args.kv_fp8 else None
    if args.graph:
        from hypha_amd.runtime.graphed_decode import GraphedDecoder

        dec = GraphedDecoder(model, args.batch, args.prefill,
                             args.decode + args.warmup, kv_quant=kvq)
        gen = dec.generate
    else:
        import functools

        gen = functools.partial(model.generate, kv_quant=kvq)

    # warm (weights/algos/graph capture), then difference two WARM runs so
    # one-time costs never land in the per-token figure
    gen(ids, max_new_tokens=args.warmup)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    gen(ids, max_new_tokens=args.warmup)
    torch.cuda.synchronize()
    t_short = time.perf_counter() - t0
    t0 = time.perf_counter()
    out = gen(ids, max_new_tokens=args.decode + args.warmup)
    torch.cuda.synchronize()
    t_long = time.perf_counter() - t0
    dt = t_long - t_short
    toks = args.batch * args.decode
    print(f"{args.model} b{args.batch} prefill{args.prefill}: "
          f"{toks / dt:.0f} decode tok/s  ({dt / args.decode * 1e3:.2f} ms/step)"
          f"  [prefill+{args.warmup}: {t_short:.2f}s]")
    assert out.shape[1] == args.prefill + args.decode + args.warmup


if __name__ == "__main__":
    main()
