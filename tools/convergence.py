"""Convergence evidence: the fused-kernel DiLoCo engine vs plain torch
AdamW on the SAME model/data (default: the GPT-2-small family on a
synthetic corpus).
Shows the whole native stack (HIP kernels + flat-param AdamW + outer
Nesterov) follows the reference-optimizer trajectory, not just per-kernel
oracles. Prints loss every `--log-every` steps for both runs."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="gpt2-small")
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--h", type=int, default=10)
    p.add_argument("--lr", type=float, default=3e-4)
    p.add_argument("--log-every", type=int, default=20)
    args = p.parse_args()

    from hypha_amd import models
    from hypha_amd.data.synthetic import SyntheticTokens
    from hypha_amd.parallel import Comm, DiLoCoConfig, DiLoCoWorker, InnerOptConfig

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"

    def data_iter():
        d = SyntheticTokens(512, args.seq, args.batch, seed=123)
        while True:
            yield d.next_batch()

    # ---- ours: fused-kernel DiLoCo (H-step inner AdamW + outer Nesterov) ----
    torch.manual_seed(7)
    model = models.build(args.model)
    w = DiLoCoWorker(
        model,
        DiLoCoConfig(h=args.h, inner=InnerOptConfig(
            lr=args.lr, warmup_steps=10, schedule="constant")),
        comm=Comm(), device=torch.device(dev))
    ours = []
    it = data_iter()
    for step in range(args.steps):
        ids, labels = next(it)
        loss = w.train_step(ids, labels)
        w.maybe_outer_sync()
        if step % args.log_every == 0 or step == args.steps - 1:
            ours.append((step, loss))

    # ---- reference: same model/init/data, torch.optim.AdamW every step ----
    torch.manual_seed(7)
    ref_model = models.build(args.model).to(dev)
    if dev != "cpu":
        ref_model = ref_model.bfloat16()
        for buf in ref_model.buffers():
            if buf.dtype is torch.bfloat16:
                buf.data = buf.data.float()
    opt = torch.optim.AdamW(ref_model.parameters(), lr=args.lr,
                            betas=(0.9, 0.95), eps=1e-8, weight_decay=0.1)
    sched = torch.optim.lr_scheduler.LambdaLR(
        opt, lambda s: min(1.0, (s + 1) / 10))
    ref = []
    it = data_iter()
    for step in range(args.steps):
        ids, labels = next(it)
        ids, labels = ids.to(dev), labels.to(dev)
        loss = ref_model(ids, labels=labels)
        loss.backward()
        torch.nn.utils.clip_grad_norm_(ref_model.parameters(), 1.0)
        opt.step()
        sched.step()
        opt.zero_grad(set_to_none=True)
        if step % args.log_every == 0 or step == args.steps - 1:
            ref.append((step, float(loss.detach().float())))

    print(f"# {args.model} b{args.batch} s{args.seq} lr{args.lr} "
          f"H{args.h} on {dev}")
    print("step, diloco_fused_loss, torch_adamw_loss")
    for (s1, l1), (s2, l2) in zip(ours, ref):
        print(f"{s1:5d}, {l1:.4f}, {l2:.4f}")


if __name__ == "__main__":
    main()
