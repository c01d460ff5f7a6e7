#!/usr/bin/env python3
"""Stock-PyTorch baseline for the tabular hot path: the same
Linear->ReLU->Linear + cross-entropy + Adam training step written the
way a user would in eager torch (bf16 autocast, fused adam when
available), measured at the same geometry/batch as the HIP kernels.

  python benchmarks/bench_torch_baseline.py --shape 784x128x10 --batch 512
"""

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--shape", default="784x128x10")
    p.add_argument("--batch", type=int, default=512)
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--compile", action="store_true", help="torch.compile the model")
    args = p.parse_args()

    inf, hid, cls = (int(x) for x in args.shape.split("x"))
    dev = "cuda:0"
    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(inf, hid), torch.nn.ReLU(), torch.nn.Linear(hid, cls)
    ).to(dev)
    if args.compile:
        model = torch.compile(model)
    try:
        opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=True)
        fused = True
    except (RuntimeError, TypeError):
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        fused = False

    B = args.batch
    X = torch.randn(B * 8, inf, device=dev)
    y = torch.randint(0, cls, (B * 8,), device=dev)

    def step(i):
        off = (i % 8) * B
        xb, yb = X[off : off + B], y[off : off + B]
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(xb), yb)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        return loss

    for i in range(args.warmup):
        step(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = step(i)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / args.steps * 1e6

    # hipGraph-captured variant of the same eager step (best case for
    # stock torch: zero launch overhead). NOTE: full-autograd capture
    # SEGFAULTS on torch 2.10 + ROCm 7.2 (also documented in
    # docs/perf_engineering.md r01); gate it behind --graph.
    graph_us = None
    try:
        if "--graph" not in sys.argv:
            raise RuntimeError("skipped (segfaults on this stack; pass --graph)")
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            for i in range(8):
                step(i)
        for _ in range(3):
            g.replay()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = max(1, args.steps // 8)
        for _ in range(reps):
            g.replay()
        torch.cuda.synchronize()
        graph_us = (time.perf_counter() - t0) / (reps * 8) * 1e6
    except RuntimeError as exc:
        print(f"graph capture unavailable: {exc}", file=sys.stderr)

    print(json.dumps({
        "baseline": "stock torch eager" + (" + torch.compile" if args.compile else ""),
        "shape": args.shape, "batch": B, "fused_adam": fused,
        "eager_us_per_step": round(us, 2),
        "eager_samples_per_sec": round(B / (us / 1e6)),
        "graph_us_per_step": round(graph_us, 2) if graph_us else None,
        "graph_samples_per_sec": round(B / (graph_us / 1e6)) if graph_us else None,
    }))


if __name__ == "__main__":
    main()
