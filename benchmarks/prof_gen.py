#!/usr/bin/env python3
"""Profile driver for the GENERALIZED tabular kernels: runs K fused
training steps + predict at a chosen geometry so `rocprofv3 --stats`
shows the per-kernel device-time split. Also prints eager step timing.

  python benchmarks/prof_gen.py --shape 784x128x10 --steps 400
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
    p.add_argument("--graph", action="store_true", help="time graph-amortized epochs too")
    args = p.parse_args()

    inf, hid, cls = (int(x) for x in args.shape.split("x"))
    from unionml_amd.ops.tabular import TabularMLP

    clf = TabularMLP(in_features=inf, hidden=hid, classes=cls, device="cuda:0", seed=0)
    g = clf.g
    B = args.batch
    torch.manual_seed(0)
    X = torch.rand(B * 8, inf, device=clf.device) * 16.0
    y = torch.randint(0, cls, (B * 8,), dtype=torch.int32, device=clf.device)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)

    rpw = clf._rows_per_wg()
    clf._ensure_slabs((B + rpw - 1) // rpw)
    loss_out = clf.grads[g.nparam : g.nparam + 1]

    def step(i):
        off = (i % 8) * B
        clf._fused_adam_step(Xbf[off : off + B], y[off : off + B], 1.0 / B, 1e-3, loss_out)

    for i in range(args.warmup):
        step(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    torch.cuda.synchronize()
    eager_us = (time.perf_counter() - t0) / args.steps * 1e6

    out = {
        "shape": args.shape,
        "padded": f"{g.inp}x{g.hid}x{g.cpad}",
        "batch": B,
        "eager_us_per_step": round(eager_us, 2),
        "samples_per_sec": round(B / (eager_us / 1e6)),
        "loss": float(loss_out.item()),
    }

    if args.graph:
        gph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gph):
            for i in range(8):
                step(i)
        for _ in range(5):
            gph.replay()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = max(1, args.steps // 8)
        for _ in range(reps):
            gph.replay()
        torch.cuda.synchronize()
        graph_us = (time.perf_counter() - t0) / (reps * 8) * 1e6
        out["graph_us_per_step"] = round(graph_us, 2)
        out["graph_samples_per_sec"] = round(B / (graph_us / 1e6))

    # predict leg
    Xq = X[:B]
    for _ in range(10):
        clf.predict(Xq)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100):
        clf.predict(Xq)
    torch.cuda.synchronize()
    out["predict_us_per_batch"] = round((time.perf_counter() - t0) / 100 * 1e6, 2)

    print(json.dumps(out))


if __name__ == "__main__":
    main()
