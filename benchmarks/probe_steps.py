"""Standalone GPU probe: where does the persistent steps kernel spend time?

Run under rocprofv3 or bare:
    python tests/probe_steps.py [--pmc-mode]
"""

import argparse
import time

import torch

from unionml_amd.ops import hip_ext
from unionml_amd.ops.reference import NPARAM
from unionml_amd.ops.tabular import TabularMLP


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pmc-mode", action="store_true", help="one fixed config only")
    args = ap.parse_args()

    ext = hip_ext(required=True)
    dev = "cuda:0"
    torch.manual_seed(0)

    def steps_time(B, n_steps, M=64):
        clf = TabularMLP(device=dev, seed=0)
        N = B * M
        Xbf = (torch.randn(N, 64, device=dev)).bfloat16()
        y = torch.randint(0, 10, (N,), dtype=torch.int32, device=dev)
        loss_out = clf.grads[NPARAM : NPARAM + 1]

        def run(k):
            ok = ext.mlp_train_steps(Xbf, y, B, k, clf.master, clf.bfmirror,
                                     clf.m, clf.v, clf.t_dev, loss_out,
                                     1e-3, 0.9, 0.999, 1e-8)
            assert ok

        run(50)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        run(n_steps)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n_steps * 1e6  # us/step

    def step_eager_time(B, iters=300):
        clf = TabularMLP(device=dev, seed=0)
        Xbf = torch.randn(B, 64, device=dev).bfloat16()
        y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)
        for _ in range(20):
            ext.mlp_step(Xbf, y, clf.W1bf, clf.W2bf, clf.master, clf.grads, 1.0 / B)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            ext.mlp_step(Xbf, y, clf.W1bf, clf.W2bf, clf.master, clf.grads, 1.0 / B)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    if args.pmc_mode:
        print("steps(B=512, 500):", steps_time(512, 500), "us/step")
        return

    for B in (128, 256, 512, 1024):
        print(f"steps kernel  B={B:5d}: {steps_time(B, 1000):8.2f} us/step")
    for B in (128, 512, 2048, 8192):
        print(f"step eager    B={B:5d}: {step_eager_time(B):8.2f} us/launch")


if __name__ == "__main__":
    main()
