#!/usr/bin/env python3
"""Randomized-geometry numerics fuzz for the generalized tabular
kernels: draws random (in_features, hidden, classes, batch) tuples,
runs one reduce-only step + predict on the GPU, and compares against
the parametric torch oracle. Exits non-zero on any mismatch.

  python benchmarks/fuzz_gen.py --trials 25 --seed 0
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--trials", type=int, default=25)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args()

    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    rng = torch.Generator().manual_seed(args.seed)

    def rand(lo, hi):
        return int(torch.randint(lo, hi + 1, (1,), generator=rng).item())

    failures = 0
    for trial in range(args.trials):
        inf = rand(2, 900)
        hid = rand(2, 256)
        cls = rand(2, 32)
        B = rand(1, 700)
        clf = TabularMLP(in_features=inf, hidden=hid, classes=cls,
                         device="cuda:0", seed=trial)
        g = clf.g
        X = torch.randn(max(B, 2), inf, generator=rng) * 1.7 + 0.2
        y = torch.randint(0, cls, (max(B, 2),), generator=rng, dtype=torch.int32)
        clf.fit_standardizer(X)
        Xbf = clf.stage(X)[:B]
        yd = y[:B].to("cuda:0")

        clf._step_reduce(Xbf, yd, 1.0 / B, 1e-3)
        torch.cuda.synchronize()
        grads = clf.grads.cpu()
        grads_ref = torch.zeros(g.nparam + 1)
        ref.mlp_step_g(g, Xbf.cpu(), y[:B], clf.W1bf.cpu(), clf.W2bf.cpu(),
                       clf.master.cpu(), grads_ref, 1.0 / B)
        scale = float(grads_ref[: g.nparam].abs().max())
        err = float((grads[: g.nparam] - grads_ref[: g.nparam]).abs().max())
        loss_err = abs(float(grads[g.nparam] - grads_ref[g.nparam]))
        grad_ok = err < max(2e-2 * scale, 3e-4) and loss_err < 3e-3

        preds, probs = clf.predict(X[:B].to("cuda:0"), return_probs=True)
        torch.cuda.synchronize()
        preds_ref, probs_ref = ref.mlp_predict_g(
            g, X[:B], clf.mean.cpu(), clf.invstd.cpu(), clf.W1bf.cpu(),
            clf.W2bf.cpu(), clf.master.cpu(), return_probs=True)
        margin = probs_ref.topk(2, dim=1).values if cls > 1 else None
        clear = (margin[:, 0] - margin[:, 1]) > 1e-4
        pred_ok = bool((preds.cpu() == preds_ref)[clear].all()) and torch.allclose(
            probs.cpu(), probs_ref, rtol=2e-2, atol=2e-3)

        status = "ok" if (grad_ok and pred_ok) else "FAIL"
        print(f"[{trial:02d}] {inf}x{hid}x{cls} B={B} -> {g.inp}x{g.hid}x{g.cpad} "
              f"grad_err={err:.2e}/{scale:.2e} loss_err={loss_err:.2e} {status}")
        if status == "FAIL":
            failures += 1

    print(f"fuzz: {args.trials - failures}/{args.trials} geometries passed")
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
