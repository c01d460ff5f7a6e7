#!/usr/bin/env python3
"""Flagship benchmark: digits-MLP training throughput (train samples/sec).

BASELINE.json metric: "train samples/sec + /predict p50 latency, MLP
digits-shape, 1/2/4/8 MI355X". This measures the training leg on
synthetic digits-shaped data (64 features, 10 classes, random-init
weights — no network for datasets) with the CDNA4 fused hot path:
one optimizer step = the fully-fused fwd/bwd/reduce/Adam MFMA kernel
(or, under DP, fused fwd/bwd kernel + RCCL gradient all-reduce + fused
Adam), with 64 optimizer steps captured per hipGraph so replay launch
overhead is amortized (--graph-steps).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W              # single GPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: per-GPU batch is fixed (default 2048); reported value is
the WHOLE-JOB samples/sec aggregated over all ranks.
"""

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=200)
    p.add_argument("--batch", type=int, default=2048, help="per-GPU batch size")
    p.add_argument("--minibatches", type=int, default=64, help="distinct minibatches cycled")
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--no-graph", action="store_true")
    p.add_argument(
        "--graph-steps",
        type=int,
        default=64,
        help="optimizer steps captured per hipGraph (amortizes replay launch; "
        "64 = one graph per full minibatch cycle, measured +2%% over 16)",
    )
    p.add_argument(
        "--engine",
        choices=["auto", "fused", "persistent", "stepwise"],
        default="auto",
        help="auto: fused single-kernel step (1 GPU) / 3-kernel+RCCL (DP)",
    )
    return p.parse_args()


def _result(args, n_gpus, B, elapsed, engine, loss):
    return {
        "metric": "train_samples_per_sec",
        "value": args.steps * B * n_gpus / elapsed,
        "unit": "samples/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": "digits_mlp_64x32x10",
            "global_batch": B * n_gpus,
            "seq_len": None,
            "parallelism": f"dp{n_gpus}",
            "engine": engine,
            "final_loss": loss,
        },
    }


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(world, 1)

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        device = "cpu"

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if use_gpu:
            # RCCL collectives inside hipGraph capture require the async
            # error watchdog off (else capture aborts / the watchdog can
            # fire mid-capture); harmless for a short bench
            os.environ.setdefault("NCCL_ASYNC_ERROR_HANDLING", "0")
            os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "0")
        try:
            # bound-device init: eager communicator creation, cleaner
            # rendezvous on one-process-per-GPU launches
            dist.init_process_group(
                "nccl" if use_gpu else "gloo",
                device_id=torch.device("cuda", local_rank) if use_gpu else None,
            )
        except TypeError:  # older torch without device_id
            dist.init_process_group("nccl" if use_gpu else "gloo")

    from unionml_amd.ops.tabular import ADAM_BETA1, ADAM_BETA2, ADAM_EPS, TabularMLP
    from unionml_amd.ops.reference import NPARAM

    B, M = args.batch, args.minibatches
    torch.manual_seed(1234 + rank)  # per-rank DATA; weights identical (seed=0)
    clf = TabularMLP(device=device, seed=0)

    # synthetic digits-shaped data, staged bf16-resident in HBM
    X = torch.rand(B * M, 64, device=clf.device) * 16.0
    y = torch.randint(0, 10, (B * M,), dtype=torch.int32, device=clf.device)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    invBtot = 1.0 / (B * world)

    engine = args.engine
    if engine == "auto":
        engine = "fused" if (use_gpu and world == 1) else "stepwise"

    if use_gpu:
        from unionml_amd.ops import hip_ext
        from unionml_amd.ops.reference import NPARAM as _NP

        ext = hip_ext(required=True)
        loss_out = clf.grads[_NP : _NP + 1]

        if engine == "fused":
            clf._ensure_slabs((B + 127) // 128)

            def eager_step(off):
                ok = ext.mlp_step_fused(
                    Xbf[off : off + B], y[off : off + B], clf.W1bf, clf.W2bf,
                    clf.master, clf.bfmirror, clf.m, clf.v, clf.t_dev,
                    clf.slabs, clf.counter, loss_out, invBtot,
                    args.lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS, wimg=clf.wimg,
                )
                assert ok

        else:
            # DP path: fused fwd/bwd + in-kernel slab reduction writing the
            # summed grads (no global atomics, no zeroing kernel), then the
            # RCCL all-reduce, then fused Adam
            clf._ensure_slabs((B + 127) // 128)

            def eager_step(off):
                ok = ext.mlp_step_fused(
                    Xbf[off : off + B], y[off : off + B], clf.W1bf, clf.W2bf,
                    clf.master, clf.bfmirror, clf.m, clf.v, clf.t_dev,
                    clf.slabs, clf.counter, loss_out, invBtot,
                    args.lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                    grads_out=clf.grads, wimg=clf.wimg,
                )
                assert ok
                if dist is not None:
                    dist.all_reduce(clf.grads)
                ext.adam_step(clf.master, clf.bfmirror, clf.grads, clf.m, clf.v,
                              clf.t_dev, args.lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                              wimg=clf.wimg)

    else:

        def eager_step(off):
            clf._step(Xbf[off : off + B], y[off : off + B], invBtot, args.lr,
                      dist is not None)

    # persistent single-workgroup engine: K steps in ONE kernel launch
    if use_gpu and world == 1 and engine == "persistent":
        def run_steps(k):
            ok = ext.mlp_train_steps(
                Xbf, y, B, k, clf.master, clf.bfmirror, clf.m, clf.v,
                clf.t_dev, loss_out, args.lr, 0.9, 0.999, 1e-8,
            )
            assert ok, "mlp_train_steps constraints unmet"

        run_steps(args.warmup)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        run_steps(args.steps)
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        loss = float(loss_out.item())
        assert loss == loss and loss < 1e6, f"training diverged: loss={loss}"
        print(json.dumps(_result(args, 1, B, elapsed, "persistent_steps_kernel", loss)))
        return

    # warm up communicator + kernels, then capture one hipGraph per minibatch
    for i in range(3):
        eager_step((i % M) * B)

    # capture G optimizer steps per graph (G | M), so one replay advances
    # G steps — all work identical, launch overhead amortized G-fold
    G = max(1, args.graph_steps)
    while M % G:
        G -= 1
    graphs = None
    if use_gpu and not args.no_graph:
        try:
            graphs = []
            for chunk in range(M // G):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    for j in range(G):
                        eager_step((chunk * G + j) * B)
                graphs.append(g)
        except RuntimeError as exc:
            print(f"[bench] graph capture unavailable ({exc}); eager stepping",
                  file=sys.stderr)
            graphs = None

    def run_steps(k0: int, nsteps: int):
        """Advance exactly ``nsteps`` optimizer steps from global step k0:
        G-aligned stretches replay whole-chunk graphs, edges run eagerly."""
        k = k0
        end = k0 + nsteps
        while k < end:
            if graphs is not None and k % G == 0 and k + G <= end:
                graphs[(k // G) % (M // G)].replay()
                k += G
            else:
                eager_step((k % M) * B)
                k += 1

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    run_steps(0, args.warmup)
    barrier_sync()
    t0 = time.perf_counter()
    run_steps(args.warmup, args.steps)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines the job)
    if dist is not None:
        t = torch.tensor([elapsed], device=clf.device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    loss = float(clf.grads[NPARAM].item())
    assert loss == loss and loss < 1e6, f"training diverged: loss={loss}"

    if rank == 0:
        engine_name = f"{engine}+{f'hipgraph{G}' if graphs is not None else 'eager'}"
        print(json.dumps(_result(args, n_gpus, B, elapsed, engine_name, loss)))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
