#!/usr/bin/env python3
"""ResNet-18 DP training benchmark.

Synthetic 224×224 images, random-init weights, bf16 autocast,
channels-last, Adam; under WORLD_SIZE>1 the framework's bucketed
RCCL gradient all-reducer (unionml_amd/parallel/ddp.py) overlaps
reduction with backward. Weak scaling: per-GPU batch fixed.

Launch:
    python benchmarks/bench_resnet.py --steps 100 --warmup 20
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 benchmarks/bench_resnet.py --steps ...
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--batch", type=int, default=256, help="per-GPU batch size")
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--classes", type=int, default=1000)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--bucket-mb", type=float, default=16.0, help="gradient bucket size; ResNet-18 grads are ~47MB fp32, so 16MB gives ~3 buckets overlapping backward")
    p.add_argument("--minibatches", type=int, default=4)
    p.add_argument(
        "--graph",
        action="store_true",
        help="EXPERIMENTAL: capture the whole train step into hipGraphs "
        "(full-autograd capture segfaults on this stack — default off)",
    )
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)
        # let MIOpen autotune each conv shape during warmup — without
        # this the default find path picks a bwd-weight kernel that is
        # ~50x slower on the 7x7 stem (measured: profiles/)
        torch.backends.cudnn.benchmark = True

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl" if use_gpu else "gloo")

    from unionml_amd.models.resnet import ResNet18
    from unionml_amd.parallel.ddp import GradientAllReducer, get_world_size

    torch.manual_seed(1234)  # identical weights on every rank
    net = ResNet18(num_classes=args.classes).to(device)
    if use_gpu:
        net = net.to(memory_format=torch.channels_last)
    net.train()

    reducer = (
        GradientAllReducer(net, bucket_mb=args.bucket_mb, sync_params=False)
        if world > 1
        else None
    )
    want_graphs = use_gpu and args.graph
    # capturable must be set BEFORE the first opt.step so Adam keeps its
    # step counters on-device (required for hipGraph capture)
    opt = torch.optim.Adam(
        net.parameters(), lr=args.lr, foreach=True, capturable=want_graphs
    )

    B, M = args.batch, args.minibatches
    gen = torch.Generator(device="cpu").manual_seed(100 + rank)
    X = [
        torch.rand(B, 3, args.image_size, args.image_size, generator=gen).to(device)
        for _ in range(M)
    ]
    if use_gpu:
        X = [x.to(memory_format=torch.channels_last) for x in X]
    y = torch.randint(0, args.classes, (M, B), generator=gen).to(device)

    amp_dtype = torch.bfloat16

    def step(k):
        mb = k % M
        with torch.autocast(device.type, dtype=amp_dtype, enabled=use_gpu):
            loss = F.cross_entropy(net(X[mb]), y[mb])
        loss.backward()
        if reducer is not None:
            reducer.finalize()
        opt.step()
        if reducer is not None:
            reducer.zero_grad()
        else:
            opt.zero_grad(set_to_none=True)
        return loss

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    for k in range(args.warmup):
        loss = step(k)
    barrier_sync()

    # capture the whole train step (fwd+bwd+reduce+Adam) per minibatch
    # into hipGraphs — a ResNet-18 step is a few hundred kernel launches
    engine = "eager"
    graphs = None
    if want_graphs:
        try:
            graphs = []
            for mb in range(M):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    step(mb)
                graphs.append(g)
            engine = "hipgraph"
        except RuntimeError as exc:
            print(f"[bench_resnet] graph capture unavailable ({exc}); eager",
                  file=sys.stderr)
            graphs = None
    barrier_sync()

    t0 = time.perf_counter()
    for k in range(args.steps):
        if graphs is not None:
            graphs[k % M].replay()
        else:
            loss = step(k)
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if graphs is not None:
        loss = step(args.steps % M)  # fresh eager step for a readable loss
        barrier_sync()

    if dist is not None:
        t = torch.tensor([elapsed], device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    final_loss = float(loss.detach().float().cpu())
    assert final_loss == final_loss, "training diverged (NaN loss)"

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "train_samples_per_sec",
                    "value": args.steps * B * max(world, 1) / elapsed,
                    "unit": "samples/s",
                    "n_gpus": max(world, 1),
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if use_gpu else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": "resnet18",
                        "global_batch": B * max(world, 1),
                        "seq_len": None,
                        "parallelism": f"dp{max(world, 1)}",
                        "image_size": args.image_size,
                        "bucket_mb": args.bucket_mb,
                        "engine": engine,
                        "final_loss": final_loss,
                    },
                }
            )
        )

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
