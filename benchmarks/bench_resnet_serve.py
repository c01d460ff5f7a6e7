#!/usr/bin/env python3
"""ResNet-18 inference latency: eager bf16 forward vs bucketed hipGraph
replay (ModuleGraphRunner). Shows what graph capture buys on a
launch-bound model (~60 kernels per forward).

    python benchmarks/bench_resnet_serve.py --iters 300
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def pct(xs, q):
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(q * len(xs)))]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=300)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--image-size", type=int, default=224)
    args = p.parse_args()

    from unionml_amd.models.resnet import ResNet18
    from unionml_amd.serving.graph_runner import ModuleGraphRunner

    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda" if use_gpu else "cpu")
    if use_gpu:
        torch.backends.cudnn.benchmark = True

    torch.manual_seed(0)
    net = ResNet18(num_classes=1000).to(device).eval()
    if use_gpu:
        net = net.to(memory_format=torch.channels_last)

    x_np = np.random.RandomState(1).rand(
        args.batch, 3, args.image_size, args.image_size
    ).astype(np.float32)
    xb = torch.from_numpy(x_np).to(device)
    if use_gpu:
        xb = xb.to(memory_format=torch.channels_last)

    def eager():
        with torch.no_grad(), torch.autocast(device.type, dtype=torch.bfloat16, enabled=use_gpu):
            out = net(xb).argmax(dim=1)
        return out.cpu()

    for _ in range(args.warmup):
        eager()
    if use_gpu:
        torch.cuda.synchronize()
    lat_e = []
    for _ in range(args.iters):
        t0 = time.perf_counter()
        eager()
        lat_e.append((time.perf_counter() - t0) * 1000.0)

    runner = ModuleGraphRunner(
        net, max_batch_size=max(args.batch, 1), channels_last=True, postprocess="argmax"
    )
    for _ in range(args.warmup):
        runner(x_np)
    lat_g = []
    for _ in range(args.iters):
        t0 = time.perf_counter()
        runner(x_np)
        lat_g.append((time.perf_counter() - t0) * 1000.0)

    print(
        json.dumps(
            {
                "metric": "resnet18_infer_p50_latency_ms",
                "value": pct(lat_g, 0.50),
                "unit": "ms",
                "n_gpus": 1 if use_gpu else 0,
                "steps": args.iters,
                "warmup": args.warmup,
                "higher_is_better": False,
                "vs_baseline": None,
                "dtype": "bf16" if use_gpu else "fp32",
                "data": "synthetic",
                "config": {
                    "model": "resnet18",
                    "batch": args.batch,
                    "image_size": args.image_size,
                    "hipgraph": {"p50_ms": pct(lat_g, 0.50), "p99_ms": pct(lat_g, 0.99)},
                    "eager": {"p50_ms": pct(lat_e, 0.50), "p99_ms": pct(lat_e, 0.99)},
                },
            }
        )
    )


if __name__ == "__main__":
    main()
