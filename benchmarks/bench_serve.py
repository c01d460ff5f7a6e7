#!/usr/bin/env python3
"""Serving benchmark: /predict p50 latency for the digits MLP app
(BASELINE.md config 3).

Trains the flagship digits MLP (synthetic data, random-init weights),
serves it through the real FastAPI app (dynamic batcher + bucketed
hipGraph inference on GPU), and measures:

- sequential single-row POST /predict latency over real HTTP
  (p50/p90/p99) — the headline "/predict p50";
- concurrent-load latency with K async clients (batcher coalescing);
- raw in-process graph-replay latency (no HTTP) for reference.

Prints one JSON line. Run (GPU box):
    python benchmarks/bench_serve.py --requests 500 --concurrency 16
"""

import argparse
import asyncio
import json
import os
import socket
import statistics
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--requests", type=int, default=500)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--concurrency", type=int, default=16)
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--no-batch", action="store_true")
    p.add_argument(
        "--workers",
        type=int,
        default=0,
        help=">0: serve via `unionml-amd serve` subprocess with N uvicorn workers",
    )
    return p.parse_args()


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def pct(xs, q):
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(q * len(xs)))]


def _load_proc(q, url, one_row, clients, reqs_per_client):
    """One load-generator process: `clients` async clients, each posting
    `reqs_per_client` sequential requests."""
    import httpx

    async def run():
        async with httpx.AsyncClient(base_url=url, timeout=60.0) as client:

            async def worker(n):
                times = []
                for _ in range(n):
                    t0 = time.perf_counter()
                    r = await client.post("/predict", json={"features": one_row})
                    times.append((time.perf_counter() - t0) * 1000.0)
                    assert r.status_code == 200
                return times

            out = await asyncio.gather(*(worker(reqs_per_client) for _ in range(clients)))
            return [t for ts in out for t in ts]

    q.put(asyncio.run(run()))


def main():
    args = parse_args()
    import torch
    import uvicorn
    from fastapi import FastAPI

    import httpx

    from unionml_amd.models.mlp import model

    use_gpu = torch.cuda.is_available()

    # train on synthetic digits-shaped data (no network for datasets)
    model.artifact = None
    model.train(
        trainer_kwargs={"epochs": 5, "lr": 3e-3},
        loader_kwargs=None,
        n=4096,
        synthetic=True,
    )

    port = args.port or _free_port()
    url = f"http://127.0.0.1:{port}"
    server = thread = proc = None
    if args.workers > 0:
        # real multi-process serving through the CLI: save the artifact,
        # spawn `unionml-amd serve` with N uvicorn workers (each loads
        # the artifact and captures its own hipGraphs)
        import subprocess
        import tempfile

        tmp = os.environ.get("UNIONML_BENCH_DIR") or tempfile.mkdtemp(
            prefix="unionml_serve_bench_"
        )
        os.makedirs(tmp, exist_ok=True)
        artifact_path = os.path.join(tmp, "model.pt")
        model.save(artifact_path)
        env = dict(os.environ)
        env.pop("UNIONML_MODEL_PATH", None)
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
        # stdout MUST not be an undrained pipe: uvicorn logs would fill the
        # pipe buffer under load and deadlock the server
        server_log = open(os.path.join(tmp, "server.log"), "wb")
        proc = subprocess.Popen(
            [
                sys.executable, "-m", "unionml_amd.cli", "serve",
                "unionml_amd.models.mlp_serve:app",
                "--model-path", artifact_path,
                "--port", str(port), "--workers", str(args.workers),
            ],
            stdout=server_log, stderr=subprocess.STDOUT, env=env,
        )
    else:
        app = FastAPI()
        model.serve(app, batch=not args.no_batch, max_batch_size=64, max_delay_ms=0.0)
        config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="warning")
        server = uvicorn.Server(config)
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()

    deadline = time.monotonic() + 120
    while time.monotonic() < deadline:
        if proc is not None and proc.poll() is not None:
            try:
                with open(server_log.name, "rb") as fh:
                    print(fh.read().decode()[-3000:], file=sys.stderr)
            except OSError:
                pass
            sys.exit(1)
        try:
            if httpx.get(f"{url}/health", timeout=1.0).status_code == 200:
                break
        except httpx.HTTPError:
            time.sleep(0.2)
    else:
        print("server never became healthy", file=sys.stderr)
        sys.exit(1)

    one_row = [{f"p{i}": float((i * 7) % 16) for i in range(64)}]

    # --- sequential latency -------------------------------------------------
    with httpx.Client(base_url=url, timeout=10.0) as client:
        for _ in range(args.warmup):
            client.post("/predict", json={"features": one_row})
        lat = []
        for _ in range(args.requests):
            t0 = time.perf_counter()
            r = client.post("/predict", json={"features": one_row})
            lat.append((time.perf_counter() - t0) * 1000.0)
            assert r.status_code == 200, r.text

    # --- concurrent load (multi-process generator: one asyncio client
    # process saturates at ~650 req/s — far below the server) ---------------
    import multiprocessing as mp

    n_procs = min(8, max(1, args.concurrency // 4))
    clients_per_proc = max(1, args.concurrency // n_procs)
    reqs_per_client = max(1, args.requests // (n_procs * clients_per_proc))

    ctx = mp.get_context("spawn")  # no CUDA state in the generator procs
    queue = ctx.Queue()
    procs = [
        ctx.Process(
            target=_load_proc,
            args=(queue, url, one_row, clients_per_proc, reqs_per_client),
        )
        for _ in range(n_procs)
    ]
    t0 = time.perf_counter()
    for p in procs:
        p.start()
    conc_lat = []
    for _ in procs:
        conc_lat.extend(queue.get())
    conc_wall = time.perf_counter() - t0
    for p in procs:
        p.join()

    # --- raw in-process graph replay (no HTTP) ------------------------------
    raw_p50 = None
    if use_gpu:
        import numpy as np

        from unionml_amd.serving.graph_runner import TabularGraphRunner

        runner = TabularGraphRunner(model.artifact.model_object, max_batch_size=64)
        x = np.asarray([[float((i * 7) % 16) for i in range(64)]], dtype=np.float32)
        for _ in range(50):
            runner(x)
        raw = []
        for _ in range(500):
            t0 = time.perf_counter()
            runner(x)
            raw.append((time.perf_counter() - t0) * 1000.0)
        raw_p50 = pct(raw, 0.50)

    batcher_stats = None
    if server is not None:
        try:
            # dig the batcher out of the serving closure for server-side stats
            for route in app.router.on_startup:
                cells = getattr(route, "__closure__", None) or []
                for c in cells:
                    v = c.cell_contents
                    if isinstance(v, dict) and v.get("batcher") is not None:
                        batcher_stats = v["batcher"].stats()
        except Exception:
            pass
        server.should_exit = True
        thread.join(timeout=10)
    if proc is not None:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except Exception:
            proc.kill()

    print(
        json.dumps(
            {
                "metric": "predict_p50_latency_ms",
                "value": pct(lat, 0.50),
                "unit": "ms",
                "n_gpus": 1 if use_gpu else 0,
                "steps": args.requests,
                "warmup": args.warmup,
                "higher_is_better": False,
                "vs_baseline": None,
                "dtype": "bf16" if use_gpu else "fp32",
                "data": "synthetic",
                "config": {
                    "model": "digits_mlp_64x32x10",
                    "batcher": not args.no_batch,
                    "workers": args.workers or 1,
                    "sequential": {
                        "p50_ms": pct(lat, 0.50),
                        "p90_ms": pct(lat, 0.90),
                        "p99_ms": pct(lat, 0.99),
                        "mean_ms": statistics.fmean(lat),
                    },
                    "concurrent": {
                        "clients": args.concurrency,
                        "p50_ms": pct(conc_lat, 0.50),
                        "p99_ms": pct(conc_lat, 0.99),
                        "requests_per_s": len(conc_lat) / conc_wall,
                    },
                    "raw_graph_replay_p50_ms": raw_p50,
                    "batcher_server_side": batcher_stats,
                },
            }
        )
    )


if __name__ == "__main__":
    main()
