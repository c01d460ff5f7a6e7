#!/usr/bin/env python3
"""Flagship benchmark: digits-MLP training throughput (train samples/sec)
plus an optional serving leg (/predict p50).

BASELINE.json metric: "train samples/sec + /predict p50 latency, MLP
digits-shape, 1/2/4/8 MI355X". The default mode measures the training
leg on synthetic digits-shaped data (64 features, 10 classes,
random-init weights — no network for datasets) with the CDNA4 fused hot
path: one optimizer step = the fully-fused fwd/bwd/reduce/Adam MFMA
kernel (or, under DP, fused fwd/bwd kernel + RCCL gradient all-reduce +
fused Adam), with up to --graph-steps optimizer steps captured per
hipGraph. ``--mode serve`` measures the /predict leg instead: real HTTP
requests against the FastAPI serving app (dynamic batcher + bucketed
hipGraph replay), reporting client-side p50 latency.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W              # single GPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: per-GPU batch is fixed (default 2048); reported value is
the WHOLE-JOB samples/sec aggregated over all ranks.

Robustness rules (first-contact world>1 safety):
  * hipGraph capture is VALIDATED before the timed loop: one step is
    captured, replayed, and checksummed against the same step run
    eagerly from identical state; mismatch or capture failure falls
    back to eager — collectively across ranks (all-reduce MIN on the
    verdict) so no rank replays graphs alone.
  * engine fallbacks are per-engine, not per-process: unmet persistent-
    kernel constraints drop to the fused engine instead of dying.
  * the reported engine label reflects what actually ran in the timed
    region (graph_replays_used is in the config blob).
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
    p.add_argument(
        "--mode",
        choices=["train", "serve"],
        default="train",
        help="train: optimizer-step throughput (driver default); "
        "serve: /predict HTTP p50 over the FastAPI app",
    )
    p.add_argument("--serve-clients", type=int, default=1,
                   help="concurrent HTTP clients in serve mode")
    p.add_argument("--serve-workers", type=int, default=1,
                   help="SO_REUSEPORT server workers in serve mode")
    p.add_argument("--serve-app", choices=["digits", "mnist"], default="digits",
                   help="serve mode: flagship digits app (specialized kernels) "
                   "or the MNIST-shape app (generalized kernels)")
    return p.parse_args()


def _result(args, n_gpus, B, elapsed, engine, loss, replays=None):
    cfg = {
        "model": "digits_mlp_64x32x10",
        "global_batch": B * n_gpus,
        "seq_len": None,
        "parallelism": f"dp{n_gpus}",
        "engine": engine,
        "final_loss": loss,
    }
    if replays is not None:
        cfg["graph_replays_used"] = replays
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
        "config": cfg,
    }


# ---------------------------------------------------------------------------
# serve mode: boot the real FastAPI app + uvicorn, measure /predict p50
# ---------------------------------------------------------------------------

def _client_loop(port, body, n_requests, out_lats, barrier=None):
    """One load client: a persistent keep-alive connection issuing POSTs
    (new-connection-per-request clients measure TCP setup, not the
    server)."""
    import http.client

    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=10)
    headers = {"Content-Type": "application/json", "Connection": "keep-alive"}
    if barrier is not None:
        barrier.wait()
    for _ in range(n_requests):
        t0 = time.perf_counter()
        try:
            conn.request("POST", "/predict", body=body, headers=headers)
            resp = conn.getresponse()
            resp.read()
            status = resp.status
        except (OSError, http.client.HTTPException):
            # connection reset (e.g. a worker restarting): reconnect,
            # count the failure, keep the client alive
            conn.close()
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=10)
            out_lats.append(float("nan"))
            continue
        if status != 200:  # count, don't kill the client thread
            out_lats.append(float("nan"))
            continue
        out_lats.append((time.perf_counter() - t0) * 1000.0)
    conn.close()


def _client_proc(port, body, n_threads, n_requests_per_thread, q, start_evt):
    import threading

    per = [[] for _ in range(n_threads)]
    barrier = threading.Barrier(n_threads)

    def run(i):
        start_evt.wait()
        _client_loop(port, body, n_requests_per_thread, per[i], barrier)

    threads = [threading.Thread(target=run, args=(i,)) for i in range(n_threads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    q.put([x for sub in per for x in sub])


def run_serve_mode(args):
    import socket
    import subprocess
    import urllib.request

    steps = args.steps if args.steps else 2000
    warmup = args.warmup if args.warmup else 200
    port = 18321
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    # train a quick artifact for the server to load (synthetic data)
    import tempfile

    if args.serve_app == "mnist":
        from unionml_amd.models.mnist import model as train_model

        train_model.artifact = None
        train_model.train(trainer_kwargs={"epochs": 5, "lr": 3e-3}, n=4096)
        serve_spec = "unionml_amd.models.mnist_serve:app"
        rng_rows = [[float((i * 7) % 16) for i in range(784)]]
    else:
        from unionml_amd.models.mlp import model as train_model

        train_model.artifact = None
        train_model.train(trainer_kwargs={"epochs": 5, "lr": 3e-3}, synthetic=True, n=4096)
        serve_spec = "unionml_amd.models.mlp_serve:app"
        rng_rows = [{f"p{i}": float((i * 7) % 16) for i in range(64)}]
    artifact_path = os.path.join(tempfile.mkdtemp(prefix="unionml_bench_"), "model.pt")
    train_model.save(artifact_path)

    env = dict(os.environ)
    env["UNIONML_MODEL_PATH"] = artifact_path
    if args.serve_workers > 1:
        # the framework's SO_REUSEPORT multi-worker supervisor (each
        # worker owns its bucketed inference hipGraphs)
        server = subprocess.Popen(
            [sys.executable, "-m", "unionml_amd.cli", "serve", serve_spec,
             "--host", "127.0.0.1",
             "--port", str(port), "--workers", str(args.serve_workers)],
            env=env,
        )
    else:
        server = subprocess.Popen(
            [sys.executable, "-m", "uvicorn", "--host", "127.0.0.1", "--port",
             str(port), "--log-level", "warning", serve_spec],
            env=env,
        )
    try:
        # wait for readiness: behind SO_REUSEPORT each worker loads its
        # artifact and captures its own hipGraphs independently, so
        # demand a long streak of consecutive healthy responses (enough
        # to have hit every worker with high probability)
        body = json.dumps({"features": rng_rows}).encode()
        deadline = time.time() + 180
        streak, need = 0, 16 * max(1, args.serve_workers)
        while streak < need:
            try:
                urllib.request.urlopen(f"http://127.0.0.1:{port}/health", timeout=2)
                streak += 1
            except Exception:
                streak = 0
                if time.time() > deadline:
                    raise RuntimeError("serve-mode server never became healthy")
                time.sleep(0.5)

        warm = []
        _client_loop(port, body, warmup, warm)
        lats = []
        if args.serve_clients <= 1:
            t_start = time.perf_counter()
            _client_loop(port, body, steps, lats)
            elapsed = time.perf_counter() - t_start
        else:
            # fan the load clients across PROCESSES (a single GIL-bound
            # client process inflates tail latencies at high rps)
            import multiprocessing as mp

            n_clients = args.serve_clients
            n_procs = min(n_clients, 8)
            threads_per = n_clients // n_procs
            per_thread = steps // (n_procs * threads_per)
            ctx = mp.get_context("fork")
            q = ctx.Queue()
            start_evt = ctx.Event()
            procs = [
                ctx.Process(
                    target=_client_proc,
                    args=(port, body, threads_per, per_thread, q, start_evt),
                )
                for _ in range(n_procs)
            ]
            for p_ in procs:
                p_.start()
            time.sleep(0.5)  # let every process build its connections
            t_start = time.perf_counter()
            start_evt.set()
            lats = []
            for _ in procs:
                lats.extend(q.get())
            for p_ in procs:
                p_.join()
            elapsed = time.perf_counter() - t_start
        errors = sum(1 for x in lats if x != x)
        lats = [x for x in lats if x == x]
        if not lats:
            raise RuntimeError(f"serve mode: all {errors} requests failed")
        steps = len(lats)
        lats.sort()
        p = lambda q: lats[min(len(lats) - 1, int(q * len(lats)))]  # noqa: E731
        print(json.dumps({
            "metric": "predict_p50_ms",
            "value": p(0.50),
            "unit": "ms",
            "n_gpus": 1,
            "steps": steps,
            "warmup": warmup,
            "ms_per_step": elapsed / steps * 1000.0,
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "mnist_mlp_784x128x10" if args.serve_app == "mnist"
                else "digits_mlp_64x32x10",
                "global_batch": 1,
                "seq_len": None,
                "parallelism": "serve1",
                "clients": args.serve_clients,
                "workers": args.serve_workers,
                "p90_ms": p(0.90),
                "p99_ms": p(0.99),
                "rps": steps / elapsed,
                "errors": errors,
            },
        }))
    finally:
        server.terminate()
        server.wait(timeout=10)


# ---------------------------------------------------------------------------
# train mode
# ---------------------------------------------------------------------------

def snapshot_state(clf):
    keys = ["master", "bfmirror", "m", "v", "t_dev", "grads"]
    s = {k: getattr(clf, k).clone() for k in keys}
    for opt in ("counter", "wimg"):
        t = getattr(clf, opt, None)
        if t is not None:
            s[opt] = t.clone()
    return s


def restore_state(clf, s):
    for k, v in s.items():
        getattr(clf, k).copy_(v)


def checksum_state(clf):
    return (
        float(clf.master.double().sum().item()),
        float(clf.master.double().abs().sum().item()),
        int(clf.t_dev.item()),
    )


def validate_graph_capture(clf, eager_step, dist, device):
    """Capture one optimizer step, replay it, and compare the resulting
    state against the same step run eagerly from identical initial
    state. Returns True only if EVERY rank validates (collective MIN),
    so no rank trusts graphs alone. RCCL-inside-hipGraph is exactly the
    kind of thing that breaks at world=8 first — never enter the timed
    loop on an unvalidated capture."""
    s0 = snapshot_state(clf)
    eager_step(0)
    torch.cuda.synchronize()
    want = checksum_state(clf)
    restore_state(clf, s0)
    torch.cuda.synchronize()

    g = None
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            eager_step(0)
    except RuntimeError as exc:
        print(f"[bench] graph capture validation failed ({exc})", file=sys.stderr)
        g = None

    # agree on capture success BEFORE anyone replays: a rank replaying a
    # captured RCCL collective whose peers never replay would hang the
    # job (capture failures are not guaranteed to be symmetric)
    if dist is not None:
        flag = torch.tensor([0.0 if g is None else 1.0], device=device)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        if flag.item() < 0.5:
            restore_state(clf, s0)
            torch.cuda.synchronize()
            return False
    elif g is None:
        restore_state(clf, s0)
        torch.cuda.synchronize()
        return False

    ok = True
    try:
        g.replay()
        torch.cuda.synchronize()
        got = checksum_state(clf)
        ok = (
            got[2] == want[2]
            and abs(got[0] - want[0]) <= 1e-6 * max(1.0, abs(want[0]))
            and abs(got[1] - want[1]) <= 1e-6 * max(1.0, want[1])
        )
        if not ok:
            print(f"[bench] graph validation mismatch: eager={want} replay={got}",
                  file=sys.stderr)
    except RuntimeError as exc:
        print(f"[bench] graph replay validation failed ({exc})", file=sys.stderr)
        ok = False
    del g
    restore_state(clf, s0)
    torch.cuda.synchronize()

    if dist is not None:
        flag = torch.tensor([1.0 if ok else 0.0], device=device)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        ok = bool(flag.item() > 0.5)
    return ok


def main():
    args = parse_args()
    if args.mode == "serve":
        run_serve_mode(args)
        return

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
    # per-engine fallback: unmet persistent-kernel constraints drop to
    # the fused engine instead of aborting the whole process
    if engine == "persistent" and not (
        use_gpu and world == 1 and B % 128 == 0 and (B * M) % B == 0
    ):
        print("[bench] persistent engine constraints unmet; falling back to fused",
              file=sys.stderr)
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
        def run_persistent(k):
            return ext.mlp_train_steps(
                Xbf, y, B, k, clf.master, clf.bfmirror, clf.m, clf.v,
                clf.t_dev, loss_out, args.lr, 0.9, 0.999, 1e-8,
            )

        if run_persistent(args.warmup):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            assert run_persistent(args.steps)
            torch.cuda.synchronize()
            elapsed = time.perf_counter() - t0
            loss = float(loss_out.item())
            assert loss == loss and loss < 1e6, f"training diverged: loss={loss}"
            print(json.dumps(_result(args, 1, B, elapsed, "persistent_steps_kernel", loss)))
            return
        print("[bench] persistent kernel declined; falling back to fused",
              file=sys.stderr)
        engine = "fused"

    # warm up communicator + kernels
    for i in range(3):
        eager_step((i % M) * B)

    # validate capture+replay before trusting graphs in the timed loop,
    # then capture G optimizer steps per graph (G | M) so one replay
    # advances G steps — all work identical, launch overhead amortized
    G = max(1, args.graph_steps)
    while M % G:
        G -= 1
    graphs = None
    if use_gpu and not args.no_graph:
        if validate_graph_capture(clf, eager_step, dist, clf.device):
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
            # all ranks must agree on the engine (a lone eager rank would
            # desync the RCCL collective order)
            if dist is not None:
                flag = torch.tensor([0.0 if graphs is None else 1.0], device=clf.device)
                dist.all_reduce(flag, op=dist.ReduceOp.MIN)
                if flag.item() < 0.5:
                    graphs = None
        else:
            print("[bench] graph validation failed; eager stepping", file=sys.stderr)

    replay_count = [0]

    def run_steps(k0: int, nsteps: int):
        """Advance exactly ``nsteps`` optimizer steps from global step k0:
        G-aligned stretches replay whole-chunk graphs, edges run eagerly."""
        k = k0
        end = k0 + nsteps
        while k < end:
            if graphs is not None and k % G == 0 and k + G <= end:
                graphs[(k // G) % (M // G)].replay()
                replay_count[0] += 1
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
    replay_count[0] = 0  # count only the timed region
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
        # the engine label must reflect what RAN in the timed region
        replays = replay_count[0]
        if graphs is not None and replays > 0:
            engine_name = f"{engine}+hipgraph{G}"
        else:
            engine_name = f"{engine}+eager"
        print(json.dumps(_result(args, n_gpus, B, elapsed, engine_name, loss,
                                 replays=replays)))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
