"""GpuAllocator: flock-guarded round-robin HIP_VISIBLE_DEVICES
assignment (the advisor-flagged race fixed in round 2). CPU-only test:
device count comes from a fake HIP_VISIBLE_DEVICES."""

import multiprocessing as mp
import os
from collections import Counter
from pathlib import Path


def _assign_worker(root, out_q):
    os.environ["HIP_VISIBLE_DEVICES"] = "0,1,2,3"
    from unionml_amd.remote import GpuAllocator

    alloc = GpuAllocator(Path(root))
    out_q.put(alloc.assign(1))


def test_round_robin_and_wraparound(tmp_path, monkeypatch):
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "0,1,2,3")
    from unionml_amd.remote import GpuAllocator

    alloc = GpuAllocator(tmp_path)
    assert alloc.n_devices == 4
    seq = [alloc.assign(1) for _ in range(8)]
    assert seq == ["0", "1", "2", "3", "0", "1", "2", "3"]
    # multi-GPU request: consecutive distinct devices
    assert alloc.assign(2) == "0,1"
    # requests larger than the node clamp to all devices
    assert alloc.assign(8) == "2,3,0,1"


def test_no_gpus_returns_none(tmp_path, monkeypatch):
    monkeypatch.delenv("HIP_VISIBLE_DEVICES", raising=False)
    from unionml_amd.remote import GpuAllocator

    alloc = GpuAllocator(tmp_path)
    if alloc.n_devices == 0:
        assert alloc.assign(1) is None
    assert alloc.assign(0) is None


def test_concurrent_assign_across_processes(tmp_path):
    """8 processes assigning concurrently must get a balanced round-robin
    (each of the 4 devices exactly twice) — the flock prevents two
    workers reading the same counter value."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_assign_worker, args=(str(tmp_path), q)) for _ in range(8)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=60) for _ in procs]
    for p in procs:
        p.join(timeout=60)
    counts = Counter(results)
    assert sorted(counts) == ["0", "1", "2", "3"]
    assert all(v == 2 for v in counts.values()), counts
