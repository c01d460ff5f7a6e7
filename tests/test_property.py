"""Property-based tests (hypothesis) for the pure-logic primitives:
row sharding, serve-bucket selection, cron evaluation, and the
type-guard signature machinery."""

import datetime

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from unionml_amd.parallel.ddp import shard
from unionml_amd.schedule import cron_matches, next_fire_time, Schedule, ScheduleType
from unionml_amd.serving.batcher import bucket_for


@given(n=st.integers(0, 5000), world=st.integers(1, 16))
@settings(max_examples=200, deadline=None)
def test_shard_partitions_exactly(n, world):
    arr = np.arange(n)
    parts = [shard(arr, r, world) for r in range(world)]
    # exact partition, order-preserving, sizes differ by at most 1
    assert np.concatenate(parts).tolist() == arr.tolist() if n else True
    sizes = [len(p) for p in parts]
    assert sum(sizes) == n
    assert max(sizes) - min(sizes) <= 1


@given(n=st.integers(1, 10000), max_batch=st.integers(1, 1024))
@settings(max_examples=200, deadline=None)
def test_bucket_for_invariants(n, max_batch):
    b = bucket_for(n, max_batch)
    assert 1 <= b <= max_batch
    if n <= max_batch:
        assert b >= n            # bucket always fits the request
        assert b & (b - 1) == 0 or b == max_batch  # power of two (or cap)
        if b > 1 and b & (b - 1) == 0 and b // 2 >= 1:
            assert b // 2 < n or b == 1  # smallest such bucket


@given(
    minute=st.integers(0, 59),
    hour=st.integers(0, 23),
    when=st.datetimes(
        min_value=datetime.datetime(2020, 1, 1),
        max_value=datetime.datetime(2030, 1, 1),
    ),
)
@settings(max_examples=200, deadline=None)
def test_cron_exact_minute_hour(minute, hour, when):
    expr = f"{minute} {hour} * * *"
    expected = when.minute == minute and when.hour == hour
    assert cron_matches(expr, when) == expected


@given(
    step=st.integers(1, 30),
    when=st.datetimes(
        min_value=datetime.datetime(2022, 1, 1),
        max_value=datetime.datetime(2028, 1, 1),
    ),
)
@settings(max_examples=100, deadline=None)
def test_cron_step_minutes(step, when):
    assert cron_matches(f"*/{step} * * * *", when) == (when.minute % step == 0)


@given(
    rate_minutes=st.integers(1, 240),
    when=st.datetimes(
        min_value=datetime.datetime(2024, 1, 1),
        max_value=datetime.datetime(2026, 1, 1),
    ),
)
@settings(max_examples=100, deadline=None)
def test_fixed_rate_next_fire_is_in_future_by_rate(rate_minutes, when):
    s = Schedule(
        type=ScheduleType.trainer,
        name="s",
        fixed_rate=datetime.timedelta(minutes=rate_minutes),
    )
    nxt = next_fire_time(s, when)
    assert nxt == when + datetime.timedelta(minutes=rate_minutes)


@given(
    when=st.datetimes(
        min_value=datetime.datetime(2024, 1, 1),
        max_value=datetime.datetime(2026, 1, 1),
    )
)
@settings(max_examples=50, deadline=None)
def test_cron_next_fire_matches_and_is_future(when):
    s = Schedule(type=ScheduleType.trainer, name="s", expression="*/5 * * * *")
    nxt = next_fire_time(s, when)
    assert nxt > when
    assert cron_matches("*/5 * * * *", nxt)
    # no earlier match in between (check minute granularity)
    probe = when.replace(second=0, microsecond=0) + datetime.timedelta(minutes=1)
    while probe < nxt.replace(second=0, microsecond=0):
        assert not cron_matches("*/5 * * * *", probe)
        probe += datetime.timedelta(minutes=1)


# ---------------- generalized-geometry invariants (r02) ----------------

from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=200, deadline=None)
@given(
    inf=st.integers(min_value=1, max_value=2048),
    hid=st.integers(min_value=1, max_value=256),
    cls=st.integers(min_value=2, max_value=32),
)
def test_geometry_invariants(inf, hid, cls):
    """Padding math holds for every constructible geometry: alignment,
    monotone containment, offset chain consistency, slab bounds."""
    from unionml_amd.ops.reference import Geometry

    g = Geometry(inf, hid, cls)
    assert g.inp % 32 == 0 and g.inp >= inf and g.inp - inf < 32
    assert g.hid >= hid and g.hid in (32, 64, 128, 256)
    assert g.cpad in (16, 32) and g.cpad >= cls
    assert g.off_b1 == g.inp * g.hid
    assert g.off_w2 == g.off_b1 + g.hid
    assert g.off_b2 == g.off_w2 + g.hid * g.cpad
    assert g.nparam == g.off_b2 + g.cpad
    assert g.slab_stride % 16 == 0
    assert g.wimg_n == g.hid * g.inp + g.hid * 32 + g.cpad * g.hid
    if g.is_specialized:
        assert (inf, hid, cls) == (64, 32, 10)


@settings(max_examples=50, deadline=None)
@given(
    inf=st.integers(min_value=2, max_value=300),
    hid=st.integers(min_value=2, max_value=256),
    cls=st.integers(min_value=2, max_value=32),
    n=st.integers(min_value=1, max_value=64),
)
def test_cpu_step_padded_params_stay_zero(inf, hid, cls, n):
    """One CPU training step at a random geometry never leaks gradient
    or parameter mass into the padded region."""
    import torch

    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    clf = TabularMLP(in_features=inf, hidden=hid, classes=cls, device="cpu", seed=0)
    X = torch.randn(n, inf)
    y = torch.randint(0, cls, (n,), dtype=torch.int32)
    clf.fit_standardizer(X)
    clf.train_epochs(clf.stage(X), y, epochs=1, batch_size=n, lr=1e-3)
    W1, b1, W2, b2 = ref.unpack_master_g(clf.g, clf.master)
    assert (W1[inf:, :] == 0).all() and (W1[:, hid:] == 0).all()
    assert (W2[hid:, :] == 0).all() and (W2[:, cls:] == 0).all()
    assert (b1[hid:] == 0).all() and (b2[cls:] == 0).all()
