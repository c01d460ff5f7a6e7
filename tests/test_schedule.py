"""Schedule unit tests (coverage shape of the reference's
tests/unit/test_schedule.py: launchplan creation for cron/fixed-rate and
error paths), plus cron-evaluator tests for this build's own scheduler."""

import datetime

import pytest

from unionml_amd.schedule import (
    LaunchPlan,
    Schedule,
    ScheduleType,
    create_scheduled_launchplan,
    cron_matches,
    next_fire_time,
)


def test_create_cron_launchplan():
    s = Schedule(type="trainer", name="nightly", expression="0 2 * * *")
    lp = create_scheduled_launchplan("m.train", "nightly", s)
    assert isinstance(lp, LaunchPlan)
    assert lp.workflow_name == "m.train"
    assert not lp.active
    lp.activate()
    assert lp.active


def test_create_fixed_rate_launchplan():
    s = Schedule(
        type=ScheduleType.predictor,
        name="often",
        fixed_rate=datetime.timedelta(minutes=5),
        inputs={"n": 3},
    )
    lp = create_scheduled_launchplan("m.predict", "often", s)
    assert lp.fixed_inputs == {"n": 3}


def test_mutual_exclusion():
    s = Schedule(
        type="trainer",
        name="bad",
        expression="* * * * *",
        fixed_rate=datetime.timedelta(minutes=1),
    )
    with pytest.raises(ValueError):
        create_scheduled_launchplan("m.train", "bad", s)


def test_neither_expression_nor_rate():
    s = Schedule(type="trainer", name="bad")
    with pytest.raises(ValueError):
        create_scheduled_launchplan("m.train", "bad", s)


def test_invalid_cron_rejected():
    s = Schedule(type="trainer", name="bad", expression="not a cron")
    with pytest.raises(ValueError):
        create_scheduled_launchplan("m.train", "bad", s)


def test_cron_matches():
    t = datetime.datetime(2026, 9, 13, 2, 0)  # a Sunday
    assert cron_matches("0 2 * * *", t)
    assert not cron_matches("0 3 * * *", t)
    assert cron_matches("*/15 * * * *", t.replace(minute=45))
    assert cron_matches("0 2 * * 0", t)  # Sunday = 0
    assert not cron_matches("0 2 * * 1", t)


def test_next_fire_time_cron():
    s = Schedule(type="trainer", name="s", expression="30 4 * * *")
    after = datetime.datetime(2026, 9, 13, 2, 0)
    assert next_fire_time(s, after) == datetime.datetime(2026, 9, 13, 4, 30)


def test_next_fire_time_fixed_rate():
    s = Schedule(type="trainer", name="s", fixed_rate=datetime.timedelta(hours=2))
    after = datetime.datetime(2026, 9, 13, 2, 0)
    assert next_fire_time(s, after) == datetime.datetime(2026, 9, 13, 4, 0)


def test_parse_iso_duration():
    from unionml_amd.schedule import parse_iso_duration

    assert parse_iso_duration("PT10M") == datetime.timedelta(minutes=10)
    assert parse_iso_duration("P2DT3H30M15S") == datetime.timedelta(
        days=2, hours=3, minutes=30, seconds=15
    )
    assert parse_iso_duration("P1W") == datetime.timedelta(weeks=1)
    assert parse_iso_duration("-PT5M") == -datetime.timedelta(minutes=5)
    for bad in ("", "P", "PT", "10M", "P1M2W", "nonsense"):
        with pytest.raises(ValueError):
            parse_iso_duration(bad)


def test_next_fire_time_cron_offset():
    """A cron offset shifts every kickoff by the duration (reference
    schedule.py:99-103 passes offset into the platform CronSchedule)."""
    s = Schedule(type="trainer", name="s", expression="30 4 * * *", offset="PT15M")
    after = datetime.datetime(2026, 9, 13, 2, 0)
    assert next_fire_time(s, after) == datetime.datetime(2026, 9, 13, 4, 45)
    # strictly-after holds across the offset boundary: at 04:40 the
    # 04:30 match (firing 04:45) is still pending
    after2 = datetime.datetime(2026, 9, 13, 4, 40)
    assert next_fire_time(s, after2) == datetime.datetime(2026, 9, 13, 4, 45)
    # past 04:45 we roll to the next day
    after3 = datetime.datetime(2026, 9, 13, 4, 50)
    assert next_fire_time(s, after3) == datetime.datetime(2026, 9, 14, 4, 45)


def test_negative_offset():
    s = Schedule(type="trainer", name="s", expression="0 5 * * *", offset="-PT30M")
    after = datetime.datetime(2026, 9, 13, 2, 0)
    assert next_fire_time(s, after) == datetime.datetime(2026, 9, 13, 4, 30)


def test_invalid_offset_rejected_at_deploy():
    s = Schedule(type="trainer", name="s", expression="0 5 * * *", offset="15 minutes")
    with pytest.raises(ValueError):
        create_scheduled_launchplan("m.train", "s", s)
