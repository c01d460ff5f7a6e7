"""Scheduling — cron / fixed-rate schedules for training & prediction.

Capability parity with the reference (unionml/schedule.py:12-123):
``Schedule`` describes when a train or predict workflow should run;
``create_scheduled_launchplan`` compiles it, together with a workflow,
into a :class:`LaunchPlan` the backend scheduler executes.

Because this build carries its own execution backend (no Flyte), the
launch plan is a plain record plus a cron evaluator; the backend
(unionml_amd/remote.py) runs an in-process scheduler loop over the
active launch plans.
"""

import datetime
import enum
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Union


class ScheduleType(enum.Enum):
    """What a schedule launches (reference: schedule.py:12-19)."""

    trainer = "trainer"
    predictor = "predictor"


@dataclass
class Schedule:
    """A cron or fixed-rate schedule (reference: schedule.py:22-64).

    Exactly one of ``expression`` (5-field cron) or ``fixed_rate``
    (timedelta) must be set. ``time_arg`` names a workflow input that
    receives the kickoff datetime. ``inputs`` are fixed workflow inputs.
    """

    type: Union[ScheduleType, str]
    name: str
    expression: Optional[str] = None
    offset: Optional[str] = None
    fixed_rate: Optional[datetime.timedelta] = None
    time_arg: Optional[str] = None
    inputs: Optional[Dict[str, Any]] = None
    reader_time_arg: Optional[str] = None
    activate_on_deploy: bool = True
    launchplan_kwargs: Optional[Dict[str, Any]] = None

    def __post_init__(self):
        if isinstance(self.type, str):
            self.type = ScheduleType(self.type)


# ----------------------------------------------------------------------
# cron evaluation (5-field: minute hour day-of-month month day-of-week)
# ----------------------------------------------------------------------


def _parse_cron_field(spec: str, lo: int, hi: int) -> List[int]:
    values = set()
    for part in spec.split(","):
        step = 1
        if "/" in part:
            part, step_s = part.split("/")
            step = int(step_s)
        if part in ("*", ""):
            rng = range(lo, hi + 1)
        elif "-" in part:
            a, b = part.split("-")
            rng = range(int(a), int(b) + 1)
        else:
            rng = range(int(part), int(part) + 1)
        values.update(v for v in rng if (v - lo) % step == 0 or (step > 1 and v % step == 0))
    bad = [v for v in values if v < lo or v > hi]
    if bad:
        raise ValueError(f"cron field '{spec}' values {bad} out of range [{lo},{hi}]")
    return sorted(values)


def cron_matches(expression: str, when: datetime.datetime) -> bool:
    """True when ``when`` matches the 5-field cron ``expression``."""
    fields = expression.split()
    if len(fields) != 5:
        raise ValueError(f"cron expression must have 5 fields, got {expression!r}")
    minute, hour, dom, month, dow = fields
    return (
        when.minute in _parse_cron_field(minute, 0, 59)
        and when.hour in _parse_cron_field(hour, 0, 23)
        and when.day in _parse_cron_field(dom, 1, 31)
        and when.month in _parse_cron_field(month, 1, 12)
        and (when.weekday() + 1) % 7 in _parse_cron_field(dow, 0, 6)
    )


def parse_iso_duration(spec: str) -> datetime.timedelta:
    """Parse an ISO-8601 duration (``P2DT3H30M15S``, ``PT10M``, ``-PT5M``)
    into a timedelta. Calendar units wider than weeks (months/years) are
    rejected — they have no fixed length."""
    import re

    m = re.fullmatch(
        r"(?P<sign>[-+])?P(?:(?P<weeks>\d+(?:\.\d+)?)W)?(?:(?P<days>\d+(?:\.\d+)?)D)?"
        r"(?:T(?:(?P<hours>\d+(?:\.\d+)?)H)?(?:(?P<minutes>\d+(?:\.\d+)?)M)?"
        r"(?:(?P<seconds>\d+(?:\.\d+)?)S)?)?",
        spec.strip(),
    )
    if not m or spec.strip().rstrip("+-") in ("P", "PT", ""):
        raise ValueError(f"invalid ISO-8601 duration: {spec!r}")
    parts = {k: float(v) for k, v in m.groupdict().items() if v not in (None, "-", "+")}
    sign = -1 if m.group("sign") == "-" else 1
    if not any(k in parts for k in ("weeks", "days", "hours", "minutes", "seconds")):
        raise ValueError(f"invalid ISO-8601 duration: {spec!r}")
    return sign * datetime.timedelta(
        weeks=parts.get("weeks", 0.0),
        days=parts.get("days", 0.0),
        hours=parts.get("hours", 0.0),
        minutes=parts.get("minutes", 0.0),
        seconds=parts.get("seconds", 0.0),
    )


def next_fire_time(schedule: Schedule, after: datetime.datetime) -> datetime.datetime:
    """Next time this schedule should fire strictly after ``after``.

    A cron ``offset`` (ISO-8601 duration, reference schedule.py:99-103)
    shifts each kickoff relative to its cron match: fire times are
    ``cron_match + offset``.
    """
    if schedule.fixed_rate is not None:
        return after + schedule.fixed_rate
    if schedule.expression is None:
        raise ValueError(f"schedule '{schedule.name}' has neither expression nor fixed_rate")
    offset = parse_iso_duration(schedule.offset) if schedule.offset else datetime.timedelta()
    base = after - offset
    t = base.replace(second=0, microsecond=0) + datetime.timedelta(minutes=1)
    for _ in range(60 * 24 * 366):  # search up to ~a year of minutes
        if cron_matches(schedule.expression, t):
            return t + offset
        t += datetime.timedelta(minutes=1)
    raise ValueError(f"cron expression {schedule.expression!r} never fires")


# ----------------------------------------------------------------------
# launch plans
# ----------------------------------------------------------------------


@dataclass
class LaunchPlan:
    """A workflow bound to a schedule (the Flyte-LaunchPlan analog)."""

    name: str
    workflow_name: str
    schedule: Schedule
    fixed_inputs: Dict[str, Any] = field(default_factory=dict)
    active: bool = False

    def activate(self):
        self.active = True

    def deactivate(self):
        self.active = False


def create_scheduled_launchplan(
    workflow_name: str,
    name: str,
    schedule: Schedule,
    **launchplan_kwargs,
) -> LaunchPlan:
    """Compile a Schedule + workflow into a LaunchPlan (reference:
    schedule.py:67-123 — mutual-exclusion checks then CronSchedule vs
    FixedRate construction)."""
    if schedule.expression is not None and schedule.fixed_rate is not None:
        raise ValueError(
            f"schedule '{name}': 'expression' and 'fixed_rate' are mutually exclusive"
        )
    if schedule.expression is None and schedule.fixed_rate is None:
        raise ValueError(f"schedule '{name}': one of 'expression' or 'fixed_rate' is required")
    if schedule.expression is not None:
        # validate eagerly so a bad expression fails at deploy time
        cron_matches(schedule.expression, datetime.datetime.now())
    if schedule.offset is not None and schedule.fixed_rate is not None:
        raise ValueError(f"schedule '{name}': 'offset' only applies to cron expressions")
    if schedule.offset is not None:
        parse_iso_duration(schedule.offset)  # fail at deploy time, not fire time

    fixed_inputs = dict(schedule.inputs or {})
    fixed_inputs.update(launchplan_kwargs.pop("fixed_inputs", {}) or {})
    lp_kwargs = {**(schedule.launchplan_kwargs or {}), **launchplan_kwargs}
    lp = LaunchPlan(
        name=name,
        workflow_name=workflow_name,
        schedule=schedule,
        fixed_inputs=fixed_inputs,
        **lp_kwargs,
    )
    return lp
