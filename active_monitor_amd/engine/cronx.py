"""Standard cron parsing with robfig/cron ParseStandard semantics.

The reference schedules cron-based HealthChecks via
``cron.ParseStandard(spec.Schedule.Cron)`` and derives
``RepeatAfterSec = int(next - now) + 1`` seconds
(healthcheck_controller.go:251-263). This module is a from-scratch
implementation of the same accepted grammar:

- five fields: minute hour day-of-month month day-of-week,
- ``*`` and ``?``, ranges ``a-b``, steps ``*/n`` ``a-b/n`` ``a/n``, lists,
- month names JAN..DEC and day names SUN..SAT (case-insensitive),
- descriptors ``@yearly|@annually``, ``@monthly``, ``@weekly``,
  ``@daily|@midnight``, ``@hourly``, and ``@every <go-duration>``,
- classic dom/dow union rule: when BOTH day fields are restricted, a time
  matches if EITHER matches.
"""
from __future__ import annotations

import re
from dataclasses import dataclass
from datetime import datetime, timedelta, timezone
from typing import FrozenSet, Optional, Tuple


class CronParseError(ValueError):
    pass


_MONTH_NAMES = {
    "jan": 1, "feb": 2, "mar": 3, "apr": 4, "may": 5, "jun": 6,
    "jul": 7, "aug": 8, "sep": 9, "oct": 10, "nov": 11, "dec": 12,
}
_DAY_NAMES = {"sun": 0, "mon": 1, "tue": 2, "wed": 3, "thu": 4, "fri": 5, "sat": 6}

# (min, max, names) per field
_FIELD_BOUNDS = [
    (0, 59, {}),            # minute
    (0, 23, {}),            # hour
    (1, 31, {}),            # day of month
    (1, 12, _MONTH_NAMES),  # month
    (0, 6, _DAY_NAMES),     # day of week (0 = Sunday; 7 accepted as Sunday)
]

_DESCRIPTORS = {
    "@yearly": "0 0 1 1 *",
    "@annually": "0 0 1 1 *",
    "@monthly": "0 0 1 * *",
    "@weekly": "0 0 * * 0",
    "@daily": "0 0 * * *",
    "@midnight": "0 0 * * *",
    "@hourly": "0 * * * *",
}

_GO_DURATION_RE = re.compile(r"(\d+(?:\.\d*)?|\.\d+)(ns|us|µs|μs|ms|s|m|h)")
_GO_UNIT_SECONDS = {
    "ns": 1e-9, "us": 1e-6, "µs": 1e-6, "μs": 1e-6, "ms": 1e-3,
    "s": 1.0, "m": 60.0, "h": 3600.0,
}


def parse_go_duration(s: str) -> float:
    """Parse a Go ``time.ParseDuration`` string ("1h2m3.5s", "300ms") into
    seconds. Used by ``@every <duration>`` descriptors."""
    orig = s
    s = s.strip()
    sign = 1.0
    if s.startswith(("+", "-")):
        sign = -1.0 if s[0] == "-" else 1.0
        s = s[1:]
    if s == "0":
        return 0.0
    if not s:
        raise CronParseError(f"invalid duration: {orig!r}")
    total = 0.0
    pos = 0
    for m in _GO_DURATION_RE.finditer(s):
        if m.start() != pos:
            raise CronParseError(f"invalid duration: {orig!r}")
        total += float(m.group(1)) * _GO_UNIT_SECONDS[m.group(2)]
        pos = m.end()
    if pos != len(s):
        raise CronParseError(f"invalid duration: {orig!r}")
    return sign * total


def _parse_value(tok: str, lo: int, hi: int, names: dict, field_hi: int) -> int:
    t = tok.lower()
    if t in names:
        return names[t]
    try:
        v = int(tok)
    except ValueError:
        raise CronParseError(f"invalid value {tok!r}")
    if field_hi == 6 and v == 7:  # classic cron: 7 == Sunday
        v = 0
    if not (lo <= v <= hi):
        raise CronParseError(f"value {v} out of range [{lo},{hi}]")
    return v


def _parse_field(field: str, lo: int, hi: int, names: dict) -> Tuple[FrozenSet[int], bool]:
    """Returns (allowed values, is_star) where is_star marks an unrestricted
    field (``*`` or ``?`` with no step)."""
    allowed = set()
    is_star = False
    for part in field.split(","):
        if not part:
            raise CronParseError(f"empty list item in {field!r}")
        step = 1
        if "/" in part:
            rng, _, step_s = part.partition("/")
            try:
                step = int(step_s)
            except ValueError:
                raise CronParseError(f"invalid step {step_s!r}")
            if step <= 0:
                raise CronParseError(f"invalid step {step}")
        else:
            rng = part
        if rng in ("*", "?"):
            start, end = lo, hi
            if "/" not in part and len(field.split(",")) == 1:
                is_star = True
        elif "-" in rng:
            a, _, b = rng.partition("-")
            start = _parse_value(a, lo, hi, names, hi)
            end = _parse_value(b, lo, hi, names, hi)
            if end < start:
                raise CronParseError(f"inverted range {rng!r}")
        else:
            start = _parse_value(rng, lo, hi, names, hi)
            # "a/n" means a..max by n; bare "a" means just a
            end = hi if "/" in part else start
        allowed.update(range(start, end + 1, step))
    return frozenset(allowed), is_star


@dataclass(frozen=True)
class Schedule:
    """A parsed schedule; ``next(after)`` returns the next activation time."""

    minutes: FrozenSet[int]
    hours: FrozenSet[int]
    dom: FrozenSet[int]
    months: FrozenSet[int]
    dow: FrozenSet[int]
    dom_star: bool
    dow_star: bool
    every: Optional[float] = None  # seconds, for @every schedules

    def _day_matches(self, t: datetime) -> bool:
        dom_ok = t.day in self.dom
        dow_ok = ((t.weekday() + 1) % 7) in self.dow  # python Mon=0 → cron Sun=0
        if self.dom_star and self.dow_star:
            return True
        if self.dom_star:
            return dow_ok
        if self.dow_star:
            return dom_ok
        return dom_ok or dow_ok  # classic union rule

    def next(self, after: datetime) -> datetime:
        """Next activation strictly after ``after`` (robfig Schedule.Next)."""
        if self.every is not None:
            # @every d: constant interval from 'after', truncated to seconds
            return after + timedelta(seconds=self.every)
        t = after.replace(second=0, microsecond=0) + timedelta(minutes=1)
        limit = t + timedelta(days=5 * 366)
        while t < limit:
            if t.month not in self.months:
                # advance to the 1st of the next month
                if t.month == 12:
                    t = t.replace(year=t.year + 1, month=1, day=1, hour=0, minute=0)
                else:
                    t = t.replace(month=t.month + 1, day=1, hour=0, minute=0)
                continue
            if not self._day_matches(t):
                t = (t + timedelta(days=1)).replace(hour=0, minute=0)
                continue
            if t.hour not in self.hours:
                t = (t + timedelta(hours=1)).replace(minute=0)
                continue
            if t.minute not in self.minutes:
                t = t + timedelta(minutes=1)
                continue
            return t
        raise CronParseError("no activation time within five years")


def parse_standard(spec: str) -> Schedule:
    """Parse a standard 5-field cron expression or descriptor
    (robfig/cron ParseStandard equivalent)."""
    spec = spec.strip()
    if not spec:
        raise CronParseError("empty spec string")
    if spec.startswith("@every "):
        secs = parse_go_duration(spec[len("@every "):])
        if secs <= 0:
            raise CronParseError(f"@every interval must be positive: {spec!r}")
        return Schedule(
            minutes=frozenset(), hours=frozenset(), dom=frozenset(),
            months=frozenset(), dow=frozenset(), dom_star=True, dow_star=True,
            every=secs,
        )
    if spec.startswith("@"):
        try:
            spec = _DESCRIPTORS[spec.lower()]
        except KeyError:
            raise CronParseError(f"unrecognized descriptor: {spec!r}")
    fields = spec.split()
    if len(fields) != 5:
        raise CronParseError(
            f"expected exactly 5 fields, found {len(fields)}: {spec!r}"
        )
    parsed = []
    for f, (lo, hi, names) in zip(fields, _FIELD_BOUNDS):
        parsed.append(_parse_field(f, lo, hi, names))
    (mins, _), (hrs, _), (dom, dom_star), (months, _), (dow, dow_star) = parsed
    return Schedule(
        minutes=mins, hours=hrs, dom=dom, months=months, dow=dow,
        dom_star=dom_star, dow_star=dow_star,
    )


def seconds_until_next(spec: str, now: Optional[datetime] = None) -> int:
    """The reference's cron→RepeatAfterSec derivation:
    ``int(next - now) + 1`` seconds, the +1 compensating integer truncation
    (healthcheck_controller.go:251-263)."""
    if now is None:
        # local time, like robfig's default Schedule.Next(time.Now()) — an
        # operator writing "0 9 * * *" means 9am on the controller's clock
        now = datetime.now().astimezone()
    sched = parse_standard(spec)
    delta = (sched.next(now) - now).total_seconds()
    return int(delta) + 1
