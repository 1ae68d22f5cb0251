"""Cron parser tests — robfig/cron ParseStandard semantics
(reference usage: healthcheck_controller.go:251-263)."""
from datetime import datetime, timezone

import pytest

from active_monitor_amd.engine.cronx import (
    CronParseError,
    parse_go_duration,
    parse_standard,
    seconds_until_next,
)


def dt(*args):
    return datetime(*args, tzinfo=timezone.utc)


def test_every_minute():
    s = parse_standard("* * * * *")
    assert s.next(dt(2026, 1, 1, 12, 30, 15)) == dt(2026, 1, 1, 12, 31)


def test_specific_minute_hour():
    s = parse_standard("30 14 * * *")
    assert s.next(dt(2026, 1, 1, 12, 0)) == dt(2026, 1, 1, 14, 30)
    assert s.next(dt(2026, 1, 1, 15, 0)) == dt(2026, 1, 2, 14, 30)


def test_step_and_range():
    s = parse_standard("*/15 9-17 * * *")
    assert s.next(dt(2026, 1, 1, 9, 16)) == dt(2026, 1, 1, 9, 30)
    assert s.next(dt(2026, 1, 1, 17, 46)) == dt(2026, 1, 2, 9, 0)


def test_list_and_names():
    s = parse_standard("0 0 * JAN,jul *")
    assert s.next(dt(2026, 2, 1)) == dt(2026, 7, 1)
    assert s.next(dt(2026, 8, 1)) == dt(2027, 1, 1)


def test_day_of_week_names():
    s = parse_standard("0 9 * * MON-FRI")
    # 2026-01-03 is a Saturday
    assert s.next(dt(2026, 1, 3, 10, 0)) == dt(2026, 1, 5, 9, 0)


def test_dow_sunday_as_0_and_7():
    # 2026-01-04 is a Sunday
    assert parse_standard("0 0 * * 0").next(dt(2026, 1, 1)) == dt(2026, 1, 4)
    assert parse_standard("0 0 * * 7").next(dt(2026, 1, 1)) == dt(2026, 1, 4)


def test_dom_dow_union_rule():
    # both restricted: fire if EITHER matches (classic cron rule)
    s = parse_standard("0 0 15 * MON")
    # from Jan 1 2026 (Thu): next Monday is Jan 5, before the 15th
    assert s.next(dt(2026, 1, 1)) == dt(2026, 1, 5)
    # from Jan 13 (Tue): the 15th (Thu) comes before next Monday (19th)
    assert s.next(dt(2026, 1, 13)) == dt(2026, 1, 15)


def test_descriptors():
    assert parse_standard("@hourly").next(dt(2026, 1, 1, 5, 30)) == dt(2026, 1, 1, 6, 0)
    assert parse_standard("@daily").next(dt(2026, 1, 1, 5, 30)) == dt(2026, 1, 2, 0, 0)
    assert parse_standard("@midnight").next(dt(2026, 1, 1, 5, 30)) == dt(2026, 1, 2, 0, 0)
    assert parse_standard("@weekly").next(dt(2026, 1, 1)) == dt(2026, 1, 4)  # Sunday
    assert parse_standard("@monthly").next(dt(2026, 1, 15)) == dt(2026, 2, 1)
    assert parse_standard("@yearly").next(dt(2026, 3, 1)) == dt(2027, 1, 1)
    assert parse_standard("@annually").next(dt(2026, 3, 1)) == dt(2027, 1, 1)


def test_every_descriptor():
    s = parse_standard("@every 1m")
    t = dt(2026, 1, 1, 0, 0, 30)
    assert (s.next(t) - t).total_seconds() == 60
    s3 = parse_standard("@every 3s")
    assert (s3.next(t) - t).total_seconds() == 3
    s_mixed = parse_standard("@every 1h30m")
    assert (s_mixed.next(t) - t).total_seconds() == 5400


def test_go_duration_parse():
    assert parse_go_duration("300ms") == pytest.approx(0.3)
    assert parse_go_duration("1h2m3.5s") == pytest.approx(3723.5)
    assert parse_go_duration("1m") == 60
    with pytest.raises(CronParseError):
        parse_go_duration("5 minutes")
    with pytest.raises(CronParseError):
        parse_go_duration("")


@pytest.mark.parametrize("bad", [
    "", "* * * *", "* * * * * *", "61 * * * *", "* 25 * * *", "* * 32 * *",
    "* * * 13 *", "* * * * 8", "@fortnightly", "a b c d e", "1-0 * * * *",
    "*/0 * * * *",
])
def test_invalid_specs_raise(bad):
    with pytest.raises(CronParseError):
        parse_standard(bad)


def test_seconds_until_next_plus_one_truncation():
    # reference derives RepeatAfterSec = int(next-now)+1 (controller :251-263)
    now = dt(2026, 1, 1, 12, 0, 30)
    # next minute boundary is 30s away → int(30)+1 = 31
    assert seconds_until_next("* * * * *", now) == 31
    assert seconds_until_next("@every 10s", now) == 11


def test_february_and_leap_year():
    s = parse_standard("0 0 29 2 *")
    assert s.next(dt(2026, 1, 1)).date().isoformat() == "2028-02-29"
