"""Property-based tests (hypothesis): wire-format round-trips and cron
invariants hold for arbitrary inputs, not just the fixtures."""
from datetime import datetime, timedelta, timezone

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from active_monitor_amd.api import HealthCheckSpec, HealthCheckStatus  # noqa: E402
from active_monitor_amd.engine.cronx import parse_standard  # noqa: E402

# -- status round-trip ------------------------------------------------------

time_str = st.datetimes(
    min_value=datetime(2000, 1, 1), max_value=datetime(2100, 1, 1)
).map(lambda d: d.strftime("%Y-%m-%dT%H:%M:%SZ"))

status_strategy = st.builds(
    HealthCheckStatus,
    error_message=st.text(max_size=40),
    remedy_error_message=st.text(max_size=40),
    started_at=st.none() | time_str,
    finished_at=st.none() | time_str,
    last_failed_at=st.none() | time_str,
    remedy_started_at=st.none() | time_str,
    remedy_finished_at=st.none() | time_str,
    remedy_last_failed_at=st.none() | time_str,
    last_failed_workflow=st.text(max_size=30),
    last_successful_workflow=st.text(max_size=30),
    success_count=st.integers(min_value=0, max_value=10**6),
    failed_count=st.integers(min_value=0, max_value=10**6),
    remedy_success_count=st.integers(min_value=0, max_value=10**6),
    remedy_failed_count=st.integers(min_value=0, max_value=10**6),
    remedy_total_runs=st.integers(min_value=0, max_value=10**6),
    total_healthcheck_runs=st.integers(min_value=0, max_value=10**6),
    status=st.sampled_from(["", "Succeeded", "Failed", "Stopped"]),
    remedy_status=st.text(max_size=60),
)


@settings(max_examples=200, deadline=None)
@given(status_strategy)
def test_status_round_trip_lossless(status):
    d = status.to_dict()
    back = HealthCheckStatus.from_dict(d)
    # omitempty drops zero values; they decode back to defaults → equal
    assert back == status
    assert back.to_dict() == d


spec_strategy = st.fixed_dictionaries({
    "repeatAfterSec": st.integers(min_value=0, max_value=10**6),
    "description": st.text(max_size=30),
    "level": st.sampled_from(["", "cluster", "namespace"]),
    "workflow": st.fixed_dictionaries({
        "generateName": st.text(
            alphabet="abcdefghijklmnopqrstuvwxyz-", min_size=1, max_size=20),
        "workflowtimeout": st.integers(min_value=0, max_value=86400),
    }),
    "backoffFactor": st.sampled_from(["", "0.5", "0.9", "not-a-number"]),
    "backoffMax": st.integers(min_value=0, max_value=3600),
    "backoffMin": st.integers(min_value=0, max_value=3600),
    "remedyRunsLimit": st.integers(min_value=0, max_value=100),
    "remedyResetInterval": st.integers(min_value=0, max_value=86400),
})


@settings(max_examples=200, deadline=None)
@given(spec_strategy)
def test_spec_round_trip_stable(d):
    spec = HealthCheckSpec.from_dict(d)
    once = spec.to_dict()
    # serialization is a fixed point after one round
    assert HealthCheckSpec.from_dict(once).to_dict() == once


# -- cron invariants --------------------------------------------------------

minute = st.integers(0, 59)
hour = st.integers(0, 23)
dom = st.integers(1, 31)
month = st.integers(1, 12)
dow = st.integers(0, 6)


@st.composite
def cron_exprs(draw):
    def field(vals, lo, hi):
        kind = draw(st.sampled_from(["star", "value", "range", "step", "list"]))
        if kind == "star":
            return "*"
        if kind == "value":
            return str(draw(vals))
        if kind == "range":
            a, b = sorted((draw(vals), draw(vals)))
            return f"{a}-{b}"
        if kind == "step":
            return f"*/{draw(st.integers(1, hi - lo + 1))}"
        return ",".join(str(draw(vals)) for _ in range(draw(st.integers(1, 3))))

    return " ".join([
        field(minute, 0, 59), field(hour, 0, 23), field(dom, 1, 31),
        field(month, 1, 12), field(dow, 0, 6),
    ])


@settings(max_examples=150, deadline=None)
@given(cron_exprs(), st.datetimes(min_value=datetime(2020, 1, 1),
                                  max_value=datetime(2030, 1, 1)))
def test_cron_next_is_strictly_future_and_matching(expr, now):
    now = now.replace(tzinfo=timezone.utc)
    try:
        sched = parse_standard(expr)
        nxt = sched.next(now)
    except Exception as e:
        # impossible dom/month combos legitimately exhaust the search window
        assert "no activation time" in str(e)
        return
    assert nxt > now
    assert nxt.second == 0 and nxt.microsecond == 0
    assert nxt.minute in sched.minutes
    assert nxt.hour in sched.hours
    assert nxt.month in sched.months
    assert sched._day_matches(nxt)
    # idempotence: asking again from just before nxt returns nxt
    again = sched.next(nxt - timedelta(seconds=1))
    assert again == nxt


@settings(max_examples=200, deadline=None)
@given(
    st.integers(min_value=0, max_value=10**5),
    st.integers(min_value=0, max_value=10**5),
    st.sampled_from(["", "0.1", "0.5", "0.99", "garbage"]),
    st.integers(min_value=0, max_value=10**6),
)
def test_backoff_params_invariants(bmax, bmin, factor, timeout):
    from active_monitor_amd.engine.backoff import compute_backoff_params

    mx, mn, f, to = compute_backoff_params(bmax, bmin, factor, timeout)
    # defaults floor at 1s; explicit values pass through verbatim
    assert mx == (bmax if bmax else max(timeout // 2, 1))
    assert mn == (bmin if bmin else max(timeout // 60, 1))
    assert 0 < f  # factor is 0.5 or the parsed value
    assert to == timeout


@settings(max_examples=150, deadline=None)
@given(st.dictionaries(
    st.text(alphabet="abc-.x/", min_size=1, max_size=12),
    st.text(alphabet="abcXYZ0-_.", max_size=12),
    max_size=4,
))
def test_label_selector_roundtrip(labels):
    from active_monitor_amd.kube.memory import parse_label_selector

    if not labels:
        assert parse_label_selector("") == {}
        return
    sel = ",".join(f"{k}={v}" for k, v in labels.items())
    assert parse_label_selector(sel) == labels


# ---------------------------------------------------------------------------
# RFC 7386 merge patch: apply(a, diff(a, b)) == b for any JSON objects
# ---------------------------------------------------------------------------

_scalars = st.one_of(st.integers(-5, 5), st.text(max_size=4), st.booleans())
_json_obj = st.recursive(
    st.dictionaries(st.text(min_size=1, max_size=4), _scalars, max_size=4),
    lambda children: st.dictionaries(
        st.text(min_size=1, max_size=4),
        st.one_of(_scalars, children, st.lists(_scalars, max_size=3)),
        max_size=4,
    ),
    max_leaves=12,
)


def _diff(a, b):
    """Build the RFC 7386 merge patch transforming a into b."""
    patch = {}
    for k in a:
        if k not in b:
            patch[k] = None
    for k, vb in b.items():
        va = a.get(k)
        if isinstance(va, dict) and isinstance(vb, dict):
            sub = _diff(va, vb)
            if sub:
                patch[k] = sub
        elif va != vb or k not in a:
            patch[k] = vb
    return patch


@given(a=_json_obj, b=_json_obj)
@settings(max_examples=200, deadline=None)
def test_merge_patch_roundtrip(a, b):
    """MemoryApiServer._merge implements RFC 7386: applying diff(a,b) to a
    yields b (None values cannot round-trip by the RFC's own design; the
    generator produces none)."""
    import copy

    from active_monitor_amd.kube.memory import MemoryApiServer

    base = copy.deepcopy(a)
    MemoryApiServer._merge(base, _diff(a, b))
    assert base == b
