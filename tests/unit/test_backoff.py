"""Inverse-exponential backoff tests
(reference: healthcheck_controller.go:575-632 + keikoproj/inverse-exp-backoff)."""
import asyncio

import pytest

from active_monitor_amd.engine.backoff import (
    IEBTimeoutError,
    InverseExponentialBackoff,
    compute_backoff_params,
)


class FakeClock:
    def __init__(self):
        self.t = 0.0
        self.sleeps = []

    def __call__(self):
        return self.t

    async def sleep(self, d):
        self.sleeps.append(d)
        self.t += d


def drive(ieb, n):
    async def go():
        for _ in range(n):
            await ieb.next()
    asyncio.run(go())


def test_intervals_decay_from_max_to_min():
    clk = FakeClock()
    ieb = InverseExponentialBackoff(8, 1, 1000, 0.5, sleep=clk.sleep, clock=clk)
    drive(ieb, 5)
    assert clk.sleeps == [8, 4, 2, 1, 1]  # decays by factor, floors at min


def test_timeout_raises():
    clk = FakeClock()
    ieb = InverseExponentialBackoff(4, 1, 10, 0.5, sleep=clk.sleep, clock=clk)
    with pytest.raises(IEBTimeoutError):
        drive(ieb, 50)
    assert clk.t == pytest.approx(10)  # sleeps capped at remaining budget


def test_sleep_never_overshoots_deadline():
    clk = FakeClock()
    ieb = InverseExponentialBackoff(100, 1, 30, 0.5, sleep=clk.sleep, clock=clk)
    with pytest.raises(IEBTimeoutError):
        drive(ieb, 10)
    assert clk.t <= 30 + 1e-9


def test_param_validation():
    with pytest.raises(ValueError):
        InverseExponentialBackoff(0, 1, 10, 0.5)
    with pytest.raises(ValueError):
        InverseExponentialBackoff(10, 20, 10, 0.5)
    with pytest.raises(ValueError):
        InverseExponentialBackoff(10, 1, 10, 1.5)
    with pytest.raises(ValueError):
        InverseExponentialBackoff(10, 1, 0, 0.5)


# --- compute_backoff_params: the reference's defaulting matrix
# (healthcheck_controller.go:575-605; unit-test parity with
#  healthcheck_controller_unit_test.go:679-753) -----------------------------

def test_backoff_defaults_from_timeout():
    mx, mn, f, to = compute_backoff_params(0, 0, "", 120)
    assert (mx, mn, f, to) == (60.0, 2.0, 0.5, 120.0)


def test_backoff_floors_at_one_second():
    mx, mn, f, to = compute_backoff_params(0, 0, "", 0)
    assert (mx, mn) == (1.0, 1.0)
    mx, mn, _, _ = compute_backoff_params(0, 0, "", 30)  # 30/60 = 0 → floor 1
    assert (mx, mn) == (15.0, 1.0)


def test_backoff_explicit_values_not_floored():
    mx, mn, f, to = compute_backoff_params(45, 3, "0.7", 120)
    assert (mx, mn, f, to) == (45.0, 3.0, 0.7, 120.0)


def test_backoff_factor_parse_error_defaults():
    _, _, f, _ = compute_backoff_params(0, 0, "not-a-float", 120)
    assert f == 0.5
