"""Dataset/Timeseries (reference cmb_dataset/cmb_timeseries) and the
logger + trial-abandon recovery path (reference cmb_logger_error ->
cimba_trial_abandon -> worker recovery)."""
import numpy as np

import cimba_amd as ca
from cimba_amd._C import Dataset, Timeseries


def test_dataset_order_stats():
    rng = np.random.default_rng(2)
    x = rng.normal(size=1001)
    d = Dataset()
    for v in x:
        d.add(v)
    assert d.size() == 1001
    assert abs(d.median() - np.median(x)) < 1e-12
    f = d.fivenum()
    assert f[0] == x.min() and f[4] == x.max()
    assert abs(f[1] - np.quantile(x, 0.25)) < 1e-9
    assert abs(f[3] - np.quantile(x, 0.75)) < 1e-9
    h = d.histogram(10)
    assert sum(h) == 1001
    s = d.summarize()
    assert abs(s.mean() - x.mean()) < 1e-12


def test_dataset_acf_pacf_ar1():
    # AR(1) with phi=0.6: ACF(k) ~ phi^k, PACF(1) ~ phi, PACF(k>1) ~ 0
    rng = np.random.default_rng(5)
    phi = 0.6
    n = 20000
    x = np.empty(n)
    x[0] = 0.0
    eps = rng.normal(size=n)
    for t in range(1, n):
        x[t] = phi * x[t - 1] + eps[t]
    d = Dataset()
    for v in x:
        d.add(v)
    acf = d.acf(5)
    for k in range(5):
        assert abs(acf[k] - phi ** (k + 1)) < 0.05, (k, acf[k])
    pacf = d.pacf(5)
    assert abs(pacf[0] - phi) < 0.05
    for k in range(1, 5):
        assert abs(pacf[k]) < 0.05


def test_dataset_merge():
    a, b = Dataset(), Dataset()
    for v in (3.0, 1.0):
        a.add(v)
    for v in (2.0, 4.0):
        b.add(v)
    a.merge(b)
    assert a.size() == 4
    assert a.median() == 2.5


def test_timeseries_weighted():
    ts = Timeseries()
    ts.add(1.0, 0.0)   # value 1 over [0, 2)
    ts.add(5.0, 2.0)   # value 5 over [2, 3)
    ts.add(2.0, 3.0)   # value 2 over [3, 6]
    s = ts.summarize(6.0)
    wmean = (1 * 2 + 5 * 1 + 2 * 3) / 6
    assert abs(s.mean() - wmean) < 1e-12
    assert ts.median(6.0) == 2.0  # time-weighted median


def test_trial_abandon_recovery():
    # scenario 14 abandons every trial via logger_error; the executive must
    # recover, run cleanup hooks, count failures, and keep going
    ca._C.logger_flags_off(ca._C.LOG_ERROR)  # quiet the expected messages
    try:
        r = ca._C.scenario_run_host(14, ntrials=6, threads=2)
    finally:
        ca._C.logger_flags_on(ca._C.LOG_ERROR)
    assert r["trials"] == 6
    assert r["abandoned"] == 6
    assert r["failed"] == 6
    assert r["cleanups"] == 6
    assert r["thread_inits"] == 2 and r["thread_exits"] == 2
    assert r["first_status"] != 0


def test_ok_scenarios_not_failed():
    r = ca._C.scenario_run_host(1, ntrials=4, threads=2)
    assert r["failed"] == 0 and r["abandoned"] == 0
