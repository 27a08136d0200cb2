"""DataSummary / WtdSummary vs numpy — counterpart of reference
test/test_data.c (moment/summary checks)."""
import numpy as np

import cimba_amd as ca


def np_summary(x):
    m = x.mean()
    d = x - m
    return dict(n=len(x), mean=m, var=x.var(ddof=1),
                skew=float((d ** 3).mean() / x.std(ddof=0) ** 3),
                kurt=float((d ** 4).mean() / x.var(ddof=0) ** 2 - 3.0))


def test_datasummary_moments():
    rng = np.random.default_rng(7)
    x = rng.gamma(2.0, 3.0, size=5000)
    s = ca.DataSummary()
    for v in x:
        s.add(v)
    ref = np_summary(x)
    assert s.count() == ref["n"]
    assert abs(s.mean() - ref["mean"]) < 1e-9
    assert abs(s.variance() - ref["var"]) < 1e-9
    assert abs(s.skewness() - ref["skew"]) < 1e-6
    assert abs(s.kurtosis() - ref["kurt"]) < 1e-6
    assert s.minimum() == x.min() and s.maximum() == x.max()


def test_datasummary_merge_exact():
    rng = np.random.default_rng(11)
    x = rng.normal(5.0, 2.0, size=4000)
    whole = ca.DataSummary()
    for v in x:
        whole.add(v)
    # merge of 4 shards must match the single-pass summary
    merged = ca.DataSummary()
    for part in np.split(x, 4):
        s = ca.DataSummary()
        for v in part:
            s.add(v)
        merged.merge(s)
    assert abs(merged.mean() - whole.mean()) < 1e-12
    assert abs(merged.variance() - whole.variance()) < 1e-9
    assert abs(merged.skewness() - whole.skewness()) < 1e-6
    assert abs(merged.kurtosis() - whole.kurtosis()) < 1e-6


def test_datasummary_raw_roundtrip():
    s = ca.DataSummary()
    for v in (1.0, 2.0, 7.5):
        s.add(v)
    t = ca.DataSummary.from_raw(*s.raw())
    assert t.mean() == s.mean() and t.variance() == s.variance()


def test_wtdsummary():
    # piecewise-constant signal: value v held for duration w
    vals = np.array([1.0, 4.0, 2.0, 8.0])
    durs = np.array([2.0, 1.0, 3.0, 0.5])
    s = ca.WtdSummary()
    for v, w in zip(vals, durs):
        s.add(v, w)
    wmean = (vals * durs).sum() / durs.sum()
    wvar = (durs * (vals - wmean) ** 2).sum() / durs.sum()
    assert abs(s.mean() - wmean) < 1e-12
    assert abs(s.variance() - wvar) < 1e-12
    assert s.minimum() == 1.0 and s.maximum() == 8.0


def test_wtdsummary_merge():
    rng = np.random.default_rng(3)
    vals = rng.normal(size=1000)
    durs = rng.random(1000) + 0.1
    whole = ca.WtdSummary()
    for v, w in zip(vals, durs):
        whole.add(v, w)
    merged = ca.WtdSummary()
    for lo in range(0, 1000, 250):
        s = ca.WtdSummary()
        for v, w in zip(vals[lo:lo + 250], durs[lo:lo + 250]):
            s.add(v, w)
        merged.merge(s)
    assert abs(merged.mean() - whole.mean()) < 1e-12
    assert abs(merged.variance() - whole.variance()) < 1e-10
