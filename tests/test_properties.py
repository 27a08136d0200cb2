"""Property-based tests (hypothesis) for the statistics layer and RNG
plumbing — the invariants the distributed reduction and the GPU result
merges rely on (counterpart of the reference's test_statistics.c checks,
generalized over random inputs)."""
import math

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import cimba_amd as ca

finite = st.floats(min_value=-1e12, max_value=1e12,
                   allow_nan=False, allow_infinity=False)
pos = st.floats(min_value=1e-6, max_value=1e6,
                allow_nan=False, allow_infinity=False)


def summarize(xs):
    ds = ca.DataSummary()
    for x in xs:
        ds.add(x)
    return ds


@settings(max_examples=40, deadline=None)
@given(st.lists(finite, min_size=2, max_size=200))
def test_datasummary_matches_numpy(xs):
    ds = summarize(xs)
    a = np.asarray(xs, dtype=np.float64)
    assert ds.count() == len(xs)
    assert ds.minimum() == a.min() and ds.maximum() == a.max()
    scale = max(1.0, abs(a.mean()))
    assert abs(ds.mean() - a.mean()) < 1e-9 * scale
    # Welford vs two-pass variance: agree to fp accuracy relative to
    # the data's spread (catastrophic-cancellation-free on both sides)
    v = a.var(ddof=1)
    assert abs(ds.variance() - v) <= 1e-6 * max(1.0, v)


@settings(max_examples=40, deadline=None)
@given(st.lists(finite, min_size=1, max_size=120),
       st.lists(finite, min_size=1, max_size=120))
def test_datasummary_merge_equals_concat(xs, ys):
    # merge(A, B) must equal summarize(A ++ B) — this is what makes the
    # RCCL rank reduction exact (parallel/experiment.py)
    m = summarize(xs)
    m.merge(summarize(ys))
    w = summarize(xs + ys)
    assert m.count() == w.count()
    assert m.minimum() == w.minimum() and m.maximum() == w.maximum()
    assert math.isclose(m.mean(), w.mean(), rel_tol=1e-12, abs_tol=1e-9)
    if w.count() > 1:
        assert math.isclose(m.variance(), w.variance(), rel_tol=1e-9,
                            abs_tol=1e-9 * max(1.0, abs(w.mean())) ** 2)


@settings(max_examples=40, deadline=None)
@given(st.lists(st.tuples(finite, pos), min_size=2, max_size=150))
def test_wtdsummary_matches_numpy(pairs):
    ws = ca.WtdSummary()
    for x, w in pairs:
        ws.add(x, w)
    xs = np.array([p[0] for p in pairs])
    w = np.array([p[1] for p in pairs])
    mean = (xs * w).sum() / w.sum()
    assert math.isclose(ws.sumw(), w.sum(), rel_tol=1e-12)
    # error model: the incremental (West/Pébay) update rounds each step
    # relative to the INTERMEDIATE running mean, which extreme weight
    # ratios can make as large as max|x| — so the absolute error is
    # bounded by ~eps * max|x| per step, not by the final mean
    scale = float(np.abs(xs).max())
    assert abs(ws.mean() - mean) < 1e-6 * max(1.0, abs(mean)) + 1e-14 * len(pairs) * scale


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(finite, pos), min_size=1, max_size=80),
       st.lists(st.tuples(finite, pos), min_size=1, max_size=80))
def test_wtdsummary_merge_equals_concat(a, b):
    wa = ca.WtdSummary()
    for x, w in a:
        wa.add(x, w)
    wb = ca.WtdSummary()
    for x, w in b:
        wb.add(x, w)
    wa.merge(wb)
    ww = ca.WtdSummary()
    for x, w in a + b:
        ww.add(x, w)
    assert math.isclose(wa.sumw(), ww.sumw(), rel_tol=1e-12)
    # error model (same as the numpy test above): the incremental update
    # rounds against the INTERMEDIATE running mean, which adversarial
    # weight ratios can push to max|x| — bound by eps * max|x| per step,
    # not by the final mean
    scale = max(abs(x) for x, _ in a + b)
    n = len(a) + len(b)
    assert abs(wa.mean() - ww.mean()) < (
        1e-8 * max(1.0, abs(ww.mean())) + 1e-14 * n * scale)


@settings(max_examples=30, deadline=None)
@given(st.lists(finite, min_size=1, max_size=300))
def test_dataset_order_statistics(xs):
    d = ca._C.Dataset()
    for x in xs:
        d.add(x)
    a = np.sort(np.asarray(xs))
    assert d.size() == len(xs)
    assert d.median() == float(np.median(a)) or \
        abs(d.median() - float(np.median(a))) < 1e-9 * max(1.0, abs(a).max())
    assert d.quantile(0.0) == a[0] and d.quantile(1.0) == a[-1]
    q = d.quantile(0.25)
    assert a[0] - 1e-12 <= q <= a[-1] + 1e-12
    hist = d.histogram(8)
    assert sum(hist) == len(xs)


@settings(max_examples=25, deadline=None)
@given(st.integers(min_value=0, max_value=2**64 - 1))
def test_fmix64_invertible_nonzero(x):
    # fmix64 is a bijection: distinct inputs produce distinct outputs
    # (seed-derivation relies on no stream collisions across trials)
    y = ca.fmix64(x)
    if x != 0:
        assert y != 0 or x == 0
    assert ca.fmix64(x) == y  # pure


@settings(max_examples=20, deadline=None)
@given(st.integers(min_value=1, max_value=2**63), st.integers(0, 10_000))
def test_trial_seed_distinct(master, i):
    a = ca.trial_seed(master, i)
    b = ca.trial_seed(master, i + 1)
    c = ca.trial_seed(master ^ 1, i)
    assert a != b and a != c
