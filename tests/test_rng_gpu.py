"""GPU bulk-sampling kernel tests (ziggurat normal/exponential on gfx950)."""
import math

import numpy as np
import pytest

import cimba_amd as ca

pytestmark = pytest.mark.gpu


def test_gpu_std_normal_moments_bulk():
    # 2^28 samples reduced on-device
    r = ca._C.rng_moments_gpu("std_normal", 0.0, 1 << 28, 1234, 0)
    n = r["n"]
    assert n == float(1 << 28)
    assert abs(r["mean"]) < 6.0 / math.sqrt(n)
    assert abs(r["var"] - 1.0) < 6.0 * math.sqrt(2.0 / n)
    skew = r["s3"] / n  # E[x^3] ~ 0
    kurt = r["s4"] / n  # E[x^4] ~ 3
    assert abs(skew) < 0.01
    assert abs(kurt - 3.0) < 0.02
    assert r["min"] < -5.0 and r["max"] > 5.0  # tail reached at this n


def test_gpu_std_exponential_moments_bulk():
    r = ca._C.rng_moments_gpu("std_exponential", 0.0, 1 << 28, 77, 0)
    n = r["n"]
    assert abs(r["mean"] - 1.0) < 6.0 / math.sqrt(n)
    assert abs(r["var"] - 1.0) < 0.01
    assert r["min"] >= 0.0
    assert r["max"] > 12.0  # deep tail via iterated offset


def test_gpu_samples_match_host_streams():
    # same per-lane seeding formula as the host sampler uses internally is
    # not required; instead check distribution agreement host vs device
    g = ca._C.rng_sample_gpu("std_normal", 0.0, 1 << 20, 99, 0)["samples"]
    h = ca.rng_sample("std_normal", [], 1 << 20, 99)
    for arr in (g, h):
        assert abs(arr.mean()) < 0.01
        assert abs(arr.std() - 1.0) < 0.01
    # KS-style coarse quantile agreement
    qg = np.quantile(g, [0.01, 0.25, 0.5, 0.75, 0.99])
    qh = np.quantile(h, [0.01, 0.25, 0.5, 0.75, 0.99])
    assert np.abs(qg - qh).max() < 0.02


def test_gpu_gamma_poisson():
    r = ca._C.rng_moments_gpu("std_gamma", 2.5, 1 << 26, 5, 0)
    assert abs(r["mean"] - 2.5) < 0.01
    assert abs(r["var"] - 2.5) < 0.02
    p = ca._C.rng_moments_gpu("poisson", 12.0, 1 << 24, 6, 0)
    assert abs(p["mean"] - 12.0) < 0.05
    assert abs(p["var"] - 12.0) < 0.1


def test_gpu_sampling_throughput_sane():
    r = ca._C.rng_moments_gpu("std_normal", 0.0, 1 << 28, 1, 0)
    # hand-written ziggurat on 256 CUs: should far exceed 10 G samples/s
    assert r["gsamples_per_sec"] > 10.0, r["gsamples_per_sec"]
