"""Statistical-quality tests for the RNG layer — the counterpart of
reference test/test_random.c (raw moments of each distribution over many
samples vs closed-form expectations, test_random.c:255-270)."""
import math

import numpy as np
import pytest

import cimba_amd as ca

N = 200_000
SEED = 0x34F05C64D7AD598F


def moments(x):
    return x.mean(), x.var(), float(((x - x.mean()) ** 3).mean())


def check_mean_var(dist, params, mean, var, seed=SEED, n=N, tol_sigma=6.0):
    x = ca.rng_sample(dist, params, n, seed)
    se_mean = math.sqrt(var / n)
    assert abs(x.mean() - mean) < tol_sigma * se_mean + 1e-12, (
        f"{dist}: mean {x.mean()} vs {mean}")
    # variance tolerance: loose CLT bound via 4th moment ~ 3 var^2
    se_var = var * math.sqrt(10.0 / n)
    assert abs(x.var() - var) < tol_sigma * se_var + 1e-12, (
        f"{dist}: var {x.var()} vs {var}")


def test_sfc64_determinism():
    a = ca.rng_sample("u64", [], 1000, 42)
    b = ca.rng_sample("u64", [], 1000, 42)
    c = ca.rng_sample("u64", [], 1000, 43)
    assert (a == b).all()
    assert (a != c).any()


def test_fmix64_known_vectors():
    # murmur3 fmix64 reference values
    assert ca.fmix64(0) == 0
    assert ca.fmix64(1) == 0xB456BCFC34C2CB2C
    assert ca.fmix64(0xDEADBEEF) == 0xD24BD59F862A1DAC


def test_uniform():
    check_mean_var("uniform", [2.0, 6.0], 4.0, 16.0 / 12.0)
    x = ca.rng_sample("uniform", [0.0, 1.0], N, SEED)
    assert x.min() >= 0.0 and x.max() < 1.0


def test_std_normal_moments():
    x = ca.rng_sample("std_normal", [], N, SEED)
    assert abs(x.mean()) < 6.0 / math.sqrt(N)
    assert abs(x.var() - 1.0) < 6.0 * math.sqrt(2.0 / N)
    # skewness ~ 0, excess kurtosis ~ 0
    z = (x - x.mean()) / x.std()
    assert abs((z ** 3).mean()) < 0.05
    assert abs((z ** 4).mean() - 3.0) < 0.12


def test_std_normal_tail():
    # ziggurat tail beyond R=3.654: P(|X|>3.654) ~ 2.58e-4
    x = ca.rng_sample("std_normal", [], 2_000_000, SEED + 7)
    frac = (np.abs(x) > 3.6541528853610088).mean()
    assert 1.0e-4 < frac < 6.0e-4
    assert np.abs(x).max() > 3.8  # tail actually samples beyond R


def test_std_exponential():
    x = ca.rng_sample("std_exponential", [], N, SEED)
    assert abs(x.mean() - 1.0) < 6.0 / math.sqrt(N)
    assert abs(x.var() - 1.0) < 0.05
    assert x.min() >= 0.0
    # tail beyond R=7.697: P ~ 4.53e-4
    y = ca.rng_sample("std_exponential", [], 2_000_000, SEED + 9)
    frac = (y > 7.6971174701310496).mean()
    assert 2.0e-4 < frac < 9.0e-4


def test_exponential_mean():
    check_mean_var("exponential", [3.0], 3.0, 9.0)


def test_lognormal():
    mu, s = 0.3, 0.5
    m = math.exp(mu + s * s / 2)
    v = (math.exp(s * s) - 1) * math.exp(2 * mu + s * s)
    check_mean_var("lognormal", [mu, s], m, v)


def test_gamma():
    for shape in (0.5, 1.0, 2.5, 9.0):
        check_mean_var("std_gamma", [shape], shape, shape)
    check_mean_var("gamma", [3.0, 2.0], 6.0, 12.0)


def test_erlang():
    check_mean_var("erlang", [4, 2.0], 2.0, 4.0 * (0.5 ** 2))


def test_beta():
    a, b = 2.0, 5.0
    m = a / (a + b)
    v = a * b / ((a + b) ** 2 * (a + b + 1))
    check_mean_var("std_beta", [a, b], m, v)


def test_weibull_pareto_rayleigh():
    k, lam = 2.0, 3.0
    m = lam * math.gamma(1 + 1 / k)
    v = lam * lam * (math.gamma(1 + 2 / k) - math.gamma(1 + 1 / k) ** 2)
    check_mean_var("weibull", [k, lam], m, v)
    a, s = 3.0, 2.0  # pareto shape 3 scale 2
    check_mean_var("pareto", [a, s], a * s / (a - 1),
                   s * s * a / ((a - 1) ** 2 * (a - 2)), tol_sigma=10.0)
    sg = 2.0
    check_mean_var("rayleigh", [sg], sg * math.sqrt(math.pi / 2),
                   (4 - math.pi) / 2 * sg * sg)


def test_triangular_pert():
    lo, mode, hi = 1.0, 3.0, 8.0
    m = (lo + mode + hi) / 3
    v = (lo*lo + mode*mode + hi*hi - lo*mode - lo*hi - mode*hi) / 18
    check_mean_var("triangular", [lo, mode, hi], m, v)
    m_pert = (lo + 4 * mode + hi) / 6
    x = ca.rng_sample("pert", [lo, mode, hi], N, SEED)
    assert abs(x.mean() - m_pert) < 0.05


def test_logistic_chisq_t():
    check_mean_var("logistic", [2.0, 0.5], 2.0, (math.pi ** 2 / 3) * 0.25)
    check_mean_var("chisquared", [5.0], 5.0, 10.0)
    df = 8.0
    check_mean_var("std_t_dist", [df], 0.0, df / (df - 2), tol_sigma=10.0)


def test_hypo_hyper_exponential():
    check_mean_var("hypoexponential", [2.0, 3.0], 5.0, 4.0 + 9.0)
    p, m1, m2 = 0.3, 1.0, 5.0
    m = p * m1 + (1 - p) * m2
    ex2 = p * 2 * m1 * m1 + (1 - p) * 2 * m2 * m2
    check_mean_var("hyperexponential", [p, m1, m2], m, ex2 - m * m)


def test_discrete():
    check_mean_var("bernoulli", [0.3], 0.3, 0.21)
    p = 0.25
    check_mean_var("geometric", [p], (1 - p) / p, (1 - p) / p ** 2)
    for lam in (3.0, 40.0):
        check_mean_var("poisson", [lam], lam, lam)
    for n, p2 in ((20, 0.3), (400, 0.4)):
        check_mean_var("binomial", [n, p2], n * p2, n * p2 * (1 - p2))
    r, pp = 5.0, 0.4
    check_mean_var("negative_binomial", [r, pp], r * (1 - pp) / pp,
                   r * (1 - pp) / pp ** 2)
    x = ca.rng_sample("discrete_uniform", [2, 11], N, SEED)
    assert x.min() == 2 and x.max() == 11
    assert abs(x.mean() - 6.5) < 0.1
    d = ca.rng_sample("dice", [6], N, SEED)
    assert d.min() == 1 and d.max() == 6


def test_discrete_nonuniform_and_alias():
    w = [1.0, 2.0, 3.0, 4.0]
    probs = np.array(w) / sum(w)
    for dist in ("discrete_nonuniform", "alias"):
        x = ca.rng_sample(dist, w, N, SEED).astype(int)
        if dist == "discrete_nonuniform":
            counts = np.bincount(x, minlength=4) / N
        else:
            counts = np.bincount(x, minlength=4) / N
        assert np.abs(counts - probs).max() < 0.01, (dist, counts)
