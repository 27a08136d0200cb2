"""Terrain + line-of-sight masking tests — the MI355X equivalent of the
reference's AWACS terrain stack (reference tutorial/tut_5_2.cu
terrain_generate_kernel/terrain_stats_kernel/prime_altitudes_kernel and
the raymarch LOS masking).  The host build (same CMB_HD code) is the
fp32 numerics reference for the GPU kernels."""
import numpy as np
import pytest

import cimba_amd as ca


def _host(cols=96, rows=80, **kw):
    kw.setdefault("base", 100.0)
    kw.setdefault("amp", 500.0)
    kw.setdefault("octaves", 5)
    kw.setdefault("seed", 11)
    return ca._C.terrain_host(cols, rows, **kw)


def test_terrain_deterministic_and_stats():
    a = _host()
    b = _host()
    h = a["heights"]
    assert np.array_equal(h, b["heights"])  # same seed -> bitwise equal
    mean, var, mn, mx = a["stats"]
    assert abs(h.mean() - mean) < 1e-6
    assert abs(float(np.var(h)) - var) < 1e-3
    assert mn == h.min() and mx == h.max()
    # ridged fbm in [0,~1] scaled by amp on top of base
    assert 100.0 <= mn < mx <= 600.0 + 1e-3
    c = _host(seed=12)
    assert not np.array_equal(h, c["heights"])  # seed moves the terrain


def test_terrain_sample_bilinear():
    r = _host(xs=[10.0, 10.5, 11.0], ys=[20.0, 20.0, 20.0])
    h = r["heights"]
    s0, smid, s1 = r["samples"]
    assert s0 == h[20, 10] and s1 == h[20, 11]
    lo, hi = min(s0, s1), max(s0, s1)
    assert lo - 1e-4 <= smid <= hi + 1e-4  # on the segment between texels
    # edge clamp: far outside returns the corner texel
    rc = _host(xs=[-50.0], ys=[-50.0])
    assert rc["samples"][0] == h[0, 0]


def test_terrain_los():
    # rays far above the peaks are clear; rays ending underground are
    # blocked; a grazing ray across rough terrain is blocked
    q = [0, 0, 10000, 95, 79, 10000,      # high above: clear
         0, 0, -10000, 95, 79, -10000,    # underground: blocked
         0, 0, 300, 95, 79, 300]          # grazing mid-heights
    r = _host(queries=q, nsteps=256)
    vis = r["vis"]
    assert vis[0] == 1 and vis[1] == 0
    h = r["heights"]
    # the grazing verdict must agree with a numpy re-march
    t = (np.arange(256) + 1) / 257.0
    xs, ys, zs = 95 * t, 79 * t, np.full_like(t, 300.0)
    hs = np.array([h[min(int(round(y)), 79), min(int(round(x)), 95)]
                   for x, y in zip(xs, ys)])
    # nearest-texel approximation: only sanity-check the obvious cases
    if (hs > 320).any():
        assert vis[2] == 0


@pytest.mark.gpu
def test_terrain_gpu_matches_host():
    rng = np.random.default_rng(3)
    xs = (rng.random(512) * 95).astype(np.float32)
    ys = (rng.random(512) * 79).astype(np.float32)
    q = np.empty((256, 6), dtype=np.float32)
    q[:, 0] = rng.random(256) * 95
    q[:, 1] = rng.random(256) * 79
    q[:, 2] = rng.random(256) * 700
    q[:, 3] = rng.random(256) * 95
    q[:, 4] = rng.random(256) * 79
    q[:, 5] = rng.random(256) * 700
    kw = dict(base=100.0, amp=500.0, octaves=5, seed=11,
              xs=xs.tolist(), ys=ys.tolist(),
              queries=q.reshape(-1).tolist(), nsteps=192)
    hh = ca._C.terrain_host(96, 80, **kw)
    gg = ca._C.terrain_gpu(96, 80, device=0, **kw)
    # generate + bilinear sample are deterministic fp32 -> bitwise equal
    assert hh["samples"] == gg["samples"]
    # LOS verdicts identical
    assert hh["vis"] == gg["vis"]
    # stats: different reduction order, same values to fp accuracy
    for a, b in zip(hh["stats"], gg["stats"]):
        assert abs(a - b) < 1e-6 * max(1.0, abs(a))


@pytest.mark.gpu
def test_terrain_gpu_large_grid():
    # a big on-device heightmap (the reference keeps ~14 GB resident;
    # here 4k x 4k = 64 MB is enough to exercise the grid-stride paths)
    g = ca._C.terrain_gpu(4096, 4096, base=0.0, amp=1000.0, octaves=6,
                          seed=5, queries=[0, 0, 5000, 4095, 4095, 5000],
                          nsteps=4096, device=0)
    mean, var, mn, mx = g["stats"]
    assert 0.0 <= mn < mean < mx <= 1000.0 + 1e-3
    assert var > 0.0
    assert g["vis"] == [1]  # 5 km altitude clears a 1 km terrain
