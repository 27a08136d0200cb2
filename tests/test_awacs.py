"""AWACS radar model (BASELINE config 5): host-path sanity + GPU parity +
MFMA beamforming numerics vs fp64 reference."""
import numpy as np
import pytest

import cimba_amd as ca


def test_awacs_host_sanity():
    r = ca._C.awacs_host(ntrials=2, duration=5.0, dwell=0.04, ntargets=200,
                         seed=7, threads=2)
    assert r["trials_ok"] == 2
    assert r["total_dwells"] == 2 * 125  # duration/dwell per trial
    assert r["total_detections"] > 0
    assert r["total_maneuvers"] > 2 * 100  # ~1 per target per 5s + resched


def test_awacs_host_deterministic():
    a = ca._C.awacs_host(ntrials=2, duration=2.0, ntargets=100, seed=3,
                         threads=1)
    b = ca._C.awacs_host(ntrials=2, duration=2.0, ntargets=100, seed=3,
                         threads=2)
    assert a["total_detections"] == b["total_detections"]
    assert a["sum_power"] == b["sum_power"]
    c = ca._C.awacs_host(ntrials=2, duration=2.0, ntargets=100, seed=4,
                         threads=1)
    assert c["total_detections"] != a["total_detections"]


@pytest.mark.gpu
def test_awacs_gpu_matches_host():
    g = ca._C.awacs_gpu(ntrials=8, duration=5.0, ntargets=512, seed=11,
                        device=0)
    h = ca._C.awacs_host(ntrials=8, duration=5.0, ntargets=512, seed=11,
                         threads=0)
    assert g["trials_ok"] == 8
    assert g["total_dwells"] == h["total_dwells"]
    # event-loop side is bit-identical (same rng stream on lane 0)
    assert g["total_maneuvers"] == h["total_maneuvers"]
    assert g["total_events"] == h["total_events"]
    # physics: MFMA fma-chain vs host mul+add differ in f32 rounding; the
    # survivor SET can also differ by beam-gate ulps (power is heavy-
    # tailed in rcs/r^-4, so a one-survivor difference moves the sum)
    assert abs(g["sum_power"] - h["sum_power"]) / h["sum_power"] < 0.05
    rel_det = abs(g["total_detections"] - h["total_detections"]) / max(
        h["total_detections"], 1)
    assert rel_det < 0.02, (g["total_detections"], h["total_detections"])


@pytest.mark.gpu
def test_awacs_mfma_beamforming_numerics():
    # device MFMA powers vs an fp64 host reference of the same formula on
    # the identical seeded target set (the required HIP-kernel-vs-plain-
    # reference numerics test)
    r = ca._C.awacs_power_check(ntargets=1000, seed=42, device=0)
    dev = np.asarray(r["device_mfma"])
    f64 = np.asarray(r["host_f64"])
    f32 = np.asarray(r["host_f32"])
    assert r["nt"] == 1000
    scale = np.abs(f64).max()
    assert scale > 0
    # f32 accumulation over K=16 elements: ~1e-6 relative class errors
    assert np.abs(dev - f64).max() / scale < 5e-5
    assert np.abs(f32 - f64).max() / scale < 5e-5
    # and the MFMA path is not silently the host path: exact zeros match
    assert np.abs(dev - f32).max() / scale < 5e-5


def test_awacs_pipeline_host_stats():
    """Full radar pipeline (terrain masking, clutter, CA-CFAR, multipath):
    triage keeps ~beamwidth/2pi of targets per dwell, mountainous terrain
    shields a large fraction, and CFAR discrimination keeps det/clear off
    the saturated 1.0 (the r01 gap: no clutter/CFAR/multipath at all)."""
    r = ca._C.awacs_host(ntrials=4, duration=10.0, ntargets=1000, seed=42,
                         threads=4)
    assert r["trials_ok"] == 4
    illum = r["total_illuminated"]
    shield = r["total_shielded"]
    det = r["total_detections"]
    clear = illum - shield
    # beam gate: ~(beamwidth + sweep)/2pi * nt per dwell ~ 8
    per_dwell = illum / r["total_dwells"]
    assert 3.0 < per_dwell < 20.0, per_dwell
    assert 0.30 < shield / illum < 0.95  # terrain really masks
    assert 0.10 < det / clear < 0.95     # CFAR really discriminates
    assert r["sum_clutter"] > 0.0
    assert det <= clear


def test_awacs_terrain_off_legacy_mode():
    """terrain=0 keeps the r01 free-space behavior (no triage counters)."""
    r = ca._C.awacs_host(ntrials=2, duration=5.0, ntargets=200, seed=7,
                         threads=2, terrain=0)
    assert r["trials_ok"] == 2
    assert r["total_illuminated"] == 0
    assert r["total_shielded"] == 0
    assert r["total_detections"] > 0


def test_awacs_pipeline_deterministic():
    a = ca._C.awacs_host(ntrials=2, duration=4.0, ntargets=500, seed=9,
                         threads=1)
    b = ca._C.awacs_host(ntrials=2, duration=4.0, ntargets=500, seed=9,
                         threads=2)
    for k in ("total_detections", "total_illuminated", "total_shielded",
              "sum_clutter", "sum_power"):
        assert a[k] == b[k], k


@pytest.mark.gpu
def test_awacs_pipeline_gpu_matches_host():
    """Device pipeline (wave-cooperative LOS + lane-parallel clutter with
    host-tree-order folds) vs the host scalar reference.  libm-vs-OCML
    transcendentals differ by ulps, so boundary decisions (beam gate, LOS,
    draws) may flip rarely — tight relative tolerances, not bitwise."""
    g = ca._C.awacs_gpu(ntrials=8, duration=8.0, ntargets=1000, seed=21,
                        device=0)
    h = ca._C.awacs_host(ntrials=8, duration=8.0, ntargets=1000, seed=21,
                         threads=0)
    assert g["trials_ok"] == 8
    assert g["total_dwells"] == h["total_dwells"]
    assert g["total_events"] == h["total_events"]
    rel = lambda a, b: abs(a - b) / max(b, 1)
    # beam-gate boundary: atan2f OCML vs libm differ by ulps
    assert rel(g["total_illuminated"], h["total_illuminated"]) < 2e-3
    assert rel(g["total_shielded"], h["total_shielded"]) < 0.01
    assert rel(g["total_detections"], h["total_detections"]) < 0.03
    # clutter folds are host-tree-order; only gate-ulp survivor-set
    # differences move the sum
    assert abs(g["sum_clutter"] - h["sum_clutter"]) < 2e-3 * max(
        h["sum_clutter"], 1e-30)
    assert g["total_shielded"] > 0
