#!/usr/bin/env python3
"""Timed mixed-workload GPU soak: rounds of all four models + device
scenarios, every result asserted.  Used for the round-1 stability
burn-ins recorded in profiles/r01_mm1_engine.md.

Usage: python scripts/soak.py [seconds]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca  # noqa: E402

budget = float(sys.argv[1]) if len(sys.argv) > 1 else 120.0
t0 = time.time()
rounds = 0
events = 0
while time.time() - t0 < budget:
    seed = 1000 + rounds
    r = ca.mm1_gpu(ntrials=262144, num_objects=2000, seed=seed, device=0)
    assert r["trials_ok"] == 262144, r
    events += r["total_events"]
    r = ca.mg1_gpu(ntrials=262144, num_objects=2000, arr_rate=0.8,
                   srv_mean=1.0, srv_scv=2.0, dist=2, seed=seed, device=0)
    assert r["trials_ok"] == 262144, r
    events += r["total_events"]
    r = ca.jobshop_gpu(ntrials=65536, entities=500, njobs=12, seed=seed,
                       device=0)
    assert r["trials_ok"] == 65536, r
    events += r["total_events"]
    r = ca._C.awacs_gpu(ntrials=1024, duration=5.0, ntargets=128, seed=seed,
                        device=0)
    assert r["trials_ok"] == 1024, r
    events += r["total_events"]
    for w in list(range(1, 14)) + [15, 16, 17, 18, 19]:
        g = ca._C.scenario_gpu(w)
        h = ca._C.scenario_host(w)
        assert g["trace"] == h["trace"], w
    rounds += 1
el = time.time() - t0
print(f"soak OK: {rounds} rounds, {events} events in {el:.0f} s "
      f"({events/el/1e9:.2f} G ev/s sustained)")
