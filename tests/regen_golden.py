#!/usr/bin/env python3
"""Regenerate tests/golden.json — fixed-seed stochastic regression outputs
(the counterpart of reference test/tools/test_stochastic.py + the
test/reference/*.txt golden files: fixed seed 0x34f05c64d7ad598f, diff the
outputs).  Run after an INTENTIONAL behavior change; the test diff shows
what moved.  Doubles are stored as hex bit patterns (exact)."""
import json
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca  # noqa: E402

SEED = 0x34F05C64D7AD598F


def dhex(x):
    return struct.pack("<d", x).hex()


def build():
    g = {}
    g["sfc64_raw"] = [ca.sfc64_raw(SEED, i) for i in range(8)]
    g["rng"] = {}
    for dist, params in [("std_normal", []), ("std_exponential", []),
                         ("std_gamma", [2.5]), ("poisson", [12.0]),
                         ("binomial", [400, 0.4])]:
        xs = ca.rng_sample(dist, params, 6, SEED)
        g["rng"][dist] = [dhex(v) for v in xs]
    r = ca.mm1_host(ntrials=4, num_objects=2000, seed=SEED, threads=2)
    g["mm1"] = {"events": r["total_events"], "wait": dhex(r["total_wait"])}
    r = ca.mg1_host(ntrials=4, num_objects=2000, arr_rate=0.8, srv_mean=1.0,
                    srv_scv=0.25, dist=1, seed=SEED, threads=2)
    g["mg1"] = {"events": r["total_events"],
                "sys": dhex(r["avg_system_time"])}
    r = ca.jobshop_host(ntrials=2, entities=500, njobs=8, seed=SEED, threads=2)
    g["jobshop"] = {"events": r["total_events"],
                    "makespan": dhex(r["mean_makespan"])}
    r = ca._C.awacs_host(ntrials=2, duration=2.0, ntargets=128, seed=SEED,
                         threads=2, terrain=0)
    g["awacs"] = {"events": r["total_events"],
                  "detections": r["total_detections"],
                  "power": dhex(r["sum_power"])}
    r = ca._C.awacs_host(ntrials=2, duration=2.0, ntargets=128, seed=SEED,
                         threads=2, terrain=1)
    g["awacs_terrain"] = {"events": r["total_events"],
                          "detections": r["total_detections"],
                          "illuminated": r["total_illuminated"],
                          "shielded": r["total_shielded"],
                          "clutter": dhex(r["sum_clutter"]),
                          "power": dhex(r["sum_power"])}
    g["scenarios"] = {str(w): ca._C.scenario_host(w)["trace"]
                      for w in list(range(1, 14)) + [15, 16, 17, 18, 19]}
    t = ca._C.terrain_host(64, 48, base=50.0, amp=400.0, octaves=5,
                           seed=SEED, xs=[3.25, 40.0], ys=[7.5, 20.0],
                           queries=[0, 0, 9000, 63, 47, 9000], nsteps=64)
    g["terrain"] = {
        "stats": [dhex(v) for v in t["stats"]],
        "samples": [struct.pack("<f", v).hex() for v in t["samples"]],
        "vis": t["vis"],
    }
    return g


if __name__ == "__main__":
    out = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "golden.json")
    with open(out, "w") as f:
        json.dump(build(), f, indent=1, sort_keys=True)
    print("wrote", out)
