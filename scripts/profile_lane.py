#!/usr/bin/env python3
"""One short lane-kernel run for rocprofv3 PMC collection.

Usage (on the GPU box, from /tmp with PYTHONPATH=/root/repo):
    rocprofv3 --pmc VALUUtilization,VALUBusy,SALUBusy,MemUnitStalled \
        -d out -o tag -- python scripts/profile_lane.py mg1|jobshop

Sized so the profiled kernel runs ~1 s: enough for stable averages,
cheap on the GPU budget.  Counter summaries are committed under
profiles/ (r01_lane_divergence.md).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca  # noqa: E402

model = sys.argv[1] if len(sys.argv) > 1 else "mg1"
if model == "mm1":
    r = ca.mm1_gpu(ntrials=262144, num_objects=5000, seed=11, device=0)
elif model == "mg1":
    r = ca.mg1_gpu(ntrials=262144, num_objects=5000, arr_rate=0.8,
                   srv_mean=1.0, srv_scv=0.25, dist=1, seed=11, device=0)
elif model == "jobshop":
    r = ca.jobshop_gpu(ntrials=131072, entities=1000, njobs=12, seed=11,
                       device=0)
else:
    sys.exit("model must be mm1, mg1 or jobshop")
ev = r["total_events"]
ms = r["elapsed_ms"]
print(f"{model}: {ev} events in {ms:.1f} ms = {ev/ms/1e6:.2f} G ev/s, "
      f"trials_ok={r['trials_ok']}")
