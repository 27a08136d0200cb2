#!/usr/bin/env python3
"""The reference's showcase experiment (reference test/test_cimba.c +
README "So, what can I use all that speed for?"): an M/G/1 queue at
4 service-variability levels x 5 utilization levels x `reps`
replications, each trial ~1e6 time units (objects = rho * 1e6 since the
mean service time is 1.0).  The reference quotes ~1.5 s for the 200-trial
grid on a 32-core Threadripper 3970X.

Usage: python scripts/mg1_grid.py [--backend cpu|gpu] [--reps N]
Prints one line per grid cell (PK-checked) and a total-time summary.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca  # noqa: E402

SCVS = [0.25, 0.5, 1.0, 2.0]   # gamma service at 4 variability levels
RHOS = [0.5, 0.6, 0.7, 0.8, 0.9]


def pk(lam, m, scv):
    es2 = scv * m * m + m * m
    return lam * es2 / (2.0 * (1.0 - lam * m)) + m


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--backend", default="cpu", choices=["cpu", "gpu"])
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--threads", type=int, default=0)
    args = ap.parse_args()

    total_events = 0
    worst = 0.0
    t0 = time.time()
    for scv in SCVS:
        for rho in RHOS:
            objs = int(rho * 1_000_000)
            kw = dict(ntrials=args.reps, num_objects=objs, arr_rate=rho,
                      srv_mean=1.0, srv_scv=scv, dist=1,
                      seed=int(scv * 100) * 1000 + int(rho * 100))
            if args.backend == "gpu":
                r = ca.mg1_gpu(device=0, **kw)
            else:
                r = ca.mg1_host(threads=args.threads, **kw)
            assert r["trials_ok"] == args.reps, r
            total_events += r["total_events"]
            theory = pk(rho, 1.0, scv)
            err = abs(r["avg_system_time"] - theory) / theory
            worst = max(worst, err)
            print(f"scv={scv:4.2f} rho={rho:.1f} E[T]={r['avg_system_time']:7.3f} "
                  f"PK={theory:7.3f} err={100*err:4.1f}%")
    el = time.time() - t0
    n = len(SCVS) * len(RHOS) * args.reps
    print(f"\n{n} trials, {total_events} events in {el:.2f} s "
          f"({total_events/el/1e6:.0f} M ev/s, backend={args.backend}); "
          f"worst PK error {100*worst:.1f}%")
    assert worst < 0.12, "PK validation failed"


if __name__ == "__main__":
    main()
