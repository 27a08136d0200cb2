#!/usr/bin/env python3
"""cimba_amd flagship benchmark: M/M/1 multi-replication events/sec.

Measures the BASELINE.json headline metric — simulated events/sec (whole
node) on the M/M/1 multi-replication workload (reference
benchmark/MM1_multi.c: arrival+service processes, unlimited queue,
exponential interarrival 1/0.9 and service 1.0, avg system time ~ 10) —
with the GPU-resident trial-per-wavefront engine.

Contract: `python bench.py --gpus N --steps K --warmup W`; for N>1 the
driver launches one rank per GPU via torch.distributed.run.  One step = a
fixed batch of replications per GPU (weak scaling).  Rank 0 prints ONE
JSON line.

vs_baseline divides by 1.0e9 events/s — the reference's published
whole-node aggregate (~25M ev/s/core x 32 cores ~ 0.8-1 G;
we use the stated "~1 G" figure).
"""
import argparse
import json
import os
import sys
import time

BASELINE_EVENTS_PER_SEC = 1.0e9  # reference 3970X whole-node aggregate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--trials", type=int, default=524288,
                    help="replications per GPU per step (524288 keeps the "
                         "block supply deep enough that tail-fill cost is "
                         "amortized; measured +18%% vs 262144)")
    ap.add_argument("--objects", type=int, default=10000,
                    help="objects per replication")
    ap.add_argument("--seed", type=lambda s: int(s, 0), default=0x34F05C64D7AD598F)
    ap.add_argument("--host", action="store_true",
                    help="debug: run the CPU host engine instead of the GPU")
    ap.add_argument("--model", choices=["mm1", "mg1", "jobshop", "awacs"],
                    default="mm1",
                    help="mm1 = headline benchmark; mg1/jobshop = "
                         "BASELINE configs 3-4")
    args = ap.parse_args()

    # Self-launch: `python bench.py --gpus N` with no external launcher
    # re-execs under torch.distributed.run, one rank per GPU over RCCL
    # (VERDICT r01 item 1).  When the driver already launched us through
    # torch.distributed.run, WORLD_SIZE is set and we skip this.
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        import socket
        import subprocess

        with socket.socket() as s:  # free rendezvous port on loopback
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", f"--nproc-per-node={args.gpus}",
               "--master-addr=127.0.0.1", f"--master-port={port}",
               os.path.abspath(__file__)] + sys.argv[1:]
        sys.exit(subprocess.call(cmd))

    import cimba_amd as ca

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # n_gpus reports the ACTUAL world size, never the requested one
    n_gpus = world

    dist = None
    torch = None
    if world > 1:
        import torch  # noqa: F811
        import torch.distributed as dist  # noqa: F811
        backend = "nccl" if (not args.host and torch.cuda.is_available()) else "gloo"
        if backend == "nccl":  # bind the device before init (rank->GPU map)
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)

    use_gpu = not args.host
    if use_gpu and ca.gpu_device_count() <= local_rank:
        print(json.dumps({"error": "no HIP device visible", "rank": rank}),
              file=sys.stderr)
        sys.exit(2)

    def barrier_sync():
        if use_gpu:
            ca.gpu_sync()
        if dist is not None:
            dist.barrier()
        if use_gpu:
            ca.gpu_sync()

    def one_step(step_idx):
        # same master seed on every rank; ranks are disjoint via trial_base
        # (global trial indices), so the union over ranks is the same trial
        # set a 1-rank run of world*trials would simulate
        seed = ca.fmix64((args.seed + step_idx + 1) & 0xFFFFFFFFFFFFFFFF)
        ntrials = args.trials
        if args.model == "mm1":
            fn = ca.mm1_gpu if use_gpu else ca.mm1_host
            kw = dict(ntrials=args.trials, num_objects=args.objects, seed=seed)
        elif args.model == "mg1":
            fn = ca.mg1_gpu if use_gpu else ca.mg1_host
            kw = dict(ntrials=args.trials, num_objects=args.objects,
                      arr_rate=0.8, srv_mean=1.0, srv_scv=0.25, dist=3,
                      seed=seed)
        elif args.model == "jobshop":
            fn = ca.jobshop_gpu if use_gpu else ca.jobshop_host
            kw = dict(ntrials=args.trials, entities=args.objects, njobs=24,
                      seed=seed)
        else:
            # AWACS cost ~ trials x duration/dwell x targets: scale the
            # defaults down to a comparable per-step budget
            ntrials = min(args.trials, 4096)
            fn = ca._C.awacs_gpu if use_gpu else ca._C.awacs_host
            kw = dict(ntrials=ntrials, duration=min(args.objects / 250.0, 60.0),
                      ntargets=1000, seed=seed)
        kw["trial_base"] = rank * ntrials
        if use_gpu:
            kw["device"] = local_rank
        else:
            kw["threads"] = 0
        r = fn(**kw)
        if r["trials_ok"] != ntrials:
            raise RuntimeError(
                f"rank {rank}: {ntrials - r['trials_ok']} trials aborted "
                f"(status {r['first_bad_status']})")
        return r["total_events"]

    # warmup (untimed)
    for i in range(args.warmup):
        one_step(-1 - i)

    barrier_sync()
    t0 = time.perf_counter()
    events = 0
    for k in range(args.steps):
        events += one_step(k)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks; sum events over ranks
    if dist is not None:
        import torch
        te = torch.tensor([elapsed], dtype=torch.float64)
        tv = torch.tensor([float(events)], dtype=torch.float64)
        if dist.get_backend() == "nccl":  # nccl reduces CUDA tensors only
            te = te.cuda()
            tv = tv.cuda()
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        dist.all_reduce(tv, op=dist.ReduceOp.SUM)
        elapsed = te.item()
        events = int(tv.item())

    if rank == 0:
        value = events / elapsed
        out = {
            "metric": "sim_events_per_sec",
            "value": value,
            "unit": "events/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / BASELINE_EVENTS_PER_SEC,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": {"mm1": "MM1_multi", "mg1": "MG1_resource",
                          "jobshop": "JobShop_pools", "awacs": "AWACS_radar"}[args.model],
                "trials_per_gpu_per_step": args.trials,
                "objects_per_trial": args.objects,
                "arrival_rate": 0.9,
                "service_rate": 1.0,
                "parallelism": f"trial-parallel dp{n_gpus}",
                "engine": "vote-gated converged lane kernel (64 trials/wave, "
                          "wave-voted dispatch paths, heap-top register "
                          "cache); trial-per-wavefront + LDS for small "
                          "batches",
                "device": "gpu" if use_gpu else "cpu-host-debug",
            },
        }
        print(json.dumps(out), flush=True)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
