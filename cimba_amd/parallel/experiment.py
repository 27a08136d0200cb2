"""Distributed experiment fan-out and statistics reduction.

The MI355X counterpart of the reference's cross-trial merge (SURVEY.md
§5.8): the reference merges cmb_datasummary partials on the host after
cimba_run returns (benchmark/MM1_multi.c:145-151); here each rank owns one
GPU, runs its shard of the replications, and the partial summaries are
merged across ranks over torch.distributed — backend "nccl" IS RCCL over
xGMI on ROCm; the payloads are KB-scale, so a latency-bound all_gather +
exact local Pébay merge beats a ring all-reduce of hand-linearized
moments (and keeps the merge exact).
"""
import os

from .. import DataSummary, WtdSummary


def shard_range(total, rank, world):
    """Contiguous [lo, hi) shard of `total` trials for `rank` of `world`."""
    base = total // world
    rem = total % world
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def _allgather_tuples(vec, group=None):
    import torch
    import torch.distributed as dist

    t = torch.tensor(vec, dtype=torch.float64)
    if dist.get_backend(group) == "nccl":
        t = t.cuda()
    world = dist.get_world_size(group)
    out = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(out, t, group=group)
    return [o.cpu().tolist() for o in out]

def allreduce_datasummary(ds, group=None):
    """Merge a DataSummary across all ranks (exact, associative)."""
    import torch.distributed as dist

    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return ds
    parts = _allgather_tuples(list(ds.raw()), group)
    merged = DataSummary()
    for p in parts:
        merged.merge(DataSummary.from_raw(*p))
    return merged


def allreduce_wtdsummary(ws, group=None):
    """Merge a WtdSummary across all ranks (exact, associative)."""
    import torch.distributed as dist

    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return ws
    parts = _allgather_tuples(list(ws.raw()), group)
    merged = WtdSummary()
    for p in parts:
        merged.merge(WtdSummary.from_raw(*p))
    return merged


def run_distributed_mm1(ntrials_total, num_objects, seed, use_gpu=None,
                        group=None):
    """Run an M/M/1 experiment sharded across the ranks of the default
    process group; returns (merged DataSummary of per-trial avg system
    times, total events).  Single-process fallback when torch.distributed
    is not initialized."""
    from .. import mm1_gpu, mm1_host, gpu_device_count

    try:
        import torch.distributed as dist
        initialized = dist.is_initialized()
    except ImportError:
        initialized = False

    rank = dist.get_rank(group) if initialized else 0
    world = dist.get_world_size(group) if initialized else 1
    lo, hi = shard_range(ntrials_total, rank, world)

    if use_gpu is None:
        use_gpu = gpu_device_count() > 0
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    # trial_base seeds each trial by its GLOBAL index, so any world size
    # simulates the identical trial set (bit-identical per-trial streams)
    if use_gpu:
        r = mm1_gpu(ntrials=hi - lo, num_objects=num_objects,
                    seed=seed, device=local_rank, trial_base=lo)
    else:
        r = mm1_host(ntrials=hi - lo, num_objects=num_objects,
                     seed=seed, threads=0, trial_base=lo)

    ds = DataSummary()
    if "per_trial_avg" in r:
        for v in r["per_trial_avg"]:
            ds.add(v)
    else:
        ds.add(r["avg_system_time"])

    events = r["total_events"]
    if initialized and world > 1:
        ds = allreduce_datasummary(ds, group)
        import torch
        te = torch.tensor([float(events)], dtype=torch.float64)
        if dist.get_backend(group) == "nccl":
            te = te.cuda()
        dist.all_reduce(te, group=group)
        events = int(te.item())
    return ds, events
