"""cimba_amd — MI355X-native discrete-event simulation framework.

A from-scratch, GPU-resident redesign of the capabilities of ambonvik/cimba
(multithreaded DES with coroutine processes): each replication runs as one
wavefront on gfx950 with its future-event list and process state in LDS;
replication fan-out across the GPUs of a node goes through
torch.distributed over RCCL/xGMI.  See SURVEY.md and docs/ for the layer
map and the reference-parity table.
"""
import os

__version__ = "0.1.0"

try:
    from . import _C  # noqa: F401
except ImportError:  # build in-tree on first use (hipcc cross-compiles on CPU)
    if os.environ.get("CIMBA_NO_BUILD"):
        raise
    from . import _build

    _build.build()
    from . import _C  # noqa: F401

from ._C import (  # noqa: E402  # noqa: F401,E402
    DataSummary,
    WtdSummary,
    fmix64,
    gpu_device_count,
    gpu_sync,
    jobshop_gpu,
    jobshop_host,
    mg1_gpu,
    mg1_host,
    mm1_gpu,
    mm1_host,
    rng_sample,
    sfc64_raw,
    trial_seed,
)


def has_gpu():
    """True when at least one HIP device is visible."""
    try:
        return gpu_device_count() > 0
    except Exception:
        return False
