"""Model registry: one call to run any built-in model on CPU or GPU.

The models themselves are native (C++/HIP, cimba_amd/csrc/models/); this
is the orchestration-side index.

    from cimba_amd.models import run
    run("mm1", ntrials=1024, num_objects=10_000)           # auto backend
    run("mg1", backend="gpu", ntrials=65_536, ...)
"""
from .. import _C, gpu_device_count

_HOST = {
    "mm1": _C.mm1_host,
    "mg1": _C.mg1_host,
    "jobshop": _C.jobshop_host,
    "awacs": _C.awacs_host,
}
_GPU = {
    "mm1": _C.mm1_gpu,
    "mg1": _C.mg1_gpu,
    "jobshop": _C.jobshop_gpu,
    "awacs": _C.awacs_gpu,
}

MODELS = tuple(sorted(_HOST))


def run(name, backend="auto", **kwargs):
    """Run `name` in {mm1, mg1, jobshop, awacs}; backend in
    {auto, cpu, gpu}.  kwargs are forwarded to the model runner
    (ntrials, seed, and model-specific parameters)."""
    if name not in _HOST:
        raise KeyError(f"unknown model {name!r}; have {MODELS}")
    if backend == "auto":
        backend = "gpu" if gpu_device_count() > 0 else "cpu"
    if backend == "gpu":
        return _GPU[name](**kwargs)
    if backend == "cpu":
        return _HOST[name](**kwargs)
    raise ValueError(f"backend must be auto/cpu/gpu, got {backend!r}")
