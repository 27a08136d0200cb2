from .experiment import (  # noqa: F401
    allreduce_datasummary,
    allreduce_wtdsummary,
    run_distributed_mm1,
    shard_range,
)
