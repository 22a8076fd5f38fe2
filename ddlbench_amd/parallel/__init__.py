from ddlbench_amd.parallel.dist_utils import (  # noqa: F401
    init_distributed, allreduce_mean_scalar, distributed_env)
from ddlbench_amd.parallel.ddp import BucketedDataParallel  # noqa: F401
