from ddlbench_amd.utils.metrics import AverageMeter, accuracy, gpu_memory_gb  # noqa: F401
from ddlbench_amd.utils.logging import BenchLogger, parse_result_line  # noqa: F401
