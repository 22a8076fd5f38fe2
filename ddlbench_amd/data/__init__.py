from ddlbench_amd.data.synthetic import (  # noqa: F401
    SyntheticImageDataset, make_loaders, synthetic_batch)
