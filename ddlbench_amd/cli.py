"""Entry helper shared by the benchmark/<dataset>_<impl>.py scripts."""

from __future__ import annotations


from ddlbench_amd.config import config_from_args, make_parser
from ddlbench_amd.strategies import run

_DEFAULT_ARCH = {
    "mnist": "resnet18", "cifar10": "resnet18",
    "imagenet": "resnet50", "highres": "resnet50",
}


def main(dataset: str, strategy: str, argv=None) -> dict:
    parser = make_parser(default_arch=_DEFAULT_ARCH[dataset])
    args = parser.parse_args(argv)
    cfg = config_from_args(dataset, strategy, args)
    return run(cfg)
