"""Real-data loading: class-per-directory trees under DATADIR.

Reference semantics: `run.sh -s` trains from real data loaded with
torchvision ImageFolder + transforms
(/root/reference/run/run/run.sh:39-41,
benchmark/mnist/mnist_pytorch.py:163-199 region). torchvision/PIL are
not in this image, so the native format is tensor files — `.npy`
(numpy, HWC or CHW, uint8 or float) and `.pt` (torch tensors) — with
JPEG/PNG decoded through PIL when it happens to be importable.

Tree layout (= ImageFolder):

    DATADIR/<split>/<class_name>/<file>

with split ∈ {train, val, test} (val preferred over test for eval, like
the reference's imagenet layout). Labels are the sorted class-directory
index. Images are resized/validated to the dataset's (C,H,W) and
normalized to mean 0 / std 1 per-image when loaded from uint8.
"""

from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

from ddlbench_amd.config import DATASET_SHAPES

_EXTS = (".npy", ".pt", ".png", ".jpg", ".jpeg", ".bmp")


def _find_split_dir(root: str, train: bool) -> str:
    names = ["train"] if train else ["val", "test", "valid"]
    for n in names:
        d = os.path.join(root, n)
        if os.path.isdir(d):
            return d
    # flat tree (classes at the root) serves both splits
    return root


class RealImageDataset(Dataset):
    def __init__(self, dataset: str, root: str, train: bool = True):
        if not os.path.isdir(root):
            raise FileNotFoundError(
                f"DATADIR {root!r} does not exist (real-data mode)")
        self.shape = DATASET_SHAPES[dataset][:3]  # (C, H, W)
        split = _find_split_dir(root, train)
        classes = sorted(d for d in os.listdir(split)
                         if os.path.isdir(os.path.join(split, d)))
        if not classes:
            raise FileNotFoundError(
                f"no class directories under {split!r}")
        self.samples: List[Tuple[str, int]] = []
        for ci, cname in enumerate(classes):
            cdir = os.path.join(split, cname)
            for f in sorted(os.listdir(cdir)):
                if f.lower().endswith(_EXTS):
                    self.samples.append((os.path.join(cdir, f), ci))
        if not self.samples:
            raise FileNotFoundError(
                f"no {'/'.join(_EXTS)} files under {split!r}")
        self.num_classes = len(classes)

    def __len__(self) -> int:
        return len(self.samples)

    def _load(self, path: str) -> torch.Tensor:
        if path.endswith(".npy"):
            arr = np.load(path)
            t = torch.from_numpy(np.ascontiguousarray(arr))
        elif path.endswith(".pt"):
            t = torch.load(path, map_location="cpu", weights_only=True)
        else:
            try:
                from PIL import Image  # optional
            except ImportError as e:
                raise RuntimeError(
                    f"{path}: JPEG/PNG loading needs PIL; convert the "
                    "tree to .npy/.pt (benchmark/"
                    "generate_synthetic_data.py --materialize writes "
                    "that format)") from e
            t = torch.from_numpy(np.asarray(Image.open(path)))
        if t.dim() == 2:  # grayscale HW -> 1HW
            t = t.unsqueeze(0)
        elif t.dim() == 3 and t.shape[-1] in (1, 3) and \
                t.shape[0] not in (1, 3):
            t = t.permute(2, 0, 1)  # HWC -> CHW
        if t.dtype == torch.uint8:
            t = t.float().div_(255.0).sub_(0.5).div_(0.5)
        else:
            t = t.float()
        c, h, w = self.shape
        if t.shape[0] != c:
            if t.shape[0] == 1 and c == 3:
                t = t.expand(3, *t.shape[1:]).clone()
            elif t.shape[0] == 3 and c == 1:
                t = t.mean(dim=0, keepdim=True)
            else:
                raise ValueError(
                    f"{t.shape[0]} channels, dataset wants {c}")
        if t.shape[1] != h or t.shape[2] != w:
            t = torch.nn.functional.interpolate(
                t.unsqueeze(0), size=(h, w), mode="bilinear",
                align_corners=False).squeeze(0)
        return t

    def __getitem__(self, i: int):
        path, label = self.samples[i]
        return self._load(path), label
