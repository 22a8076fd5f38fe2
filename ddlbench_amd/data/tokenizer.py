"""Word-level tokenizer + parallel-text dataset for the GNMT workload.

Reference position: seq2seq/data/tokenizer.py (Moses-style vocab with
PAD/UNK/BOS/EOS specials) and seq2seq/data/dataset.py (padded parallel
TextDataset) in /root/reference/pipedream-fork/runtime/translation.
The reference ships WMT En-De; with no network this module builds the
vocab from whatever parallel line files DATADIR holds
(`<prefix>.src` / `<prefix>.tgt`, one sentence per line) — the
synthetic token stream (data/translation.py) stays the default.
"""

from __future__ import annotations

import os
from collections import Counter
from typing import List, Optional, Tuple

import torch
from torch.utils.data import Dataset

PAD, UNK, BOS, EOS = 0, 1, 2, 3
SPECIALS = ["<pad>", "<unk>", "<s>", "</s>"]


class Tokenizer:
    """Whitespace word-level vocab with the GNMT special tokens."""

    def __init__(self, vocab: Optional[List[str]] = None):
        self.itos: List[str] = list(SPECIALS)
        if vocab:
            self.itos += [w for w in vocab if w not in SPECIALS]
        self.stoi = {w: i for i, w in enumerate(self.itos)}

    @classmethod
    def build(cls, lines, max_size: int = 32000,
              min_freq: int = 1) -> "Tokenizer":
        counts = Counter()
        for ln in lines:
            counts.update(ln.strip().split())
        keep = [w for w, c in counts.most_common(max_size)
                if c >= min_freq]
        return cls(keep)

    def __len__(self) -> int:
        return len(self.itos)

    def encode(self, line: str, bos: bool = True,
               eos: bool = True) -> List[int]:
        ids = [self.stoi.get(w, UNK) for w in line.strip().split()]
        if bos:
            ids = [BOS] + ids
        if eos:
            ids = ids + [EOS]
        return ids

    def decode(self, ids) -> str:
        out = []
        for i in ids:
            i = int(i)
            if i == EOS:
                break
            if i in (PAD, BOS):
                continue
            out.append(self.itos[i] if 0 <= i < len(self.itos)
                       else SPECIALS[UNK])
        return " ".join(out)

    def save(self, path: str) -> None:
        with open(path, "w") as f:
            f.write("\n".join(self.itos))

    @classmethod
    def load(cls, path: str) -> "Tokenizer":
        with open(path) as f:
            words = [w for w in f.read().splitlines() if w]
        t = cls()
        t.itos = words
        t.stoi = {w: i for i, w in enumerate(words)}
        return t


class TextTranslationDataset(Dataset):
    """Parallel corpus with the SyntheticTranslationDataset item
    contract — unpadded (src_ids, tgt_ids) per item plus src_len(idx)
    for the bucketing sampler; collate_translation pads per batch."""

    def __init__(self, root: str, prefix: str = "train",
                 tokenizer_src: Optional[Tokenizer] = None,
                 tokenizer_tgt: Optional[Tokenizer] = None,
                 max_len: int = 50, min_len: int = 2):
        src_path = os.path.join(root, prefix + ".src")
        tgt_path = os.path.join(root, prefix + ".tgt")
        for p in (src_path, tgt_path):
            if not os.path.exists(p):
                raise FileNotFoundError(
                    f"parallel corpus file {p!r} missing")
        with open(src_path) as f:
            src_lines = f.read().splitlines()
        with open(tgt_path) as f:
            tgt_lines = f.read().splitlines()
        if len(src_lines) != len(tgt_lines):
            raise ValueError("src/tgt line counts differ")
        self.tok_src = tokenizer_src or Tokenizer.build(src_lines)
        self.tok_tgt = tokenizer_tgt or Tokenizer.build(tgt_lines)
        pairs: List[Tuple[List[int], List[int]]] = []
        for s, t in zip(src_lines, tgt_lines):
            es = self.tok_src.encode(s)
            et = self.tok_tgt.encode(t)
            if min_len <= len(es) <= max_len and \
                    min_len <= len(et) <= max_len:
                pairs.append((es, et))
        if not pairs:
            raise ValueError("no sentence pairs within length bounds")
        self.pairs = pairs
        self.src_max = max(len(s) for s, _ in pairs)
        self.tgt_max = max(len(t) for _, t in pairs)

    def __len__(self) -> int:
        return len(self.pairs)

    def src_len(self, idx: int) -> int:
        return len(self.pairs[idx][0])

    def __getitem__(self, idx: int):
        s, t = self.pairs[idx]
        return (torch.tensor(s, dtype=torch.long),
                torch.tensor(t, dtype=torch.long))
