"""GNMT inference (greedy + beam search) and BLEU scoring.

Parity with the reference's Translator/beam_search and BLEU tooling
(/root/reference/pipedream-fork/runtime/translation/seq2seq/inference/
{inference,beam_search}.py, compute_bleu_scores.py). Implemented from
the algorithms, sized for batch decode on one GPU."""

from __future__ import annotations

import collections
import math
from typing import Sequence

import torch

from ddlbench_amd.models.gnmt import BOS, EOS, GNMT, PAD, varlen_mask


class Translator:
    def __init__(self, model: GNMT, max_len: int = 80,
                 beam_size: int = 5, len_norm: float = 0.6):
        self.model = model
        self.max_len = max_len
        self.beam_size = beam_size
        self.len_norm = len_norm

    @torch.no_grad()
    def greedy(self, src: torch.Tensor, src_len: torch.Tensor):
        model = self.model
        model.eval()
        B = src.size(1)
        device = src.device
        context = model.encode(src, src_len)
        src_mask = varlen_mask(src_len, src.size(0)).transpose(0, 1)
        tok = torch.full((1, B), BOS, dtype=torch.long, device=device)
        hidden = None
        done = torch.zeros(B, dtype=torch.bool, device=device)
        outs = []
        for _ in range(self.max_len):
            logits, hidden = model.decoder(tok, context, src_mask, hidden)
            tok = logits[-1].argmax(dim=-1, keepdim=True).t()
            tok = tok.masked_fill(done.unsqueeze(0), PAD)
            outs.append(tok.squeeze(0).clone())
            done |= tok.squeeze(0) == EOS
            if bool(done.all()):
                break
        return torch.stack(outs)  # (T, B)

    @torch.no_grad()
    def beam(self, src: torch.Tensor, src_len: torch.Tensor):
        """Batched beam search; returns best hypothesis per sentence."""
        model = self.model
        model.eval()
        K = self.beam_size
        B = src.size(1)
        device = src.device
        context = model.encode(src, src_len)          # (Ts, B, H)
        Ts, _, H = context.shape
        src_mask = varlen_mask(src_len, Ts).transpose(0, 1)  # (B, Ts)
        # expand to B*K
        context = context.unsqueeze(2).expand(Ts, B, K, H) \
            .reshape(Ts, B * K, H)
        src_mask = src_mask.unsqueeze(1).expand(B, K, Ts) \
            .reshape(B * K, Ts)
        scores = torch.full((B, K), -1e9, device=device)
        scores[:, 0] = 0.0
        tok = torch.full((1, B * K), BOS, dtype=torch.long, device=device)
        hidden = None
        alive = torch.ones(B, K, dtype=torch.bool, device=device)
        seqs = torch.zeros(self.max_len, B, K, dtype=torch.long,
                           device=device)
        for t in range(self.max_len):
            logits, hidden = model.decoder(tok, context, src_mask, hidden)
            logp = torch.log_softmax(logits[-1].float(), dim=-1) \
                .view(B, K, -1)                       # (B, K, V)
            V = logp.size(-1)
            logp = logp.masked_fill(~alive.unsqueeze(-1), -1e9)
            # finished beams keep their score by emitting PAD at cost 0
            pad_free = torch.full_like(logp[..., PAD], 0.0)
            logp[..., PAD] = torch.where(alive, logp[..., PAD], pad_free)
            cand = scores.unsqueeze(-1) + logp        # (B, K, V)
            flat = cand.view(B, K * V)
            scores, idx = flat.topk(K, dim=-1)        # (B, K)
            beam_idx = idx // V
            tok_idx = idx % V
            # reorder state
            gather = (torch.arange(B, device=device).unsqueeze(1) * K
                      + beam_idx).view(-1)
            hidden = [(h[0][:, gather], h[1][:, gather]) for h in hidden]
            seqs = seqs[:, torch.arange(B, device=device).unsqueeze(1),
                        beam_idx]
            seqs[t] = tok_idx
            alive = alive.gather(1, beam_idx) & (tok_idx != EOS) \
                & (tok_idx != PAD)
            tok = tok_idx.view(1, B * K)
            if not bool(alive.any()):
                break
        # length-normalized best
        lens = (seqs != PAD).sum(0).clamp_min(1).float()
        norm = ((5 + lens) / 6) ** self.len_norm
        best = (scores / norm).argmax(dim=-1)
        return seqs[:, torch.arange(B, device=device), best]  # (T, B)


def bleu(hyps: Sequence[Sequence[int]], refs: Sequence[Sequence[int]],
         max_n: int = 4) -> float:
    """Corpus BLEU over token-id sequences (uniform n-gram weights,
    brevity penalty)."""
    p_num = [0] * max_n
    p_den = [0] * max_n
    hyp_len = ref_len = 0
    for hyp, ref in zip(hyps, refs):
        hyp = [t for t in hyp if t not in (PAD, BOS, EOS)]
        ref = [t for t in ref if t not in (PAD, BOS, EOS)]
        hyp_len += len(hyp)
        ref_len += len(ref)
        for n in range(1, max_n + 1):
            hc = collections.Counter(
                tuple(hyp[i:i + n]) for i in range(len(hyp) - n + 1))
            rc = collections.Counter(
                tuple(ref[i:i + n]) for i in range(len(ref) - n + 1))
            p_num[n - 1] += sum(min(c, rc[g]) for g, c in hc.items())
            p_den[n - 1] += max(sum(hc.values()), 0)
    if min(p_den) == 0 or min(p_num) == 0:
        return 0.0
    logp = sum(math.log(n / d) for n, d in zip(p_num, p_den)) / max_n
    bp = min(1.0, math.exp(1 - ref_len / max(hyp_len, 1)))
    return 100.0 * bp * math.exp(logp)
