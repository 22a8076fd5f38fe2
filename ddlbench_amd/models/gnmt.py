"""GNMT v2 translation model family.

Parity with the reference's second workload
(/root/reference/pipedream-fork/runtime/translation/seq2seq/models/ —
SURVEY.md §2.12): ResidualRecurrentEncoder (emulated-bidirectional LSTM
layer 0 + unidirectional residual LSTMs), ResidualRecurrentDecoder with
Bahdanau RecurrentAttention and a Classifier head, hidden=1024,
vocab=32320. LSTM cells go through torch.nn.LSTM (MIOpen RNN on ROCm);
the emulated bidirectional layer uses the hand-written revert_varlen
HIP kernel (ops/csrc/seq_utils.hip) with a torch fallback on CPU."""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ddlbench_amd import ops as _ops

PAD, BOS, EOS = 0, 1, 2


# ------------------------------------------------------------- revert op
class _RevertVarlen(torch.autograd.Function):
    """out[t,b] = x[len_b-1-t, b] for t < len_b else 0. Self-inverse —
    the backward is the same op (reference uses the same kernel in both
    directions, encoder.py:13-23)."""

    @staticmethod
    def forward(ctx, x, lengths):
        ctx.save_for_backward(lengths)
        if _ops.use_native(x, "auto"):
            return _ops.require_extension().revert_varlen(
                x.contiguous(), lengths)
        return _revert_torch(x, lengths)

    @staticmethod
    def backward(ctx, dy):
        (lengths,) = ctx.saved_tensors
        if _ops.use_native(dy, "auto"):
            return (_ops.require_extension().revert_varlen(
                dy.contiguous(), lengths), None)
        return _revert_torch(dy, lengths), None


def _revert_torch(x: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    T, B, Fdim = x.shape
    out = torch.zeros_like(x)
    for b in range(B):
        n = int(lengths[b])
        out[:n, b] = x[:n, b].flip(0)
    return out


def revert_varlen(x: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    return _RevertVarlen.apply(x, lengths)


def varlen_mask(lengths: torch.Tensor, T: int) -> torch.Tensor:
    """(T, B) bool valid-timestep mask."""
    if lengths.is_cuda and _ops.extension_available():
        return _ops.require_extension().varlen_mask(lengths, T).bool()
    ar = torch.arange(T, device=lengths.device).unsqueeze(1)
    return ar < lengths.unsqueeze(0)


# ------------------------------------------------------------- encoder
class EmuBidirLSTM(nn.Module):
    """Bidirectional layer as two unidirectional LSTMs + time reversal,
    so a pipeline partitioner can split it (reference encoder.py:25-46)."""

    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        self.fwd = nn.LSTM(input_size, hidden_size)
        self.bwd = nn.LSTM(input_size, hidden_size)

    def forward(self, x, lengths):
        y_f, _ = self.fwd(x)
        rev = revert_varlen(x, lengths)
        y_b, _ = self.bwd(rev)
        y_b = revert_varlen(y_b, lengths)
        return torch.cat([y_f, y_b], dim=2)


class ResidualRecurrentEncoder(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int = 1024,
                 num_layers: int = 4, dropout: float = 0.2):
        super().__init__()
        self.embedder = nn.Embedding(vocab_size, hidden_size,
                                     padding_idx=PAD)
        self.bidir = EmuBidirLSTM(hidden_size, hidden_size)
        self.layer1 = nn.LSTM(2 * hidden_size, hidden_size)
        self.layers = nn.ModuleList(
            [nn.LSTM(hidden_size, hidden_size)
             for _ in range(num_layers - 2)])
        self.dropout = nn.Dropout(dropout)

    def forward(self, src, src_len):
        # src: (T, B) int64
        x = self.embedder(src)
        x = self.dropout(x)
        x = self.bidir(x, src_len)
        x = self.dropout(x)
        x, _ = self.layer1(x)
        for rnn in self.layers:
            res = x
            x = self.dropout(x)
            x, _ = rnn(x)
            x = x + res
        return x  # (T, B, H)


# ------------------------------------------------------------ attention
class BahdanauAttention(nn.Module):
    """Normalized additive attention (reference attention.py:12-115)."""

    def __init__(self, query_size: int, key_size: int, hidden: int):
        super().__init__()
        self.linear_q = nn.Linear(query_size, hidden, bias=False)
        self.linear_k = nn.Linear(key_size, hidden, bias=False)
        self.v = nn.Parameter(torch.empty(hidden))
        self.b = nn.Parameter(torch.zeros(hidden))
        self.g = nn.Parameter(torch.tensor(math.sqrt(1.0 / hidden)))
        nn.init.uniform_(self.v, -1.0 / math.sqrt(hidden),
                         1.0 / math.sqrt(hidden))

    def score(self, q, k):
        # q: (B, Tq, H'), k: (B, Tk, H')
        vn = self.g * self.v / self.v.norm()
        s = torch.tanh(q.unsqueeze(2) + k.unsqueeze(1) + self.b)
        return torch.einsum("bqkh,h->bqk", s, vn)

    def forward(self, query, keys, mask: Optional[torch.Tensor] = None):
        # query: (B, Tq, Hq); keys: (B, Tk, Hk); mask: (B, Tk) bool valid
        q = self.linear_q(query)
        k = self.linear_k(keys)
        scores = self.score(q, k)
        if mask is not None:
            scores = scores.masked_fill(~mask.unsqueeze(1), -65504.0)
        attn = F.softmax(scores.float(), dim=-1).to(keys.dtype)
        ctx = torch.bmm(attn, keys)
        return ctx, attn


class RecurrentAttention(nn.Module):
    def __init__(self, input_size: int, context_size: int,
                 hidden_size: int, dropout: float = 0.2):
        super().__init__()
        self.rnn = nn.LSTM(input_size, hidden_size)
        self.attn = BahdanauAttention(hidden_size, context_size,
                                      hidden_size)
        self.dropout = nn.Dropout(dropout)

    def forward(self, x, context, src_mask, hidden=None):
        # x: (T, B, H); context: (Tk, B, H)
        self.rnn.flatten_parameters()
        y, hidden = self.rnn(self.dropout(x), hidden)
        ctx, attn = self.attn(y.transpose(0, 1),
                              context.transpose(0, 1), src_mask)
        return y, ctx.transpose(0, 1), attn, hidden


class Classifier(nn.Module):
    def __init__(self, hidden_size: int, vocab_size: int):
        super().__init__()
        self.fc = nn.Linear(hidden_size, vocab_size)

    def forward(self, x):
        return self.fc(x)


class ResidualRecurrentDecoder(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int = 1024,
                 num_layers: int = 4, dropout: float = 0.2):
        super().__init__()
        self.embedder = nn.Embedding(vocab_size, hidden_size,
                                     padding_idx=PAD)
        self.att_rnn = RecurrentAttention(hidden_size, hidden_size,
                                          hidden_size, dropout)
        self.layer1 = nn.LSTM(2 * hidden_size, hidden_size)
        self.layers = nn.ModuleList(
            [nn.LSTM(2 * hidden_size, hidden_size)
             for _ in range(num_layers - 2)])
        self.classifier = Classifier(hidden_size, vocab_size)
        self.dropout = nn.Dropout(dropout)

    def forward(self, tgt, context, src_mask, hidden=None):
        x = self.embedder(tgt)
        x, ctx, _, h0 = self.att_rnn(x, context, src_mask,
                                     hidden[0] if hidden else None)
        new_hidden = [h0]
        y = torch.cat([x, ctx], dim=2)
        y = self.dropout(y)
        y, h1 = self.layer1(y, hidden[1] if hidden else None)
        new_hidden.append(h1)
        for i, rnn in enumerate(self.layers):
            res = y
            y2 = torch.cat([self.dropout(y), ctx], dim=2)
            y, hi = rnn(y2, hidden[i + 2] if hidden else None)
            new_hidden.append(hi)
            y = y + res
        return self.classifier(y), new_hidden


class GNMT(nn.Module):
    """Seq2seq wrapper (reference gnmt.py:13-56). forward() is the
    training path (teacher forcing)."""

    def __init__(self, vocab_size: int = 32320, hidden_size: int = 1024,
                 num_layers: int = 4, dropout: float = 0.2):
        super().__init__()
        self.vocab_size = vocab_size
        self.encoder = ResidualRecurrentEncoder(vocab_size, hidden_size,
                                                num_layers, dropout)
        self.decoder = ResidualRecurrentDecoder(vocab_size, hidden_size,
                                                num_layers, dropout)

    def encode(self, src, src_len):
        return self.encoder(src, src_len)

    def forward(self, src, src_len, tgt_in):
        context = self.encoder(src, src_len)
        src_mask = varlen_mask(src_len, src.size(0)).transpose(0, 1)
        logits, _ = self.decoder(tgt_in, context, src_mask)
        return logits  # (Ttgt, B, V)


# ---------------------------------------------------- pipeline stages
class _EncEmbed(nn.Module):
    """(src, src_len, tgt_in) -> (x, src_len, tgt_in)"""

    def __init__(self, enc: ResidualRecurrentEncoder):
        super().__init__()
        self.embedder = enc.embedder
        self.bidir = enc.bidir
        self.dropout = enc.dropout

    def forward(self, src, src_len, tgt_in):
        x = self.dropout(self.embedder(src.long()))
        x = self.bidir(x, src_len.long())
        return x, src_len, tgt_in


class _EncLayer1(nn.Module):
    def __init__(self, enc):
        super().__init__()
        self.layer1 = enc.layer1
        self.dropout = enc.dropout

    def forward(self, x, src_len, tgt_in):
        y, _ = self.layer1(self.dropout(x))
        return y, src_len, tgt_in


class _EncResidual(nn.Module):
    def __init__(self, rnn, dropout):
        super().__init__()
        self.rnn = rnn
        self.dropout = dropout

    def forward(self, x, src_len, tgt_in):
        y, _ = self.rnn(self.dropout(x))
        return x + y, src_len, tgt_in


class _DecAttn(nn.Module):
    """(context, src_len, tgt_in) -> (y, ctx, tgt_in)"""

    def __init__(self, dec: ResidualRecurrentDecoder):
        super().__init__()
        self.embedder = dec.embedder
        self.att_rnn = dec.att_rnn

    def forward(self, context, src_len, tgt_in):
        mask = varlen_mask(src_len.long(), context.size(0)).transpose(0, 1)
        x = self.embedder(tgt_in.long())
        y, ctx, _, _ = self.att_rnn(x, context, mask)
        return y, ctx


class _DecLayer1(nn.Module):
    def __init__(self, dec):
        super().__init__()
        self.layer1 = dec.layer1
        self.dropout = dec.dropout

    def forward(self, y, ctx):
        z, _ = self.layer1(self.dropout(torch.cat([y, ctx], dim=2)))
        return z, ctx


class _DecResidual(nn.Module):
    def __init__(self, rnn, dropout):
        super().__init__()
        self.rnn = rnn
        self.dropout = dropout

    def forward(self, y, ctx):
        z, _ = self.rnn(torch.cat([self.dropout(y), ctx], dim=2))
        return y + z, ctx


class _DecHead(nn.Module):
    def __init__(self, dec):
        super().__init__()
        self.classifier = dec.classifier

    def forward(self, y, ctx):
        return self.classifier(y)


def gnmt_pipeline_units(model: GNMT):
    """Flatten GNMT into tuple-I/O pipeline units (the counterpart of
    the reference's generated gnmt stage modules,
    pipedream-fork/runtime/translation/models/gnmt/gpus=4/). LSTM hidden
    state never crosses a unit boundary (each minibatch starts fresh),
    so any contiguous grouping of units is a valid stage."""
    enc, dec = model.encoder, model.decoder
    units = [_EncEmbed(enc), _EncLayer1(enc)]
    units += [_EncResidual(r, enc.dropout) for r in enc.layers]
    units += [_DecAttn(dec), _DecLayer1(dec)]
    units += [_DecResidual(r, dec.dropout) for r in dec.layers]
    units += [_DecHead(dec)]
    return units


def gnmt_edge_specs(model: GNMT, n_units: int, Tsrc: int, Ttgt: int,
                    B: int, dtype):
    """Static TensorSpec lists for every unit boundary, given fixed
    (padded) sequence lengths — pipeline mode trades the bucketing
    sampler's ragged batches for static shapes."""
    from ddlbench_amd.parallel.pipeline.runtime import TensorSpec
    H = model.encoder.layer1.hidden_size
    n_enc = 2 + len(model.encoder.layers)   # units producing (x,len,tgt)
    specs = []
    for i in range(n_units):
        if i < n_enc:  # after unit i: (x, src_len, tgt_in)
            specs.append([
                TensorSpec((Tsrc, B, 2 * H if i == 0 else H), dtype, True),
                TensorSpec((B,), torch.long, False),
                TensorSpec((Ttgt, B), torch.long, False)])
        elif i < n_units - 1:  # decoder body: (y, ctx)
            specs.append([TensorSpec((Ttgt, B, H), dtype, True),
                          TensorSpec((Ttgt, B, H), dtype, True)])
        else:  # logits
            specs.append([TensorSpec((Ttgt, B, model.vocab_size), dtype,
                                     True)])
    return specs


class LabelSmoothingLoss(nn.Module):
    """Per-token label-smoothed CE ignoring PAD
    (reference train/smoothing.py:7-18)."""

    def __init__(self, smoothing: float = 0.1, padding_idx: int = PAD):
        super().__init__()
        self.smoothing = smoothing
        self.padding_idx = padding_idx

    def forward(self, logits, target):
        # logits: (T, B, V); target: (T, B)
        V = logits.size(-1)
        logits = logits.reshape(-1, V).float()
        target = target.reshape(-1)
        non_pad = target != self.padding_idx
        logp = F.log_softmax(logits, dim=-1)
        nll = -logp.gather(1, target.clamp_min(0).unsqueeze(1)).squeeze(1)
        smooth = -logp.mean(dim=-1)
        loss = (1 - self.smoothing) * nll + self.smoothing * smooth
        n_tokens = non_pad.sum().clamp_min(1)
        return (loss * non_pad).sum() / n_tokens
