"""GNMT training/eval runner (single-device or DP over RCCL).

The reference runs GNMT only through the pipedream driver
(SURVEY.md §2.12); here the translation workload is a first-class
benchmark: teacher-forced training with label smoothing, per-token loss,
validation loss + greedy-decode BLEU, with the standard result-log
contract (samples/sec = sentences/sec)."""

from __future__ import annotations

import time

import torch

from ddlbench_amd.engine import compute_dtype, resolve_device
from ddlbench_amd.models.gnmt import GNMT, LabelSmoothingLoss
from ddlbench_amd.data.translation import (BucketingSampler,
                                           SyntheticTranslationDataset,
                                           collate_translation)
from ddlbench_amd.ops.sgd import FusedSGD
from ddlbench_amd.utils import AverageMeter, BenchLogger, gpu_memory_gb


def run_gnmt(epochs=3, batch_size=64, dataset_size=2000, vocab=32320,
             hidden=1024, layers=4, lr=0.25e-3, dtype="float32",
             device="auto", log_interval=25, seed=42, max_len=50,
             ddp=False, kernel_backend="auto", bleu_batches=2) -> dict:
    from ddlbench_amd.parallel import (BucketedDataParallel,
                                       allreduce_mean_scalar,
                                       init_distributed)
    env = init_distributed() if ddp else None
    world = env.world_size if env else 1
    rank = env.rank if env else 0
    torch.manual_seed(seed)

    class _Cfg:
        pass

    cfg = _Cfg()
    cfg.device = device
    dev = resolve_device(cfg, env.local_rank if env else 0)
    dt = torch.bfloat16 if dtype == "bfloat16" else torch.float32

    model = GNMT(vocab_size=vocab, hidden_size=hidden,
                 num_layers=layers).to(dev)
    if dt != torch.float32:
        model = model.to(dt)
    dp = BucketedDataParallel(model) if world > 1 else None
    # Adam is the reference GNMT optimizer; SGD keeps the fused path —
    # throughput benchmarking is optimizer-agnostic, use fused SGD
    opt = FusedSGD(model.parameters(), lr=lr * world, momentum=0.9,
                   backend=kernel_backend)
    loss_fn = LabelSmoothingLoss(0.1)

    train_ds = SyntheticTranslationDataset(dataset_size, vocab,
                                           max_len=max_len, seed=seed)
    val_ds = SyntheticTranslationDataset(max(dataset_size // 10, 8),
                                         vocab, max_len=max_len,
                                         seed=seed + 1)
    sampler = BucketingSampler(train_ds, batch_size, world, rank, seed)
    log = BenchLogger(rank)

    def batches(ds, smp):
        for idx in smp:
            yield collate_translation([ds[i] for i in idx])

    def to_dev(src, src_len, tgt_in, tgt_out):
        return (src.to(dev), src_len.to(dev), tgt_in.to(dev),
                tgt_out.to(dev))

    epoch_sps, epoch_secs = [], []
    val_loss = score = 0.0
    for epoch in range(1, epochs + 1):
        sampler.set_epoch(epoch)
        model.train()
        losses = AverageMeter()
        seen = 0
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        t0 = time.perf_counter()
        wstart, wseen = t0, 0
        for i, batch in enumerate(batches(train_ds, sampler)):
            src, src_len, tgt_in, tgt_out = to_dev(*batch)
            logits = (dp or model)(src, src_len, tgt_in)
            loss = loss_fn(logits, tgt_out)
            if dp is not None:
                dp.zero_grad_buckets()
                loss.backward()
                dp.finalize_backward()
            else:
                opt.zero_grad(set_to_none=True)
                loss.backward()
            opt.step()
            bs = src.size(1)
            losses.update(loss.item(), bs)
            seen += bs
            wseen += bs
            if log_interval and (i + 1) % log_interval == 0:
                if dev.type == "cuda":
                    torch.cuda.synchronize(dev)
                sps = wseen * world / (time.perf_counter() - wstart)
                a, r, t = gpu_memory_gb(dev)
                log.train_step(epoch, epochs,
                               int(100 * (i + 1) / len(sampler)), sps,
                               a, r, t)
                wstart, wseen = time.perf_counter(), 0
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        secs = time.perf_counter() - t0
        sps = seen * world / secs

        # validation: per-token loss + greedy BLEU on a few batches
        model.eval()
        from ddlbench_amd.translation import Translator, bleu
        vls = AverageMeter()
        hyps, refs = [], []
        with torch.no_grad():
            vs = BucketingSampler(val_ds, batch_size, 1, 0, seed)
            for j, batch in enumerate(batches(val_ds, vs)):
                src, src_len, tgt_in, tgt_out = to_dev(*batch)
                vls.update(loss_fn(model(src, src_len, tgt_in),
                                   tgt_out).item(), src.size(1))
                if j < bleu_batches:
                    out = Translator(model, max_len=max_len).greedy(
                        src, src_len)
                    hyps += out.t().tolist()
                    refs += tgt_out.t().tolist()
        score = bleu(hyps, refs)
        val_loss = vls.avg
        if world > 1:
            val_loss = allreduce_mean_scalar(
                val_loss, dev if dev.type == "cuda" else None)
        epoch_sps.append(sps)
        epoch_secs.append(secs)
        log.epoch(epoch, epochs, losses.avg, sps, val_loss, score)
    avg_sps = sum(epoch_sps) / max(len(epoch_sps), 1)
    avg_secs = sum(epoch_secs) / max(len(epoch_secs), 1)
    log.final(score, avg_sps, avg_secs)
    return {"bleu": score, "samples_per_sec": avg_sps,
            "sec_per_epoch": avg_secs, "valid_loss": val_loss}
