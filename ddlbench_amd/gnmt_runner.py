"""GNMT training/eval runner (single-device or DP over RCCL).

The reference runs GNMT only through the pipedream driver
(SURVEY.md §2.12); here the translation workload is a first-class
benchmark: teacher-forced training with label smoothing, per-token loss,
validation loss + greedy-decode BLEU, with the standard result-log
contract (samples/sec = sentences/sec)."""

from __future__ import annotations

import time

import torch

from ddlbench_amd.engine import resolve_device
from ddlbench_amd.models.gnmt import GNMT, LabelSmoothingLoss
from ddlbench_amd.data.translation import (BucketingSampler,
                                           SyntheticTranslationDataset,
                                           collate_translation)
from ddlbench_amd.ops.sgd import FusedSGD
from ddlbench_amd.utils import AverageMeter, BenchLogger, gpu_memory_gb


def run_gnmt(epochs=3, batch_size=64, dataset_size=2000, vocab=32320,
             hidden=1024, layers=4, lr=0.25e-3, dtype="float32",
             device="auto", log_interval=25, seed=42, max_len=50,
             ddp=False, kernel_backend="auto", bleu_batches=2,
             data_dir="", optimizer="sgd") -> dict:
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

    if data_dir:
        # real parallel corpus (reference seq2seq/data position):
        # <data_dir>/{train,val}.{src,tgt}; vocab from the train split
        from ddlbench_amd.data.tokenizer import TextTranslationDataset
        train_ds = TextTranslationDataset(data_dir, "train",
                                          max_len=max_len)
        try:
            val_ds = TextTranslationDataset(
                data_dir, "val", tokenizer_src=train_ds.tok_src,
                tokenizer_tgt=train_ds.tok_tgt, max_len=max_len)
        except FileNotFoundError:
            val_ds = train_ds
        vocab = max(len(train_ds.tok_src), len(train_ds.tok_tgt), 8)
    else:
        train_ds = SyntheticTranslationDataset(dataset_size, vocab,
                                               max_len=max_len,
                                               seed=seed)
        val_ds = SyntheticTranslationDataset(max(dataset_size // 10, 8),
                                             vocab, max_len=max_len,
                                             seed=seed + 1)

    model = GNMT(vocab_size=vocab, hidden_size=hidden,
                 num_layers=layers).to(dev)
    if dt != torch.float32:
        model = model.to(dt)
    dp = BucketedDataParallel(model) if world > 1 else None
    # Adam is the reference GNMT optimizer (runtime/translation
    # main_with_runtime.py position); both run the fused multi-tensor
    # step — SGD stays the throughput default
    if optimizer == "adam":
        from ddlbench_amd.ops.adam import FusedAdam
        opt = FusedAdam(model.parameters(), lr=lr * world,
                        backend=kernel_backend)
    else:
        opt = FusedSGD(model.parameters(), lr=lr * world, momentum=0.9,
                       backend=kernel_backend)
    loss_fn = LabelSmoothingLoss(0.1)

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
            # eval shards statically across ranks (reference
            # StaticDistributedSampler); metrics are rank-local, rank 0
            # logs its shard
            from ddlbench_amd.data.translation import (
                StaticDistributedSampler)
            vs = StaticDistributedSampler(val_ds, batch_size, world,
                                          rank)
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


def run_gnmt_pipeline(epochs=1, batch_size=32, n_minibatches=16,
                      vocab=32320, hidden=1024, layers=4, lr=2.5e-4,
                      dtype="float32", device="auto", seed=42,
                      src_len_max=48, tgt_len=48, log_interval=0,
                      data_dir="") -> dict:
    """GNMT through the 1F1B pipeline — the reference's
    translation/main_with_runtime.py flow on the
    tuple-I/O StageRuntime. Straight pipeline, one stage per rank;
    fixed (padded) sequence lengths give static edge shapes.

    data_dir: parallel corpus root (<dir>/train.{src,tgt}) — batches
    are padded/truncated to the fixed (src_len_max, tgt_len) shapes the
    static pipeline edges need; every rank builds the same dataset and
    epoch permutation, so the first stage's inputs and the last stage's
    targets line up without a data channel (the reference ships the
    target through the pipeline instead; runtime.py:540-543)."""
    import torch.distributed as dist

    from ddlbench_amd.models.gnmt import (GNMT, LabelSmoothingLoss,
                                          gnmt_edge_specs,
                                          gnmt_pipeline_units)
    from ddlbench_amd.parallel import init_distributed
    from ddlbench_amd.parallel.pipeline.comm import PipelineTransport
    from ddlbench_amd.parallel.pipeline.balance import (partition_minmax,
                                                        profile_unit_times)
    from ddlbench_amd.parallel.pipeline.runtime import (StagePlan,
                                                        StageRuntime)
    from ddlbench_amd.parallel.pipeline.stash import VersionedOptimizer

    env = init_distributed()
    world = env.world_size

    class _C:
        pass

    cfg = _C()
    cfg.device = device
    dev = resolve_device(cfg, env.local_rank)
    dt = torch.bfloat16 if dtype == "bfloat16" else torch.float32

    ds = None
    if data_dir:
        from ddlbench_amd.data.tokenizer import TextTranslationDataset
        ds = TextTranslationDataset(
            data_dir, "train", max_len=max(src_len_max, tgt_len + 1))
        vocab = max(len(ds.tok_src), len(ds.tok_tgt), 8)

    torch.manual_seed(seed)
    model = GNMT(vocab_size=vocab, hidden_size=hidden, num_layers=layers,
                 dropout=0.0)
    if dt != torch.float32:
        model = model.to(dt)
    units = gnmt_pipeline_units(model)
    assert world <= len(units), \
        f"at most {len(units)} pipeline stages for this GNMT config"
    specs = gnmt_edge_specs(model, len(units), src_len_max, tgt_len,
                            batch_size, dt)

    # profiled stage balance: rank 0 times each unit on one minibatch
    # shape and broadcasts the split (the 1F1B image runner's
    # profile->partition->broadcast flow, here over tuple units)
    if world > 1 and dist.is_initialized():
        payload = [None]
        if env.rank == 0:
            if dev.type == "cuda":
                for u in units:
                    u.to(dev)  # profile on the real device
            g0 = torch.Generator().manual_seed(seed)
            sample = (torch.randint(3, vocab, (src_len_max, batch_size),
                                    generator=g0),
                      torch.full((batch_size,), src_len_max,
                                 dtype=torch.long),
                      torch.randint(3, vocab, (tgt_len, batch_size),
                                    generator=g0))
            times = profile_unit_times(units, sample)
            payload = [partition_minmax(times, world)]
        dist.broadcast_object_list(payload, src=0)
        sizes = payload[0]
    else:
        sizes = partition_minmax([1.0] * len(units), world)
    if env.rank == 0 and dev.type == "cuda":
        # release the units that belong to other stages
        b0 = sum(sizes[:env.rank])
        for i, u in enumerate(units):
            if not (b0 <= i < b0 + sizes[env.rank]):
                u.to("cpu")
        torch.cuda.empty_cache()
    bounds = [0]
    for sz in sizes:
        bounds.append(bounds[-1] + sz)
    plan = StagePlan(replicas=[1] * world)
    stage = env.rank
    my_units = units[bounds[stage]:bounds[stage + 1]]

    class _Tuple(torch.nn.Module):
        def __init__(self, mods):
            super().__init__()
            self.mods = torch.nn.ModuleList(mods)

        def forward(self, *xs):
            for m in self.mods:
                out = m(*xs)
                xs = (out,) if torch.is_tensor(out) else out
            return xs

    stage_mod = _Tuple(my_units).to(dev)
    transport = PipelineTransport(plan.edges(),
                                  dist.get_backend() if world > 1
                                  else "gloo")
    in_specs = None if stage == 0 else specs[bounds[stage] - 1]
    out_specs = specs[bounds[stage + 1] - 1]
    loss_fn = LabelSmoothingLoss()
    rt = StageRuntime(plan, env.rank, stage_mod, transport, in_specs,
                      out_specs, dev, dt,
                      loss_fn=lambda out, tgt: loss_fn(out, tgt))
    opt = VersionedOptimizer(
        FusedSGD(stage_mod.parameters(), lr=lr, momentum=0.9),
        versioned=plan.num_warmup(stage) > 0)

    def providers(epoch):
        if ds is not None:
            # real corpus: identical epoch permutation on every rank,
            # fixed-shape padding (PAD=0) for the static edges
            g = torch.Generator().manual_seed(seed * 131 + epoch * 7919)
            perm = torch.randperm(len(ds), generator=g).tolist()

            def fixed(mb):
                start = (mb * batch_size) % len(ds)
                src = torch.zeros(src_len_max, batch_size,
                                  dtype=torch.long)
                src_len = torch.ones(batch_size, dtype=torch.long)
                tgt_full = torch.zeros(tgt_len + 1, batch_size,
                                       dtype=torch.long)
                for b in range(batch_size):
                    s, t = ds[perm[(start + b) % len(ds)]]
                    s, t = s[:src_len_max], t[:tgt_len + 1]
                    src[:len(s), b] = s
                    src_len[b] = len(s)
                    tgt_full[:len(t), b] = t
                return src, src_len, tgt_full

            def input_provider(mb):
                src, src_len, tf = fixed(mb)
                return src, src_len, tf[:-1]

            def target_provider(mb):
                return fixed(mb)[2][1:]

            return input_provider, target_provider

        def tgt_full(mb):
            g = torch.Generator().manual_seed(
                seed * 131 + epoch * 7919 + mb)
            return torch.randint(3, vocab, (tgt_len + 1, batch_size),
                                 generator=g)

        def input_provider(mb):
            g = torch.Generator().manual_seed(
                seed * 131 + epoch * 7919 + mb + 500_009)
            src = torch.randint(3, vocab, (src_len_max, batch_size),
                                generator=g)
            src_len = torch.randint(src_len_max // 2, src_len_max + 1,
                                    (batch_size,), generator=g)
            return src, src_len, tgt_full(mb)[:-1]

        def target_provider(mb):
            return tgt_full(mb)[1:]

        return input_provider, target_provider

    log = BenchLogger(0 if rt.is_last else 1)
    epoch_sps = []
    last_loss = 0.0
    for epoch in range(1, epochs + 1):
        inp, tgt = providers(epoch)
        mbs = rt.my_minibatches(n_minibatches)
        warmup = min(plan.num_warmup(stage), len(mbs))
        losses = []
        if dist.is_initialized():
            dist.barrier()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        t0 = time.perf_counter()
        for k in range(warmup):
            loss, _ = rt.run_forward(mbs[k], inp, tgt, training=True)
            if loss is not None:
                losses.append(loss.detach())
        for k in range(len(mbs)):
            if warmup + k < len(mbs):
                loss, _ = rt.run_forward(mbs[warmup + k], inp, tgt,
                                         training=True)
                if loss is not None:
                    losses.append(loss.detach())
            opt.zero_grad(set_to_none=False)
            rt.run_backward()
            opt.step()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        if dist.is_initialized():
            dist.barrier()
        secs = time.perf_counter() - t0
        sps = n_minibatches * batch_size / secs
        epoch_sps.append(sps)
        last_loss = (torch.stack(losses).mean().item() if losses else 0.0)
        log.epoch(epoch, epochs, last_loss, sps, last_loss, 0.0)
    avg = sum(epoch_sps) / max(len(epoch_sps), 1)
    log.final(0.0, avg, 0.0)
    return {"samples_per_sec": avg, "train_loss": last_loss,
            "stage": stage, "num_units": len(units)}
