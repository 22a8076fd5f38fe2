"""PipeDream-style training driver: profile -> partition -> 1F1B runtime.

One process per GPU (torchrun). Rank 0 profiles the layer chain and runs
the partitioner; the plan broadcasts to all ranks (replacing the
reference's offline profiler -> optimizer -> codegen -> bash-parsed
stage map flow, SURVEY.md §3.3 — stages are sliced from the live module
list at runtime; the conf.json contract is still written to disk).

Epoch flow mirrors main_with_runtime.py:406-500: warmup forwards, 1F1B
steady state with per-minibatch optimizer steps, backward drain,
forward-only validation with ack clocking."""

from __future__ import annotations

import os
import time

import torch
import torch.distributed as dist

from ddlbench_amd.config import BenchConfig
from ddlbench_amd.data import SyntheticImageDataset
from ddlbench_amd.engine import compute_dtype, resolve_device
from ddlbench_amd.models import build_sequential
from ddlbench_amd.ops import functional as NF
from ddlbench_amd.ops.modules import set_default_backend
from ddlbench_amd.ops.sgd import FusedSGD
from ddlbench_amd.parallel import BucketedDataParallel, init_distributed
from ddlbench_amd.parallel.pipeline.comm import (PipelineTransport,
                                                 dry_run_shapes)
from ddlbench_amd.parallel.pipeline.partition import (PartitionResult,
                                                      partition_chain)
from ddlbench_amd.parallel.pipeline.profiler import profile_sequential
from ddlbench_amd.parallel.pipeline.runtime import StagePlan, StageRuntime
from ddlbench_amd.parallel.pipeline.stash import VersionedOptimizer
from ddlbench_amd.utils import AverageMeter, BenchLogger, accuracy


def _make_plan(cfg: BenchConfig, seq, device, world: int):
    """Rank 0 profiles + partitions; everyone gets the same plan."""
    straight = cfg.straight_pipeline
    rank = dist.get_rank() if dist.is_initialized() else 0
    payload = [None]
    if rank == 0:
        from ddlbench_amd.data import synthetic_batch
        sample, _ = synthetic_batch(cfg, batch_size=min(cfg.batch_size, 8),
                                    device=device,
                                    dtype=compute_dtype(cfg))
        graph = profile_sequential(seq, sample, device=device,
                                   iters=3, warmup=1)
        result = partition_chain(graph, world, straight=straight)
        prof_dir = os.path.join(
            os.environ.get("DDLB_PROFILE_DIR", "profiles"), cfg.arch)
        os.makedirs(prof_dir, exist_ok=True)
        graph.save(os.path.join(prof_dir, "graph.txt"))
        result.save(os.path.join(prof_dir, "conf.json"))
        print(result.describe(), flush=True)  # the reference's split
        # analysis stdout (optimizer_graph_hierarchical.py:169-191)
        payload = [{
            "module_to_stage_map": result.module_to_stage_map,
            "replicas": [s.replicas for s in result.stages],
        }]
    if dist.is_initialized():
        dist.broadcast_object_list(payload, src=0)
    return payload[0]


def run_1f1b_training(cfg: BenchConfig) -> dict:
    set_default_backend(cfg.kernel_backend)
    env = init_distributed()
    world = env.world_size
    device = resolve_device(cfg, env.local_rank)
    dtype = compute_dtype(cfg)

    torch.manual_seed(cfg.seed)  # identical init everywhere
    seq = build_sequential(cfg.dataset, cfg.arch)
    if dtype != torch.float32:
        seq = seq.to(dtype)

    plan_info = _make_plan(cfg, seq, device, world)
    m2s = plan_info["module_to_stage_map"]
    plan = StagePlan(replicas=plan_info["replicas"])
    assert plan.world_size == world, (plan.replicas, world)

    stage, replica = plan.stage_of_rank(env.rank)
    my_layers = [i for i, s in enumerate(m2s) if s == stage]
    stage_mod = torch.nn.Sequential(
        *[seq[i] for i in my_layers]).to(device)

    # static shapes from a tiny CPU dry run, batch dim patched to B
    stage_slices = []
    off = 0
    for s in range(plan.num_stages):
        layers = [i for i, t in enumerate(m2s) if t == s]
        stage_slices.append(torch.nn.Sequential(*[seq[i] for i in layers]))
    # profiling moved `seq` (and thus the slices) onto `device` on rank 0,
    # so the probe must follow it
    probe = torch.zeros((2,) + tuple(cfg.shape), dtype=dtype)
    probe_dev = next(seq.parameters()).device
    shapes = dry_run_shapes(stage_slices, probe, device=probe_dev)
    B = cfg.batch_size

    def with_batch(shape):
        return torch.Size((B,) + tuple(shape)[1:])

    in_shape = (None if stage == 0
                else with_batch(shapes[stage - 1]))
    out_shape = with_batch(shapes[stage])

    backend = dist.get_backend() if dist.is_initialized() else "gloo"
    transport = PipelineTransport(plan.edges(), backend)

    # per-stage DP groups (every rank participates in every new_group)
    dp = None
    for s in range(plan.num_stages):
        ranks = plan.stage_ranks(s)
        if len(ranks) > 1:
            g = dist.new_group(ranks)
            if s == stage:
                dp = BucketedDataParallel(stage_mod, process_group=g)

    loss_fn = (lambda out, tgt: NF.cross_entropy(
        out, tgt, backend=cfg.kernel_backend))
    rt = StageRuntime(plan, env.rank, stage_mod, transport, in_shape,
                      out_shape, device, dtype, loss_fn=loss_fn,
                      dp_wrapper=dp)

    # per-stage LR scaling for replicated stages (reference
    # runtime.py:695-701: DP-style x-replicas within a stage)
    stage_lr = cfg.lr * plan.replicas[stage]
    opt = VersionedOptimizer(
        FusedSGD(stage_mod.parameters(), lr=stage_lr,
                 momentum=cfg.momentum,
                 weight_decay=cfg.weight_decay,
                 backend=cfg.kernel_backend),
        versioned=(not cfg.no_input_pipelining
                   and plan.num_warmup(stage) > 0))

    # deterministic synthetic streams: stage 0 reads inputs, last stage
    # reads the matching targets (no label transport needed)
    train_ds = SyntheticImageDataset(cfg.dataset, train=True,
                                     seed=cfg.seed,
                                     scale=cfg.synthetic_scale)
    test_ds = SyntheticImageDataset(cfg.dataset, train=False,
                                    seed=cfg.seed,
                                    scale=cfg.synthetic_scale)

    def make_providers(ds, epoch):
        n_mb = len(ds) // B
        lcm = plan.lcm_replicas()
        n_mb = max((n_mb // lcm) * lcm, 0)
        if device.type == "cuda":
            # device-resident synthetic stream: stage 0 (inputs) and the
            # last stage (targets) derive the SAME minibatch from the
            # same (seed, epoch, mb) — no CPU dataset, no H2D copies
            c, h, w = cfg.shape

            def input_provider(mb):
                g = torch.Generator(device=device).manual_seed(
                    cfg.seed * 977 + epoch * 65537 + mb)
                return torch.randn((B, c, h, w), generator=g,
                                   device=device, dtype=dtype)

            def target_provider(mb):
                g = torch.Generator(device=device).manual_seed(
                    cfg.seed * 977 + epoch * 65537 + mb + 1)
                return torch.randint(cfg.num_classes, (B,), generator=g,
                                     device=device)

            return n_mb, input_provider, target_provider
        g = torch.Generator().manual_seed(cfg.seed * 977 + epoch)
        perm = torch.randperm(len(ds), generator=g)

        def input_provider(mb):
            idx = perm[mb * B:(mb + 1) * B]
            return torch.stack([ds[i.item()][0] for i in idx])

        def target_provider(mb):
            idx = perm[mb * B:(mb + 1) * B]
            return torch.tensor([ds[i.item()][1] for i in idx])

        return n_mb, input_provider, target_provider

    is_output = rt.is_last and replica == 0
    log = BenchLogger(0 if is_output else 1)

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    def train_epoch(epoch):
        stage_mod.train()
        n_mb, inp, tgt = make_providers(train_ds, epoch)
        mbs = rt.my_minibatches(n_mb)
        # --no_input_pipelining: pure model parallel, one minibatch in
        # flight (reference main_with_runtime.py:66-67, 233-235)
        warmup = (0 if cfg.no_input_pipelining
                  else min(plan.num_warmup(stage), len(mbs)))
        loss_vals = []  # detached tensors; .item() deferred to epoch end
        if dist.is_initialized():
            dist.barrier()
        sync()
        t0 = time.perf_counter()
        for k in range(warmup):
            loss, _ = rt.run_forward(mbs[k], inp, tgt, training=True)
            if loss is not None:
                loss_vals.append(loss.detach())
        window_start, window_n = time.perf_counter(), 0
        for k in range(len(mbs)):
            if warmup + k < len(mbs):
                loss, _ = rt.run_forward(mbs[warmup + k], inp, tgt,
                                         training=True)
                if loss is not None:
                    loss_vals.append(loss.detach())
            if dp is not None:
                dp.zero_grad_buckets()
            else:
                opt.zero_grad(set_to_none=False)
            rt.run_backward()
            if dp is not None:
                dp.finalize_backward()
            opt.step()
            window_n += 1
            if (cfg.log_interval and is_output
                    and (k + 1) % cfg.log_interval == 0):
                sync()
                now = time.perf_counter()
                from ddlbench_amd.utils import gpu_memory_gb
                sps_w = (window_n * B * plan.replicas[-1]
                         / (now - window_start))
                a, r, t = gpu_memory_gb(device)
                log.train_step(epoch, cfg.epochs,
                               int(100 * (k + 1) / len(mbs)), sps_w,
                               a, r, t)
                window_start, window_n = time.perf_counter(), 0
        sync()
        if dist.is_initialized():
            dist.barrier()
        elapsed = time.perf_counter() - t0
        sps = n_mb * B / max(elapsed, 1e-9)
        avg_loss = (torch.stack(loss_vals).mean().item()
                    if loss_vals else 0.0)
        return avg_loss, sps, elapsed

    @torch.no_grad()
    def validate():
        stage_mod.eval()
        n_mb, inp, tgt = make_providers(test_ds, 0)
        mbs = rt.my_minibatches(n_mb)
        window = plan.num_stages
        losses = AverageMeter()
        accs = AverageMeter()
        outstanding = 0
        for k, mb in enumerate(mbs):
            loss, extras = rt.run_forward(mb, inp, tgt, training=False)
            # per-hop credit: ack upstream as soon as mb is consumed
            if not rt.is_first:
                rt.send_ack(mb)
            if rt.is_last:
                y = extras["target"]
                out = extras["output"]
                losses.update(loss.item(), y.numel())
                accs.update(accuracy(out, y)[0], y.numel())
                rt.pop_eval()
            else:
                outstanding += 1
                if outstanding >= window:
                    mb_old, _ = rt.pop_eval()
                    rt.recv_ack(mb_old)
                    outstanding -= 1
        while outstanding > 0 and not rt.is_last:
            mb_old, _ = rt.pop_eval()
            rt.recv_ack(mb_old)
            outstanding -= 1
        if dist.is_initialized():
            dist.barrier()
        return losses.avg, accs.avg

    start_epoch = 1
    if cfg.resume and cfg.checkpoint_dir:
        from ddlbench_amd.utils.checkpoint import load_stage_checkpoint
        state = load_stage_checkpoint(cfg.checkpoint_dir, stage, stage_mod,
                                      opt, map_location=device)
        if state is not None:
            start_epoch = state["epoch"] + 1

    # per-epoch LR adjustment (reference main_with_runtime.py:441
    # adjust_learning_rate; applies over the stage-scaled base LR)
    from ddlbench_amd.utils.lr import apply_lr, make_lr_schedule
    lr_sched = make_lr_schedule(cfg.lr_schedule, world_size=world,
                                warmup_epochs=cfg.warmup_epochs)
    base_lrs = [g["lr"] for g in opt.inner.param_groups]

    epoch_sps, epoch_secs = [], []
    val_loss = val_acc = 0.0
    train_loss = 0.0
    for epoch in range(start_epoch, cfg.epochs + 1):
        apply_lr(opt.inner, base_lrs, lr_sched(epoch, 0.0))
        train_loss, sps, secs = train_epoch(epoch)
        val_loss, val_acc = validate()
        epoch_sps.append(sps)
        epoch_secs.append(secs)
        log.epoch(epoch, cfg.epochs, train_loss, sps, val_loss, val_acc)
        # per-stage checkpoint by rank_in_stage 0 (reference
        # main_with_runtime.py:393-403)
        if cfg.checkpoint_dir and replica == 0:
            from ddlbench_amd.utils.checkpoint import save_stage_checkpoint
            save_stage_checkpoint(cfg.checkpoint_dir, stage, epoch,
                                  cfg.arch, stage_mod, opt, val_acc)
    avg_sps = sum(epoch_sps) / max(len(epoch_sps), 1)
    avg_secs = sum(epoch_secs) / max(len(epoch_secs), 1)
    log.final(val_acc, avg_sps, avg_secs)
    return {"valid_accuracy": val_acc, "samples_per_sec": avg_sps,
            "sec_per_epoch": avg_secs, "stage": stage,
            "replicas": plan.replicas}
