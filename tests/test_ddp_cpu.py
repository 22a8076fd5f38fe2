"""BucketedDataParallel semantics on gloo, world_size 2, CPU.

The multi-process analogue of the reference's horovod path — verifies
gradient averaging, parameter sync, and bucket bookkeeping without a GPU
(SURVEY.md §4 'Implication': fake-transport/multi-process CPU tests the
reference lacked)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker_avg(rank, world, port):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from ddlbench_amd.parallel import BucketedDataParallel

    torch.manual_seed(100 + rank)  # different init per rank on purpose
    model = torch.nn.Sequential(torch.nn.Linear(4, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 2))
    dp = BucketedDataParallel(model, bucket_mb=0.0001)  # force >1 bucket
    assert len(dp._buckets) > 1

    # after wrap, params must match rank 0's
    for p in model.parameters():
        ref = p.data.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(ref, p.data)

    # rank-dependent data -> grads must end up averaged
    torch.manual_seed(rank)
    x = torch.randn(8, 4)
    dp.zero_grad_buckets()
    dp(x).sum().backward()
    dp.finalize_backward()

    # reference: average of per-rank grads computed on a fresh clone
    clone = torch.nn.Sequential(torch.nn.Linear(4, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 2))
    clone.load_state_dict(model.state_dict())
    clone.zero_grad()
    clone(x).sum().backward()
    for p, q in zip(model.parameters(), clone.parameters()):
        g = q.grad.clone()
        dist.all_reduce(g)
        g /= world
        torch.testing.assert_close(p.grad, g, rtol=1e-5, atol=1e-6)

    # second iteration reuses buckets correctly
    dp.zero_grad_buckets()
    dp(x).sum().backward()
    dp.finalize_backward()
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_grad_averaging_two_ranks(free_port):
    mp.spawn(_worker_avg, args=(2, free_port), nprocs=2, join=True)


def _worker_train(rank, world, port):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from ddlbench_amd.parallel import BucketedDataParallel
    from ddlbench_amd.ops.sgd import FusedSGD

    torch.manual_seed(0)
    model = torch.nn.Linear(4, 4)
    dp = BucketedDataParallel(model, bucket_mb=1)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, backend="torch")
    torch.manual_seed(50 + rank)
    for _ in range(3):
        x = torch.randn(6, 4)
        dp.zero_grad_buckets()
        dp(x).pow(2).sum().backward()
        dp.finalize_backward()
        opt.step()
    # params must stay bit-identical across ranks
    for p in model.parameters():
        ref = p.data.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(ref, p.data), "ranks diverged"
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_params_stay_in_sync(free_port):
    mp.spawn(_worker_train, args=(2, free_port), nprocs=2, join=True)


def test_layout_view_channels_last():
    """Bucket grad views must carry the param's memory format (the
    round-1 DDP+channels_last bug: row-major views orphaned by
    FusedSGD's layout remediation)."""
    from ddlbench_amd.parallel.ddp import BucketedDataParallel

    p = torch.empty(2, 8, 3, 3).to(memory_format=torch.channels_last)
    flat = torch.zeros(p.numel())
    v = BucketedDataParallel._layout_view(flat, p)
    assert v.shape == p.shape
    assert v.stride() == p.stride()
    # writes through the view land in the flat payload
    v.copy_(torch.arange(p.numel(), dtype=torch.float32).view_as(p))
    assert flat.abs().sum() > 0
    # a contiguous param keeps a contiguous view
    q = torch.empty(4, 5)
    v2 = BucketedDataParallel._layout_view(torch.zeros(20), q)
    assert v2.stride() == q.stride()


def _worker_channels_last(rank, world, port):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from ddlbench_amd.parallel import BucketedDataParallel
    from ddlbench_amd.ops.sgd import FusedSGD

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Conv2d(8, 16, 3, padding=1, bias=False),
        torch.nn.ReLU(),
        torch.nn.Conv2d(16, 8, 1, bias=False),
    ).to(memory_format=torch.channels_last)
    dp = BucketedDataParallel(model, bucket_mb=1)
    # grad views must match param layout (strides on size-1 dims are
    # arbitrary — same rule as FusedSGD._same_dense_layout)
    from ddlbench_amd.ops.sgd import _same_dense_layout
    for p in model.parameters():
        assert p.grad is not None
        assert _same_dense_layout(p, p.grad), \
            (p.shape, p.grad.stride(), p.stride())
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9,
                   backend="torch")
    torch.manual_seed(123 + rank)
    grads_seen = []
    for it in range(3):
        x = torch.randn(2, 8, 6, 6).to(memory_format=torch.channels_last)
        dp.zero_grad_buckets()
        dp(x).pow(2).mean().backward()
        dp.finalize_backward()
        # grads must be live (non-zero) and still be the bucket views
        b = dp._param_bucket[next(model.parameters())]
        assert b.flat.abs().sum() > 0, "bucket never received grads"
        for p in model.parameters():
            assert p.grad.data_ptr() == dp._buckets[
                dp._buckets.index(dp._param_bucket[p])].views[p].data_ptr()
        grads_seen.append(
            next(model.parameters()).grad.flatten()[:4].clone())
        opt.step()
    # grads differ between iterations (i.e. no stale accumulation)
    assert not torch.equal(grads_seen[0], grads_seen[1])
    # params bit-identical across ranks
    for p in model.parameters():
        ref = p.data.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(ref, p.data), "ranks diverged (channels_last)"
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_channels_last_buckets_two_ranks(free_port):
    mp.spawn(_worker_channels_last, args=(2, free_port), nprocs=2,
             join=True)


def _worker_reduce_log(rank, world, port):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from ddlbench_amd.parallel import BucketedDataParallel

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(4, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 2))
    dp = BucketedDataParallel(model, bucket_mb=0.0001,
                              log_reduce_times=True)
    nb = len(dp._buckets)
    assert nb > 1
    dp.zero_grad_buckets()
    dp(torch.randn(8, 4)).sum().backward()
    dp.finalize_backward()
    times = dp.pop_reduce_times()
    assert len(times) == nb
    assert all(t >= 0.0 for t in times)
    assert dp.pop_reduce_times() == []  # drained
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_reduce_time_logging_two_ranks(free_port):
    """Per-bucket all-reduce spans logged and drained (the reference's
    extract_reduce_times pipeline, utils/all_reduce/)."""
    mp.spawn(_worker_reduce_log, args=(2, free_port), nprocs=2, join=True)
