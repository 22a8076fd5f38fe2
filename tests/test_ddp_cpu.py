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
