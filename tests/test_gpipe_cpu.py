"""GPipe engine semantics on CPU: pipeline == plain sequential."""

import torch

from ddlbench_amd.parallel.pipeline.balance import partition_minmax
from ddlbench_amd.parallel.pipeline.gpipe import GPipeModel


def _seq():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(8, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 32), torch.nn.ReLU(),
        torch.nn.Linear(32, 4))


def test_partition_minmax_exact():
    assert partition_minmax([1, 1, 1, 1], 2) == [2, 2]
    assert partition_minmax([5, 1, 1, 1], 2) == [1, 3]
    assert partition_minmax([1, 1, 1, 5], 2) == [3, 1]
    sizes = partition_minmax([1.0] * 10, 3)
    assert sum(sizes) == 10 and len(sizes) == 3


def test_pipeline_matches_sequential_forward_backward():
    seq = _seq()
    ref = _seq()
    ref.load_state_dict(seq.state_dict())

    pipe = GPipeModel(seq, balance=[2, 2, 1],
                      devices=[torch.device("cpu")], chunks=4,
                      checkpoint="never")
    x = torch.randn(8, 8)
    y = pipe(x)
    y_ref = ref(x)
    torch.testing.assert_close(y, y_ref, rtol=1e-6, atol=1e-6)

    y.pow(2).sum().backward()
    y_ref.pow(2).sum().backward()
    for p, q in zip(pipe.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, q.grad, rtol=1e-5, atol=1e-6)


def test_pipeline_checkpointing_grads_match():
    seq = _seq()
    ref = _seq()
    ref.load_state_dict(seq.state_dict())
    pipe = GPipeModel(seq, balance=[3, 2], devices=[torch.device("cpu")],
                      chunks=2, checkpoint="except_last")
    pipe.train()
    x = torch.randn(4, 8)
    pipe(x).pow(2).sum().backward()
    ref(x).pow(2).sum().backward()
    for p, q in zip(pipe.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, q.grad, rtol=1e-5, atol=1e-6)


def test_uneven_chunks():
    seq = _seq()
    pipe = GPipeModel(seq, balance=[2, 3], devices=[torch.device("cpu")],
                      chunks=3, checkpoint="never")
    y = pipe(torch.randn(7, 8))  # 7 doesn't divide by 3
    assert y.shape == (7, 4)
