"""FusedSGD semantics vs torch.optim.SGD (CPU reference path)."""

import torch

from ddlbench_amd.ops.sgd import FusedSGD


def _models():
    torch.manual_seed(0)
    a = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                            torch.nn.Linear(16, 4))
    b = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                            torch.nn.Linear(16, 4))
    b.load_state_dict(a.state_dict())
    return a, b


def test_matches_torch_sgd():
    a, b = _models()
    oa = FusedSGD(a.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4,
                  backend="torch")
    ob = torch.optim.SGD(b.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=1e-4)
    for step in range(5):
        x = torch.randn(4, 8)
        for m, o in ((a, oa), (b, ob)):
            o.zero_grad()
            m(x).pow(2).sum().backward()
            o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-6, atol=1e-6)


def test_no_momentum():
    a, b = _models()
    oa = FusedSGD(a.parameters(), lr=0.05, backend="torch")
    ob = torch.optim.SGD(b.parameters(), lr=0.05)
    x = torch.randn(4, 8)
    for m, o in ((a, oa), (b, ob)):
        o.zero_grad()
        m(x).sum().backward()
        o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-6, atol=1e-6)
