"""GNMT on the MI355X: revert kernel numerics + one train step."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_revert_varlen_native_matches_torch():
    from ddlbench_amd.models.gnmt import _revert_torch
    from ddlbench_amd.ops import require_extension
    ext = require_extension()
    torch.manual_seed(0)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(12, 5, 16, device="cuda", dtype=dtype)
        lengths = torch.tensor([12, 7, 3, 1, 9], device="cuda")
        out = ext.revert_varlen(x.contiguous(), lengths)
        ref = _revert_torch(x.cpu(), lengths.cpu()).to("cuda", dtype)
        torch.testing.assert_close(out, ref)


def test_varlen_mask_native():
    from ddlbench_amd.ops import require_extension
    ext = require_extension()
    lengths = torch.tensor([3, 1, 5], device="cuda")
    m = ext.varlen_mask(lengths, 5).bool()
    assert m.shape == (5, 3)
    assert m[:, 0].tolist() == [True] * 3 + [False] * 2
    assert m[:, 2].all()


def test_gnmt_train_step_gpu():
    from ddlbench_amd.models.gnmt import GNMT, LabelSmoothingLoss
    from ddlbench_amd.ops.adam import FusedAdam
    torch.manual_seed(0)
    dev = torch.device("cuda", 0)
    model = GNMT(vocab_size=512, hidden_size=128, num_layers=4).to(dev)
    opt = FusedAdam(model.parameters(), lr=1e-3, backend="native")
    src = torch.randint(3, 512, (20, 8), device=dev)
    src_len = torch.randint(5, 21, (8,), device=dev).sort(
        descending=True).values
    tgt = torch.randint(3, 512, (18, 8), device=dev)
    logits = model(src, src_len, tgt[:-1])
    loss = LabelSmoothingLoss()(logits, tgt[1:])
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)


def test_fused_adam_native_matches_cpu_reference():
    from ddlbench_amd.ops.adam import FusedAdam
    torch.manual_seed(0)
    dev = torch.device("cuda", 0)
    a = torch.nn.Linear(16, 16).to(dev)
    b = torch.nn.Linear(16, 16).to(dev)
    b.load_state_dict(a.state_dict())
    oa = FusedAdam(a.parameters(), lr=1e-2, weight_decay=1e-2,
                   backend="native")
    ob = torch.optim.Adam(b.parameters(), lr=1e-2, weight_decay=1e-2)
    for _ in range(5):
        x = torch.randn(8, 16, device=dev)
        for m, o in ((a, oa), (b, ob)):
            o.zero_grad()
            m(x).pow(2).sum().backward()
            o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-4, atol=1e-5)
