"""Weight versioning for asynchronous 1F1B pipelining.

PipeDream requirement (/root/reference/pipedream-fork/runtime/optimizer.py:
58-116): the backward pass of minibatch i must use the SAME weight
version its forward used, although the optimizer steps (warmup-depth)
times in between. The reference implements this with an explicit deque of
cloned state_dicts swapped in before every backward (load_old_params /
load_new_params).

MI355X-native mechanism — copy-on-step: PyTorch autograd already saves
references to the exact weight tensors a forward used; what breaks
consistency is the optimizer's *in-place* update mutating those saved
storages (verified: an in-place `p.data.add_` leaks the update into an
in-flight backward, while rebinding `p.data = fresh` does not). So
VersionedOptimizer rebinds every parameter to a fresh clone right before
the fused SGD kernel updates it:

  * backward of minibatch i automatically computes with i's forward-time
    weights — no swap bookkeeping, no load_old/load_new;
  * old versions stay alive exactly as long as an in-flight backward
    references them (stash depth == warmup+1, enforced by autograd's own
    lifetime tracking, not by a manual deque);
  * gradients always apply to the LATEST weights, like the reference.
"""

from __future__ import annotations

import torch


class VersionedOptimizer:
    """Wraps an optimizer so each step writes a NEW weight storage.

    Use for any pipelined stage with in-flight minibatches. The wrapped
    optimizer may update in place (our fused SGD kernel does)."""

    def __init__(self, optimizer: torch.optim.Optimizer,
                 versioned: bool = True):
        self.inner = optimizer
        self.versioned = versioned

    def zero_grad(self, set_to_none: bool = True) -> None:
        self.inner.zero_grad(set_to_none=set_to_none)

    def step(self) -> None:
        if self.versioned:
            for group in self.inner.param_groups:
                for p in group["params"]:
                    if p.grad is not None:
                        p.data = p.data.clone()
            # fused kernels cache param data_ptrs — rebinding moved them
            cache = getattr(self.inner, "_cache", None)
            if cache is not None:
                cache.clear()
        self.inner.step()

    @property
    def param_groups(self):
        return self.inner.param_groups
