"""Hand-written CDNA4 HIP kernels and their dispatch layer.

The native extension (``ddlbench_amd/ops/_hip_ops.so``) is built in-tree
for gfx950 only (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``). Policy:

* On a GPU box the native kernels ARE the compute path. If the extension
  is missing there, ops raise instead of silently falling back to eager
  PyTorch (``kernel_backend="torch"`` opts out explicitly).
* On CPU (no HIP device) every op has a plain-PyTorch reference
  implementation — that is what the numerics tests compare against.
"""

from __future__ import annotations


_ext = None
_ext_err: Exception | None = None


def _load_extension():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        import torch  # noqa: F401 — loads libc10/libtorch for the ext
        from ddlbench_amd.ops import _hip_ops  # built in-tree .so
        _ext = _hip_ops
    except ImportError as e:
        _ext_err = e
    return _ext


def extension():
    """The native module, or None when unavailable."""
    return _load_extension()


def extension_available() -> bool:
    return _load_extension() is not None


def require_extension():
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "ddlbench_amd native HIP extension (_hip_ops) is not built; "
            "run `python setup.py build_ext --inplace` (gfx950). "
            f"Original import error: {_ext_err}")
    return ext


def use_native(tensor_or_device, backend: str = "auto") -> bool:
    """Decide native-vs-torch dispatch for an op.

    native  -> always (raises later if extension missing)
    torch   -> never
    auto    -> native iff the tensor lives on a HIP device. On a GPU a
               missing extension is a hard error (no silent eager
               fallback — the framework's GPU compute path is the HIP
               kernels)."""
    import torch
    if backend == "torch":
        return False
    dev = (tensor_or_device.device
           if isinstance(tensor_or_device, torch.Tensor) else tensor_or_device)
    on_gpu = dev.type == "cuda"
    if backend == "native":
        require_extension()
        return True
    if on_gpu:
        require_extension()
        return True
    return False
