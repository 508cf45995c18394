"""Op dispatch: CPU → torch reference impls, GPU → gfx950 HIP kernels.

On a GPU box the HIP extension is REQUIRED for ops that have a native
implementation — a silent PyTorch fallback on GPU would defeat the whole
point of the framework, so dispatch raises if a CUDA tensor arrives and the
extension is missing.  Ops without a native kernel yet are listed in
`_TORCH_OK_ON_GPU` and run through torch(+MIOpen/hipBLASLt) until their
HIP kernel lands.
"""

from __future__ import annotations

import functools
import os
from typing import Callable, Dict

import torch

from . import reference

_native_mod = None
_native_checked = False
_native_err = None


def native():
    """Return the loaded HIP extension module, or None on CPU-only hosts."""
    global _native_mod, _native_checked, _native_err
    if not _native_checked:
        _native_checked = True
        try:
            from . import native_loader
            _native_mod = native_loader.load()
        except Exception as e:
            _native_mod = None
            _native_err = e
    if _native_err is not None and torch.cuda.is_available() and \
            not os.environ.get("COS_AMD_ALLOW_EAGER_FALLBACK"):
        raise RuntimeError(
            f"gfx950 HIP extension failed to load: {_native_err}")
    return _native_mod


def native_available() -> bool:
    return native() is not None


class _Dispatcher:
    """Per-op dispatch table.  GPU entries are registered by the native
    loader; reference impls serve the CPU path."""

    def __init__(self):
        self.gpu_impls: Dict[str, Callable] = {}

    def register_gpu(self, name: str, fn: Callable) -> None:
        self.gpu_impls[name] = fn


dispatcher = _Dispatcher()

# Ops that are allowed to run through torch on GPU because their native
# kernel hasn't landed yet.  Shrink this list as kernels land.
_TORCH_OK_ON_GPU = set()

# Activation/compute ops of the DECLARED fp32 GPU path (`-dtype fp32`):
# the hand-written CDNA4 kernels are bf16-compute/fp32-accumulate; exact
# fp32 Caffe numerics instead run the reference torch impls on ROCm
# (rocBLAS GEMMs, MIOpen conv — library calls, not hidden kernel
# fallbacks; BASELINE.md states the per-config dtype policy).  Solver
# update ops are excluded: the fused fp32-arena update kernels serve both
# dtypes natively.
_FP32_REF_ON_GPU = {
    "conv2d_forward", "conv2d_backward", "fc_forward", "fc_backward",
    "relu_forward", "relu_backward", "sigmoid_forward", "sigmoid_backward",
    "tanh_forward", "tanh_backward", "maxpool_forward", "maxpool_backward",
    "avgpool_forward", "avgpool_backward", "global_avgpool_forward",
    "global_avgpool_backward", "lrn_forward", "lrn_backward",
    "softmax_forward", "softmax_backward", "softmax_loss_forward",
    "softmax_loss_backward", "dropout_forward", "dropout_backward",
    "embed_forward", "embed_backward", "lstm_unit_forward",
    "lstm_unit_backward", "bn_forward_train", "bn_forward_infer",
    "bn_backward", "accuracy", "bias_add",
}

# Ops whose fp32 routing can ALSO be decided by dtype sniffing when no net
# has declared a mode (direct op calls in tests/pycaffe).  Backward ops
# that consume an opaque forward context are excluded: the bf16 native
# forward packs fp32 tensors into its context, so dtype alone cannot tell
# the paths apart — those rely on the net-declared mode below.
_FP32_SNIFF_OK = _FP32_REF_ON_GPU - {"softmax_loss_backward",
                                     "lstm_unit_backward",
                                     "maxpool_backward", "bn_backward"}

# Net.forward/backward declare their compute mode here ("fp32" routes the
# _FP32_REF_ON_GPU set to the reference impls; "bf16" forces native; None
# = no net active, fall back to dtype sniffing)
_ACTIVE_GPU_MODE = None


def set_active_gpu_mode(mode):
    global _ACTIVE_GPU_MODE
    _ACTIVE_GPU_MODE = mode


def _args_on_gpu(args) -> bool:
    for a in args:
        if isinstance(a, torch.Tensor):
            if a.is_cuda:
                return True
        elif isinstance(a, (tuple, list)):
            for b in a:
                if isinstance(b, torch.Tensor) and b.is_cuda:
                    return True
    return False


def _any_bf16(args) -> bool:
    for a in args:
        if isinstance(a, torch.Tensor):
            if a.dtype == torch.bfloat16:
                return True
        elif isinstance(a, (tuple, list)):
            for b in a:
                if isinstance(b, torch.Tensor) and b.dtype == torch.bfloat16:
                    return True
    return False


def _dispatch(name: str):
    ref_fn = getattr(reference, name)

    @functools.wraps(ref_fn)
    def wrapper(*args, **kwargs):
        on_gpu = _args_on_gpu(args)
        if on_gpu:
            native()  # force extension load (raises loudly if missing)
            if name in _FP32_REF_ON_GPU and (
                    _ACTIVE_GPU_MODE == "fp32"
                    or (_ACTIVE_GPU_MODE is None
                        and name in _FP32_SNIFF_OK
                        and not _any_bf16(args))):
                # declared fp32 GPU path (see _FP32_REF_ON_GPU docstring)
                return ref_fn(*args, **kwargs)
            fn = dispatcher.gpu_impls.get(name)
            if fn is not None:
                return fn(*args, **kwargs)
            if name not in _TORCH_OK_ON_GPU and not os.environ.get(
                    "COS_AMD_ALLOW_EAGER_FALLBACK"):
                raise RuntimeError(
                    f"op {name!r} has no gfx950 native implementation and is "
                    "not whitelisted for torch fallback on GPU")
        return ref_fn(*args, **kwargs)

    return wrapper


# public op surface — one callable per op, same signatures as reference.py
conv2d_forward = _dispatch("conv2d_forward")
conv2d_backward = _dispatch("conv2d_backward")
fc_forward = _dispatch("fc_forward")
fc_backward = _dispatch("fc_backward")
relu_forward = _dispatch("relu_forward")
relu_backward = _dispatch("relu_backward")
sigmoid_forward = _dispatch("sigmoid_forward")
sigmoid_backward = _dispatch("sigmoid_backward")
tanh_forward = _dispatch("tanh_forward")
tanh_backward = _dispatch("tanh_backward")
maxpool_forward = _dispatch("maxpool_forward")
maxpool_backward = _dispatch("maxpool_backward")
avgpool_forward = _dispatch("avgpool_forward")
avgpool_backward = _dispatch("avgpool_backward")
global_avgpool_forward = _dispatch("global_avgpool_forward")
global_avgpool_backward = _dispatch("global_avgpool_backward")
lrn_forward = _dispatch("lrn_forward")
lrn_backward = _dispatch("lrn_backward")
softmax_forward = _dispatch("softmax_forward")
softmax_backward = _dispatch("softmax_backward")
softmax_loss_forward = _dispatch("softmax_loss_forward")
softmax_loss_backward = _dispatch("softmax_loss_backward")
dropout_forward = _dispatch("dropout_forward")
dropout_backward = _dispatch("dropout_backward")
embed_forward = _dispatch("embed_forward")
embed_backward = _dispatch("embed_backward")
lstm_unit_forward = _dispatch("lstm_unit_forward")
lstm_unit_backward = _dispatch("lstm_unit_backward")
bn_forward_train = _dispatch("bn_forward_train")
bn_forward_infer = _dispatch("bn_forward_infer")
bn_backward = _dispatch("bn_backward")
accuracy = _dispatch("accuracy")
bias_add = _dispatch("bias_add")
sgd_update = _dispatch("sgd_update")
nesterov_update = _dispatch("nesterov_update")
adam_update = _dispatch("adam_update")


def gpu_op(name):
    """GPU-only op (no CPU reference), or None if unavailable."""
    native()
    return dispatcher.gpu_impls.get(name)
