"""Native ops layer.

The HIP/CDNA4 extension (_heterofl_hip) provides hand-written gfx950 kernels
for the hot path: MFMA GEMM (+im2col conv), fused static-BatchNorm, Scaler /
ReLU fusions, masked cross-entropy, the fused clip+momentum-SGD optimizer
step, and the federation pack/unpack + combine kernels.

Policy: on a CUDA/ROCm device the native extension is REQUIRED — ops raise
if it is missing (no silent eager fallback on GPU).  On CPU the pure-torch
reference implementations run (they are also the numerics oracle in tests).
Set HETEROFL_FORCE_EAGER=1 to force the torch path on GPU (debug only).
"""
import os

import torch

_ext = None
_ext_err = None


def _load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from . import _heterofl_hip as ext  # built in-tree by setup.py
        _ext = ext
    except ImportError as e:
        _ext_err = e
    return _ext


def native_available():
    return _load() is not None


def require_native():
    """Call on the GPU path: fail loudly if the extension is missing."""
    if os.environ.get('HETEROFL_FORCE_EAGER') == '1':
        return None
    ext = _load()
    if ext is None:
        raise RuntimeError(
            f'heterofl_amd native HIP extension is not built '
            f'(import error: {_ext_err}). Build it with '
            f'`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950).')
    return ext


def use_native(tensor_or_device=None):
    """True when running on a ROCm device and the extension should be used."""
    if os.environ.get('HETEROFL_FORCE_EAGER') == '1':
        return False
    if isinstance(tensor_or_device, torch.Tensor):
        is_cuda = tensor_or_device.is_cuda
    elif tensor_or_device is not None:
        is_cuda = torch.device(tensor_or_device).type == 'cuda'
    else:
        is_cuda = torch.cuda.is_available()
    return is_cuda


def native_conv_enabled():
    """The MFMA conv path can be toggled off for A/B benchmarking."""
    return os.environ.get('HETEROFL_NATIVE_CONV', '1') == '1'


_FP8 = False


def set_fp8(on):
    """fp8 local-training mode (BASELINE config 5): the conv GEMMs run on
    CDNA4 fp8 MFMA (e4m3 weights/activations x e5m2 grads), tensors stay
    bf16 in memory."""
    global _FP8
    _FP8 = bool(on)


def fp8_enabled():
    return _FP8
