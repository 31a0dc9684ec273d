"""Loader for the in-tree gfx950 HIP extension (_adanet_hip).

Policy: on CUDA (ROCm) tensors every op MUST run through the hand-written
CDNA4 kernels — if the extension is missing on a GPU box we raise instead of
silently falling back to eager PyTorch. CPU tensors use plain fp32 PyTorch
reference implementations (they exist for GPU-less CI and as the numerics
oracle in tests/).
"""

from __future__ import annotations

_ext = None
_err = None
_tried = False


def load():
    global _ext, _err, _tried
    if not _tried:
        _tried = True
        try:
            from adanet_amd import _adanet_hip  # noqa: F401

            _ext = _adanet_hip
        except ImportError as e:  # pragma: no cover - build problem
            _err = e
    return _ext


def available() -> bool:
    return load() is not None


def require():
    m = load()
    if m is None:  # pragma: no cover
        raise RuntimeError(
            "adanet_amd HIP extension (_adanet_hip) is not built, but a GPU "
            "tensor reached an adanet_amd op. Build it in-tree with "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
            "GPU ops never fall back to eager PyTorch by design. "
            "Original import error: %r" % (_err,))
    return m
