"""HIP/CDNA4 extension loader.

The native extension ``_sparkhip`` (built in-tree by ``setup.py build_ext
--inplace`` / ``__graft_entry__.build()`` for gfx950) provides the hand-written
kernels: fused Adam/SGD on flat buckets, bf16/f32 MFMA linear fwd/bwd, fused
bias+ReLU, cross-entropy, and the fp64->fp32 vector pack.

Policy: on a GPU the HIP path is mandatory — ops raise if the extension is
missing rather than silently falling back to eager PyTorch.  On CPU (this
build sandbox has no GPU) pure-torch fallbacks keep the logic testable.
"""

from __future__ import annotations

_EXT = None


def _try_load():
    global _EXT
    if _EXT is None:
        try:
            from sparktorch_amd.ops import _sparkhip  # type: ignore

            _EXT = _sparkhip
        except ImportError:
            _EXT = False
    return _EXT


def available() -> bool:
    return bool(_try_load())


def ext():
    e = _try_load()
    if not e:
        raise RuntimeError(
            "sparktorch_amd native HIP extension (_sparkhip) is not built. "
            "Build it in-tree with: PYTORCH_ROCM_ARCH=gfx950 python setup.py "
            "build_ext --inplace  (GPU ops refuse to fall back to eager torch)."
        )
    return e
