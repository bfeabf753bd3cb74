"""Loader for the gfx950 HIP extension.

Builds in-tree (``ops/_build``) so the compiled .so travels with the repo
snapshot to GPU boxes.  On a machine WITH a GPU a missing/broken extension
raises — the device ops must never silently fall back to eager PyTorch.  On
CPU-only machines ``require()`` raises and callers (CPU engine) simply never
ask for it.
"""
import logging
import os

log = logging.getLogger("dampr_amd")

_EXT = None
_ERR = None

_HERE = os.path.dirname(os.path.abspath(__file__))
SOURCES = [os.path.join(_HERE, "hip", "dampr_kernels.hip"),
           os.path.join(_HERE, "hip", "dampr_sort.hip")]
# Overridable for kernel-parameter sweeps (scripts/sweep_tfidf.py): each
# variant builds into its own in-tree dir and is selected per process.
BUILD_DIR = os.environ.get(
    "DAMPR_HIP_BUILD_DIR", os.path.join(_HERE, "_build"))


def build(verbose=False):
    """Compile the extension for gfx950 (works without a GPU: hipcc
    cross-compiles)."""
    global _EXT, _ERR
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils import cpp_extension
    defines = [d for d in os.environ.get("DAMPR_HIP_DEFINES",
                                         "").split(",") if d]
    _EXT = cpp_extension.load(
        name="dampr_hip",
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"] + ["-D" + d for d in defines],
        extra_cuda_cflags=["-O3"] + ["-D" + d for d in defines],
        verbose=verbose,
    )
    _ERR = None
    return _EXT


def get():
    """The extension module, building it on first use; raises on failure."""
    global _EXT, _ERR
    if _EXT is not None:
        return _EXT
    if _ERR is not None:
        raise RuntimeError(
            "dampr_hip extension previously failed to load: {}".format(_ERR))
    try:
        return build()
    except Exception as e:       # noqa: BLE001 - report and re-raise
        _ERR = e
        raise RuntimeError(
            "dampr_hip HIP extension failed to build/load; the GPU engine "
            "refuses to run without its native kernels: {}".format(e)) from e


def require():
    """Alias used by GPU-side callers: never returns a fallback."""
    return get()
