"""Loader for the in-tree HIP extension (csrc/).

The extension is built IN-TREE (csrc/build/) so the .so travels with the
repo snapshot to GPU boxes.  On a machine with a GPU, ops FAIL LOUDLY if
the extension is missing — there is no silent PyTorch fallback on the
device compute path (the CPU path, used for host-only tests, is a
separate torch-f32 reference implementation in functional.py).
"""

import importlib.util
import os
import sys

_EXT_NAME = "_shallowspeed_hip"
_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
_BUILD_DIR = os.path.join(_REPO_ROOT, "csrc", "build")

_ext = None
_ext_err = None


def _find_so():
    if not os.path.isdir(_BUILD_DIR):
        return None
    for fn in os.listdir(_BUILD_DIR):
        if fn.startswith(_EXT_NAME) and fn.endswith(".so"):
            return os.path.join(_BUILD_DIR, fn)
    return None


def load_ext(required=False):
    """Import the built HIP extension; build lazily only if asked.

    Returns the module or None.  required=True raises if unavailable.
    """
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    so = _find_so()
    if so is not None:
        try:
            import torch  # noqa: F401  (libtorch symbols must be loaded first)

            spec = importlib.util.spec_from_file_location(_EXT_NAME, so)
            mod = importlib.util.module_from_spec(spec)
            sys.modules[_EXT_NAME] = mod
            spec.loader.exec_module(mod)
            _ext = mod
            return _ext
        except Exception as e:  # pragma: no cover
            _ext_err = e
    if required:
        raise RuntimeError(
            f"shallowspeed_amd HIP extension not available "
            f"(looked in {_BUILD_DIR}; last error: {_ext_err}). "
            f"Run `python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"from the repo root to build it for gfx950."
        )
    return None


def build_ext(verbose=True):
    """Compile the HIP extension for gfx950 into csrc/build (in-tree).

    Uses torch.utils.cpp_extension (drives hipcc for .hip sources with
    PYTORCH_ROCM_ARCH=gfx950).  Cross-compiles fine on a GPU-less host.
    """
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    os.makedirs(_BUILD_DIR, exist_ok=True)
    src_dir = os.path.join(_REPO_ROOT, "csrc")
    sources = [
        os.path.join(src_dir, "bindings.cpp"),
        os.path.join(src_dir, "gemm.hip"),
        os.path.join(src_dir, "gemm256.hip"),
        os.path.join(src_dir, "gemm256w.hip"),
        os.path.join(src_dir, "wgrad256.hip"),
        os.path.join(src_dir, "fp8.hip"),
        os.path.join(src_dir, "elementwise.hip"),
        os.path.join(src_dir, "norm.hip"),
        os.path.join(src_dir, "fused_mlp.hip"),
    ]
    mod = load(
        name=_EXT_NAME,
        sources=sources,
        build_directory=_BUILD_DIR,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=True,
    )
    global _ext
    _ext = mod
    return mod
