"""In-tree build of the gfx950 HIP extension.

Built with torch.utils.cpp_extension (hipcc under PYTORCH_ROCM_ARCH=gfx950)
into ``ops/_build`` INSIDE the repo, so the compiled .so travels to GPU
boxes with the source snapshot. hipcc cross-compiles on CPU-only hosts;
``__graft_entry__.build()`` calls :func:`build` as the driver's build check.
"""

from __future__ import annotations

import glob
import os
import sys
import threading

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC_DIR = os.path.join(_THIS_DIR, "csrc")
BUILD_DIR = os.path.join(_THIS_DIR, "_build")
EXT_NAME = "distrl_llm_amd_C"

_ext = None
_ext_err: Exception | None = None
_lock = threading.Lock()


def _sources():
    hip = sorted(glob.glob(os.path.join(CSRC_DIR, "*.hip")))
    cpp = sorted(glob.glob(os.path.join(CSRC_DIR, "*.cpp")))
    return cpp + hip


def build(verbose: bool = False):
    """Compile (if needed) and load the extension. Returns the module."""
    global _ext, _ext_err
    with _lock:
        if _ext is not None:
            return _ext
        srcs = _sources()
        if not srcs:
            raise RuntimeError(f"no kernel sources found under {CSRC_DIR}")
        os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
        os.makedirs(BUILD_DIR, exist_ok=True)
        from torch.utils.cpp_extension import load
        _ext = load(
            name=EXT_NAME,
            sources=srcs,
            build_directory=BUILD_DIR,
            extra_cflags=["-O3", "-std=c++17"],
            extra_cuda_cflags=[
                "-O3", "-std=c++17", "--offload-arch=gfx950", "-ffast-math",
            ],
            verbose=verbose,
        )
        _ext_err = None
        return _ext


def get_extension():
    """The loaded extension, or None if it cannot be built/loaded.
    Never raises; callers decide the failure policy (ops.functional
    requires it for GPU tensors)."""
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    if _ext_err is not None:
        return None
    try:
        return build(verbose=False)
    except Exception as e:  # remember the failure; don't retry every call
        _ext_err = e
        sys.stderr.write(f"[distrl_llm_amd.ops] extension unavailable: {e}\n")
        return None


def extension_available() -> bool:
    return get_extension() is not None


if __name__ == "__main__":
    mod = build(verbose=True)
    print(f"built + loaded {EXT_NAME}: {mod.__file__ if hasattr(mod, '__file__') else mod}")
