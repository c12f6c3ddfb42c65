"""In-tree build of the HIP dataplane extension (gfx950).

The built bng_amd/dataplane/_C.so travels with the repo snapshot to GPU
boxes (a JIT cache under ~/.cache would not).  `python -m
bng_amd.dataplane.build` (or __graft_entry__.build()) compiles it.
"""
from __future__ import annotations

import os
import shutil
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
BUILD_DIR = os.path.join(PKG_DIR, "_build")
OUT_SO = os.path.join(PKG_DIR, "_C.so")

SOURCES = [
    os.path.join(CSRC, "bng_ext.cpp"),
    os.path.join(CSRC, "bng_kernels.hip"),
]


def _sources_mtime() -> float:
    hdrs = [os.path.join(CSRC, h) for h in ("bng_abi.h", "bng_device.h")]
    return max(os.path.getmtime(f) for f in SOURCES + hdrs)


def build(verbose: bool = False, force: bool = False) -> str:
    """Compile the extension for gfx950 and place _C.so in-tree."""
    if not force and os.path.exists(OUT_SO) and \
            os.path.getmtime(OUT_SO) >= _sources_mtime():
        return OUT_SO
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load
    # experiment toggles (A/B on the GPU box without editing sources)
    defs = []
    if os.environ.get("BNG_SC1_PROBES") in ("0", "1"):
        defs.append("-DBNG_SC1_PROBES=" + os.environ["BNG_SC1_PROBES"])
    mod = load(
        name="bng_dataplane_C",
        sources=SOURCES,
        extra_cflags=["-O3", "-std=c++17"] + defs,
        extra_cuda_cflags=["-O3", "-std=c++17"] + defs,
        build_directory=BUILD_DIR,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    built = os.path.join(BUILD_DIR, "bng_dataplane_C.so")
    shutil.copy2(built, OUT_SO)
    return OUT_SO


_ext = None


def get_ext(required: bool = False):
    """Import the built extension; build lazily if sources changed.

    On a GPU host the extension is REQUIRED: dataplane ops must never fall
    back silently to a CPU path there (the golden model is a test oracle,
    not a production fallback)."""
    global _ext
    if _ext is not None:
        return _ext
    import importlib.util
    so = OUT_SO
    if not os.path.exists(so):
        try:
            so = build()
        except Exception:
            if required:
                raise
            return None
    try:
        import torch  # noqa: F401  (extension links against torch)
        spec = importlib.util.spec_from_file_location("bng_dataplane_C", so)
        _ext = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(_ext)
    except Exception:
        if required:
            raise
        _ext = None
    return _ext


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv, force="-f" in sys.argv)
    print(f"built {path}")
