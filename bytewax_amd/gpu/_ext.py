"""In-tree build/load of the HIP kernel extension for gfx950.

The built ``.so`` lives under ``bytewax_amd/_native/build/`` so it
travels with repository snapshots (JIT caches under ``~/.cache`` do
not).  On a machine with a GPU, a missing or unloadable extension is a
hard error — the framework must never fall back to an eager emulation
silently.
"""

import os
from pathlib import Path

_NATIVE_DIR = Path(__file__).resolve().parent.parent / "_native"
_BUILD_DIR = _NATIVE_DIR / "build"
_SOURCES = [_NATIVE_DIR / "stream_kernels.hip"]

_ext = None


def build(verbose: bool = False):
    """Compile the HIP extension for gfx950 (works without a GPU)."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    _BUILD_DIR.mkdir(parents=True, exist_ok=True)
    from torch.utils.cpp_extension import load

    return load(
        name="bytewax_amd_stream_kernels",
        sources=[str(s) for s in _SOURCES],
        build_directory=str(_BUILD_DIR),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        verbose=verbose,
    )


def ext():
    """Load (building if necessary) the kernel extension."""
    global _ext
    if _ext is None:
        try:
            _ext = build()
        except Exception as ex:
            import torch

            if torch.cuda.is_available():
                msg = (
                    "bytewax_amd HIP kernel extension failed to load on a "
                    "GPU machine; refusing to fall back to an eager path. "
                    f"Original error: {ex}"
                )
                raise RuntimeError(msg) from ex
            raise
    return _ext
