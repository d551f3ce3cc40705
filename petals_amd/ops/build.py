"""Build the HIP extension IN-TREE (the .so must travel with the repo snapshot
to GPU boxes): `python -m petals_amd.ops.build`.

Cross-compiles for gfx950 (no GPU needed at build time).
"""

from __future__ import annotations

import os
import shutil
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
SRC_DIR = os.path.join(PKG_DIR, "csrc")
BUILD_DIR = os.path.join(PKG_DIR, "_build")
SO_PATH = os.path.join(PKG_DIR, "_hip_ops.so")

SOURCES = ["bindings.cpp", "elementwise.hip", "gemv.hip", "attention.hip", "nf4.hip", "int8.hip", "prefill_attn.hip", "moe.hip"]


def build(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(min(8, os.cpu_count() or 4)))
    from torch.utils.cpp_extension import load

    os.makedirs(BUILD_DIR, exist_ok=True)
    # torch's hipify writes `<name>_hip.<ext>` next to each source and skips
    # regeneration based on mtimes; a repo snapshot (gpurun) can scramble those
    # and leave a stale artifact compiling instead of the real source. Purge
    # them so every build regenerates from the true sources.
    import glob

    for stale in glob.glob(os.path.join(SRC_DIR, "*_hip.*")):
        os.unlink(stale)
    sources = [os.path.join(SRC_DIR, s) for s in SOURCES if os.path.exists(os.path.join(SRC_DIR, s))]
    module = load(
        name="petals_amd_hip_ops",
        sources=sources,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "-ffast-math"],
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
        keep_intermediates=True,
    )
    # copy the built .so in-tree under a stable name
    built = os.path.join(BUILD_DIR, "petals_amd_hip_ops.so")
    if os.path.exists(built):
        shutil.copy2(built, SO_PATH)
    return SO_PATH


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print(f"built {path}")
