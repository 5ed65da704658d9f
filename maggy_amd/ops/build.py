"""In-tree build of the HIP extension for gfx950.

``python -m maggy_amd.ops.build`` compiles maggy_kernels.hip + bindings.cpp
with hipcc (via torch.utils.cpp_extension, PYTORCH_ROCM_ARCH=gfx950) into
``maggy_amd/ops/_build/_maggy_hip.so``.  The .so is git-ignored but travels
with the repo snapshot to GPU boxes; ops/__init__.py loads it directly from
that path without rebuilding.
"""
import os
import shutil
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(PKG_DIR, "hip")
BUILD_DIR = os.path.join(PKG_DIR, "_build")
SO_NAME = "_maggy_hip"
SO_PATH = os.path.join(BUILD_DIR, SO_NAME + ".so")
SOURCES = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "maggy_kernels.hip"),
    os.path.join(HIP_DIR, "fused_bn.hip"),
    os.path.join(HIP_DIR, "fused_rms.hip"),
    os.path.join(HIP_DIR, "gemm.hip"),
]


def _sources_mtime():
    return max(os.path.getmtime(s) for s in SOURCES)


def is_built():
    return os.path.exists(SO_PATH) and \
        os.path.getmtime(SO_PATH) >= _sources_mtime()


def build(verbose=False, force=False):
    """Compile the extension in-tree; returns the .so path."""
    if is_built() and not force:
        return SO_PATH
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    load(
        name=SO_NAME,
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
        is_python_module=True,
        keep_intermediates=True,
    )
    built = os.path.join(BUILD_DIR, SO_NAME + ".so")
    if not os.path.exists(built):
        # ninja names it <name>.so in build_directory; fall back to search
        for f in os.listdir(BUILD_DIR):
            if f.startswith(SO_NAME) and f.endswith(".so"):
                shutil.copy(os.path.join(BUILD_DIR, f), built)
                break
    if not os.path.exists(built):
        raise RuntimeError("HIP extension build produced no .so in {}".format(
            BUILD_DIR))
    return built


if __name__ == "__main__":
    path = build(verbose=("-q" not in sys.argv), force=("-f" in sys.argv))
    print("built:", path)
