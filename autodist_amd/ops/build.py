"""In-tree builder for the gfx950 HIP extension.

Cross-compiles on CPU-only machines (hipcc targets gfx950 without a GPU).
The resulting _autodist_hip.so is written NEXT TO ops/api.py so it travels
with the repo snapshot to GPU boxes (a JIT cache under ~/.cache would not).

Run: python -m autodist_amd.ops.build
"""
import glob
import os
import shutil
import sys


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "csrc", "ext.hip")
    build_dir = os.path.join(here, "_build")
    os.makedirs(build_dir, exist_ok=True)
    load(
        name="_autodist_hip",
        sources=[src],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=False,
        keep_intermediates=True,
    )
    built = glob.glob(os.path.join(build_dir, "_autodist_hip.so"))
    if not built:
        raise RuntimeError(f"extension build produced no .so in {build_dir}")
    dest = os.path.join(here, "_autodist_hip.so")
    shutil.copy2(built[0], dest)
    return dest


if __name__ == "__main__":
    path = build()
    print(f"built {path}")
    sys.exit(0)
