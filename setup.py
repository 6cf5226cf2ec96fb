"""Build/install for autodist_amd.

The gfx950 HIP extension builds IN-TREE so the .so travels with the source
checkout: `python setup.py build_ext --inplace` (or
`python -m autodist_amd.ops.build`). PYTORCH_ROCM_ARCH defaults to gfx950.
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext_modules = []
cmdclass = {}
try:
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    here = os.path.dirname(os.path.abspath(__file__))

    class InplaceBuild(BuildExtension):
        def build_extensions(self):
            super().build_extensions()
            # also drop the .so next to ops/api.py for snapshot shipping
            import glob
            import shutil
            for so in glob.glob(os.path.join(self.build_lib, "**",
                                             "_autodist_hip*.so"),
                                recursive=True):
                shutil.copy2(so, os.path.join(here, "autodist_amd", "ops",
                                              "_autodist_hip.so"))

    ext_modules = [CUDAExtension(
        name="autodist_amd.ops._autodist_hip",
        sources=["autodist_amd/ops/csrc/ext.hip"],
        extra_compile_args={"nvcc": ["-O3", "--offload-arch=gfx950"]},
    )]
    cmdclass = {"build_ext": InplaceBuild}
except ImportError:
    pass

setup(
    name="autodist_amd",
    version="0.1.0",
    description="MI355X-native distributed training engine with "
                "petuum/autodist's capabilities",
    packages=find_packages(include=["autodist_amd", "autodist_amd.*"]),
    python_requires=">=3.10",
    install_requires=["torch>=2.1", "numpy", "pyyaml"],
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
