"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces distributedes_amd/ops/_hip_ops*.so (git-ignored; travels to GPU
boxes with the repo snapshot).  hipcc cross-compiles without a GPU present.
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
# keep hipcc from probing for a GPU
os.environ.setdefault("ROCM_PATH", "/opt/rocm")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "distributedes_amd" / "csrc"

sources = sorted(str(p) for p in CSRC.glob("*.cpp")) + \
          sorted(str(p) for p in CSRC.glob("*.hip")
                 if not p.name.endswith("_hip.hip"))  # exclude hipify copies

class CleanHipifyBuildExt(BuildExtension):
    """Delete torch-hipify's `*_hip.hip` byproducts after building: they are
    verbatim copies of the sources (which are already native HIP) and only
    pollute the tree / gpurun snapshots."""

    def run(self):
        super().run()
        for p in CSRC.glob("*_hip.hip"):
            p.unlink()


setup(
    name="distributedes_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="distributedes_amd.ops._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": CleanHipifyBuildExt.with_options(no_python_abi_suffix=False)},
)
