"""In-tree build of the vescale_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands at vescale_amd/ops/_C*.so and travels to the GPU box
with the repo snapshot.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension


class _FreshBuildExtension(BuildExtension.with_options(no_python_abi_suffix=False)):
    """ninja's depfile does not track .hip files #included by the main TU
    (hipify renames the TU, breaking header dependency discovery), so edits
    to kernel files silently reuse the stale .o. Touch the TU (and remove
    its hipified copy) before every build so it always recompiles."""

    def build_extensions(self):
        here = os.path.dirname(os.path.abspath(__file__))
        csrc = os.path.join(here, "vescale_amd", "ops", "csrc")
        os.utime(os.path.join(csrc, "extension.hip"), None)
        # hipify skips regeneration on equal content; force the copy stale
        hip_copy = os.path.join(csrc, "extension_hip.hip")
        if os.path.exists(hip_copy):
            os.utime(hip_copy, (0, 0))
        # and force ninja to recompile the object even if hipify no-ops
        import glob
        for o in glob.glob(os.path.join(here, "build", "temp*", "vescale_amd", "ops", "csrc", "*.o")):
            os.remove(o)
        super().build_extensions()

setup(
    name="vescale_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="vescale_amd.ops._C",
            sources=["vescale_amd/ops/csrc/extension.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": _FreshBuildExtension},
)
