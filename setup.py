"""Package + in-tree gfx950 HIP extension build.

    python setup.py build_ext --inplace   # build ops/nvs3d_hip.so
    pip install -e .                      # editable install

(The extension can also be built via `__graft_entry__.build()` or lazily by
novel_view_synthesis_3d_amd.ops.build.build_extensions().)
"""

import os
import sys

from setuptools import Command, find_packages, setup


class BuildHipExt(Command):
    """Builds ops/hip/*.hip with hipcc --offload-arch=gfx950 into an in-tree
    .so (torch.utils.cpp_extension under the hood)."""

    user_options = [("inplace", "i", "build in-tree (always true here)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
        from novel_view_synthesis_3d_amd.ops.build import build_extensions
        build_extensions(verbose=True)


setup(
    name="novel_view_synthesis_3d_amd",
    version="0.1.0",
    description="MI355X-native 3DiM novel-view-synthesis engine "
                "(PyTorch-ROCm + CDNA4 HIP kernels + RCCL)",
    packages=find_packages(include=["novel_view_synthesis_3d_amd*"]),
    package_data={"novel_view_synthesis_3d_amd.ops": ["*.so", "hip/*"]},
    python_requires=">=3.10",
    cmdclass={"build_ext": BuildHipExt},
)
