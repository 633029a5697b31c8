"""In-tree build of the gfx950 HIP extension.

Compiles ops/hip/*.hip (+ bindings.cpp) with hipcc --offload-arch=gfx950 via
torch.utils.cpp_extension into novel_view_synthesis_3d_amd/ops/nvs3d_hip.so —
in-tree so the built .so travels to the GPU box with the gpurun snapshot
(JIT caches under ~/.cache do not).
"""

from __future__ import annotations

import glob
import os
import shutil
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(HERE, "hip")
EXT_NAME = "nvs3d_hip"


def hip_sources():
    hips = [s for s in sorted(glob.glob(os.path.join(HIP_DIR, "*.hip")))
            if not s.endswith("_hip.hip")]  # hipify-generated copies
    return hips + sorted(glob.glob(os.path.join(HIP_DIR, "*.cpp")))


def built_path() -> str:
    return os.path.join(HERE, EXT_NAME + ".so")


def build_extensions(verbose: bool = False, force: bool = False) -> str:
    """Compile the extension if sources are newer than the built .so."""
    srcs = hip_sources()
    if not srcs:
        if verbose:
            print("build_extensions: no HIP sources yet — nothing to build")
        return ""
    out = built_path()
    if not force and os.path.exists(out):
        newest = max(os.path.getmtime(s) for s in srcs)
        if os.path.getmtime(out) >= newest:
            if verbose:
                print(f"build_extensions: {out} up to date")
            return out

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)
    mod = load(
        name=EXT_NAME,
        sources=srcs,
        build_directory=build_dir,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,  # just build; we copy the .so in-tree
    )
    so = glob.glob(os.path.join(build_dir, EXT_NAME + "*.so"))
    assert so, f"build produced no .so in {build_dir}"
    shutil.copy2(so[0], out)
    if verbose:
        print(f"build_extensions: built {out}")
    return out


if __name__ == "__main__":
    build_extensions(verbose=True, force="--force" in sys.argv)
