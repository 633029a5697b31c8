"""In-tree build of the gfx950 HIP extension.

Compiles ops/hip/*.hip (+ bindings.cpp) with hipcc --offload-arch=gfx950
directly (parallel hipcc -c, then one shared link) into
novel_view_synthesis_3d_amd/ops/nvs3d_hip.so — in-tree so the built .so
travels to the GPU box with the gpurun snapshot (JIT caches under ~/.cache
do not). Include/library paths come from torch.utils.cpp_extension so the
build tracks the installed torch. Driving hipcc ourselves (instead of
torch's cpp_extension JIT loader) also avoids the hipify pass that used to
drop *_hip.hip byte-copies next to the sources.
"""

from __future__ import annotations

import glob
import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(HERE, "hip")
EXT_NAME = "nvs3d_hip"


def hip_sources():
    hips = [s for s in sorted(glob.glob(os.path.join(HIP_DIR, "*.hip")))
            if not s.endswith("_hip.hip")]  # stale hipify-era copies
    return hips + sorted(glob.glob(os.path.join(HIP_DIR, "*.cpp")))


def built_path() -> str:
    return os.path.join(HERE, EXT_NAME + ".so")


def _torch_paths():
    from torch.utils import cpp_extension as ce
    incs = ce.include_paths()
    libs = ce.library_paths()
    return incs, libs


_COMMON_DEFS = [
    "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
    f"-DTORCH_EXTENSION_NAME={EXT_NAME}", "-DTORCH_API_INCLUDE_EXTENSION_H",
    "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
    "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
]


def _compile_one(src: str, out: str, incs) -> None:
    is_hip = src.endswith(".hip")
    cc = ["/opt/rocm/bin/hipcc"] if is_hip else ["c++"]
    cmd = cc + ["-O3", "-std=c++17", "-fPIC", "-c", src, "-o", out]
    cmd += _COMMON_DEFS
    for i in incs:
        cmd += ["-isystem", i]
    cmd += ["-isystem", "/opt/rocm/include",
            "-isystem", f"/usr/include/python{sys.version_info.major}."
                        f"{sys.version_info.minor}"]
    if is_hip:
        cmd += ["--offload-arch=gfx950", "-fno-gpu-rdc"]
    subprocess.run(cmd, check=True)


def build_extensions(verbose: bool = False, force: bool = False) -> str:
    """Compile the extension if sources are newer than the built .so."""
    srcs = hip_sources()
    if not srcs:
        if verbose:
            print("build_extensions: no HIP sources yet — nothing to build")
        return ""
    out = built_path()
    hdrs = glob.glob(os.path.join(HIP_DIR, "*.h"))
    if not force and os.path.exists(out):
        newest = max(os.path.getmtime(s) for s in srcs + hdrs + [__file__])
        if os.path.getmtime(out) >= newest:
            if verbose:
                print(f"build_extensions: {out} up to date")
            return out

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    incs, libdirs = _torch_paths()
    build_dir = os.path.join(HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)

    objs = []
    jobs = []
    with ThreadPoolExecutor(max_workers=max(4, os.cpu_count() or 4)) as ex:
        for src in srcs:
            obj = os.path.join(
                build_dir,
                os.path.splitext(os.path.basename(src))[0] + ".o")
            objs.append(obj)
            if (force or not os.path.exists(obj)
                    or os.path.getmtime(obj) < max(
                        os.path.getmtime(src),
                        *(os.path.getmtime(h) for h in hdrs) if hdrs
                        else [0])):
                if verbose:
                    print(f"  hipcc -c {os.path.basename(src)}")
                jobs.append(ex.submit(_compile_one, src, obj, incs))
        for j in jobs:
            j.result()  # raise on compile error

    link = ["c++", "-shared"] + objs + ["-o", out]
    for d in libdirs:
        link += [f"-L{d}"]
    link += ["-lc10", "-lc10_hip", "-ltorch_cpu", "-ltorch_hip", "-ltorch",
             "-ltorch_python", "-L/opt/rocm/lib", "-lamdhip64"]
    subprocess.run(link, check=True)
    if verbose:
        print(f"build_extensions: built {out}")
    return out


if __name__ == "__main__":
    build_extensions(verbose=True, force="--force" in sys.argv)
