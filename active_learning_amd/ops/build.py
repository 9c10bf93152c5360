"""In-tree build of the gfx950 HIP extension (active_learning_amd._C).

Everything — the *.hip device TUs AND the torch binding — is compiled by
hipcc with --offload-arch=gfx950 and linked into a single .so placed inside
the package (so it travels with repo snapshots to GPU machines). No hipify,
no JIT cache outside the tree.

Usage: python -m active_learning_amd.ops.build  (or setup.py build_ext).
"""

import os
import subprocess
import sys
import sysconfig

HIP_SOURCES = ["bn.hip", "pool.hip", "scoring.hip", "optim.hip", "igemm.hip",
               "wgrad.hip", "kcenter.hip", "linear.hip"]
BINDING = "bindings.cpp"

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))  # .../active_learning_amd
HIP_DIR = os.path.join(PKG_DIR, "ops", "hip")
OUT_SO = os.path.join(PKG_DIR, "_C.so")
BUILD_DIR = os.path.join(PKG_DIR, "ops", "hip", "build")

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_flags():
    import torch
    from torch.utils import cpp_extension as ce
    includes = [f"-I{p}" for p in ce.include_paths(device_type="cuda")]
    includes.append(f"-I{sysconfig.get_paths()['include']}")
    lib_dirs = [f"-L{p}" for p in ce.library_paths(device_type="cuda")]
    libs = ["-ltorch", "-ltorch_python", "-ltorch_hip", "-lc10", "-lc10_hip",
            "-lamdhip64"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defines = [f"-D_GLIBCXX_USE_CXX11_ABI={abi}", "-DUSE_ROCM=1",
               "-D__HIP_PLATFORM_AMD__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
               "-DTORCH_API_INCLUDE_EXTENSION_H",
               "-DTORCH_EXTENSION_NAME=_C"]
    return includes, lib_dirs, libs, defines


def _run(cmd):
    print("  " + " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def _stale(obj, src):
    return (not os.path.exists(obj)
            or os.path.getmtime(obj) < os.path.getmtime(src)
            or os.path.getmtime(obj) < os.path.getmtime(
                os.path.join(HIP_DIR, "al_common.h")))


def build(verbose=True):
    os.makedirs(BUILD_DIR, exist_ok=True)
    includes, lib_dirs, libs, defines = _torch_flags()
    common = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC"]
    objs = []
    for src in HIP_SOURCES:
        obj = os.path.join(BUILD_DIR, src.replace(".hip", ".o"))
        objs.append(obj)
        if _stale(obj, os.path.join(HIP_DIR, src)):
            _run(["hipcc", *common, "-c", os.path.join(HIP_DIR, src), "-o", obj])
    bind_obj = os.path.join(BUILD_DIR, "bindings.o")
    if _stale(bind_obj, os.path.join(HIP_DIR, BINDING)):
        _run(["hipcc", *common, *defines, *includes, "-c",
              os.path.join(HIP_DIR, BINDING), "-o", bind_obj])
    objs.append(bind_obj)
    import torch
    torch_lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    _run(["hipcc", "-shared", "-fPIC", *objs, *lib_dirs, *libs,
          f"-Wl,-rpath,{torch_lib}", "-Wl,-rpath,/opt/rocm/lib", "-o", OUT_SO])
    print(f"built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    build()
