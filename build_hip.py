"""In-tree gfx950 HIP extension build.

Compiles deepfake_detection_amd/ops/hip/*.hip with hipcc
(--offload-arch=gfx950) and links deepfake_detection_amd/_hip_ops.so against
libtorch. Built in-tree so the .so travels to GPU boxes with the repo
snapshot. Used by setup.py and __graft_entry__.build().
"""

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))
SRC_DIR = os.path.join(REPO, "deepfake_detection_amd", "ops", "hip")
OUT_SO = os.path.join(REPO, "deepfake_detection_amd", "_hip_ops.so")
BUILD_DIR = os.path.join(REPO, "build", "hip")

SOURCES = ["bn_act.hip", "normalize.hip", "pool.hip", "se.hip", "optim.hip", "dwconv.hip", "pwconv.hip", "stemconv.hip", "head.hip", "ext.hip"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    import torch.utils.cpp_extension as cpp_ext

    includes = cpp_ext.include_paths()
    lib_dir = os.path.join(os.path.dirname(torch.__file__), "lib")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return includes, lib_dir, abi


def _needs_rebuild(obj, src, headers):
    if not os.path.exists(obj):
        return True
    ot = os.path.getmtime(obj)
    if os.path.getmtime(src) > ot:
        return True
    return any(os.path.getmtime(h) > ot for h in headers)


def build(verbose=True, force=False):
    includes, lib_dir, abi = _torch_paths()
    os.makedirs(BUILD_DIR, exist_ok=True)
    hipcc = os.environ.get("HIPCC", "hipcc")

    common = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    inc_flags = [f"-I{p}" for p in includes]
    inc_flags.append(f"-I{sysconfig.get_paths()['include']}")

    headers = [os.path.join(SRC_DIR, "common.h")]
    objs = []
    compiled = 0
    for src in SOURCES:
        src_path = os.path.join(SRC_DIR, src)
        obj = os.path.join(BUILD_DIR, src.replace(".hip", ".o"))
        objs.append(obj)
        if not force and not _needs_rebuild(obj, src_path, headers):
            continue
        cmd = [hipcc, "-c", src_path, "-o", obj] + common + inc_flags
        if verbose:
            print("[build_hip]", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)
        compiled += 1

    if compiled or not os.path.exists(OUT_SO) or force:
        link = [hipcc, "-shared", "-fPIC", "-o", OUT_SO] + objs + [
            f"-L{lib_dir}", "-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip",
            "-ltorch_python", f"-Wl,-rpath,{lib_dir}",
        ]
        if verbose:
            print("[build_hip]", " ".join(link), flush=True)
        subprocess.check_call(link)
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("built", OUT_SO)
