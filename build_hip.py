#!/usr/bin/env python3
"""Build edl_amd._C for gfx950 by driving hipcc DIRECTLY (no hipify pass,
no CUDA shims) and linking against this interpreter's libtorch.

    python build_hip.py            # -> edl_amd/_C.cpython-*.so (in-tree)

hipcc cross-compiles gfx950 with no GPU present; the in-tree .so travels
with the repo snapshot to the GPU box."""
import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(REPO, "edl_amd", "csrc")
BUILD = os.path.join(REPO, "build", "hip")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    return {
        "includes": ce.include_paths() + [sysconfig.get_paths()["include"]],
        "lib_dir": os.path.join(os.path.dirname(torch.__file__), "lib"),
        "abi": int(torch._C._GLIBCXX_USE_CXX11_ABI),
    }


def newer(srcs, target):
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    return any(os.path.getmtime(s) > t for s in srcs if os.path.exists(s))


def run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build(verbose=True):
    tp = torch_paths()
    os.makedirs(BUILD, exist_ok=True)
    common_flags = [
        "-O3", "-std=c++17", "-fPIC", "--offload-arch=" + ARCH,
        "-D_GLIBCXX_USE_CXX11_ABI=%d" % tp["abi"],
        "-DTORCH_EXTENSION_NAME=_C",
        "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fno-gpu-rdc", "-parallel-jobs=4",
        "-Wno-deprecated-declarations", "-Wno-unused-result",
    ] + ["-I" + i for i in tp["includes"]] + ["-I" + CSRC]

    objs = []
    hdr = [os.path.join(CSRC, "common.h")]
    sources = sorted(
        f for f in os.listdir(CSRC) if f.endswith(".hip") or f.endswith(".cpp")
    )
    for src in sources:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.rsplit(".", 1)[0] + ".o")
        objs.append(op)
        if newer([sp] + hdr, op):
            run([HIPCC, "-c", sp, "-o", op] + common_flags +
                (["-x", "hip"] if src.endswith(".hip") else []))

    ext_suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    out = os.path.join(REPO, "edl_amd", "_C" + ext_suffix)
    if newer(objs, out):
        run([HIPCC, "-shared", "-o", out] + objs + [
            "-L" + tp["lib_dir"],
            "-Wl,-rpath," + tp["lib_dir"],
            "-ltorch", "-ltorch_hip", "-ltorch_cpu", "-lc10", "-lc10_hip",
            "-ltorch_python", "-lamdhip64",
        ])
    print("built", out)
    return out


if __name__ == "__main__":
    build()
