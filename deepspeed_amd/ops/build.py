"""Build the gfx950 HIP extension in-tree with hipcc (no hipify, no
CUDAExtension shim). Produces deepspeed_amd/ops/_hip_ops.so.

Usage: python -m deepspeed_amd.ops.build  (or via setup.py / __graft_entry__)
"""
import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT = os.path.join(HERE, "_hip_ops.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    troot = os.path.dirname(torch.__file__)
    return {
        "includes": [
            os.path.join(troot, "include"),
            os.path.join(troot, "include", "torch", "csrc", "api", "include"),
            sysconfig.get_paths()["include"],
        ],
        "lib": os.path.join(troot, "lib"),
        "abi": int(getattr(__import__("torch")._C, "_GLIBCXX_USE_CXX11_ABI",
                           True)),
    }


def _sources():
    srcs = []
    for f in sorted(os.listdir(CSRC)):
        if f.endswith((".hip", ".cpp")):
            srcs.append(os.path.join(CSRC, f))
    return srcs


def needs_rebuild():
    if not os.path.exists(OUT):
        return True
    out_mtime = os.path.getmtime(OUT)
    for src in _sources() + [os.path.join(CSRC, "common.h")]:
        if os.path.exists(src) and os.path.getmtime(src) > out_mtime:
            return True
    return False


def build(verbose=True, force=False):
    if not force and not needs_rebuild():
        if verbose:
            print(f"[ops.build] up-to-date: {OUT}")
        return OUT
    tp = _torch_paths()
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3", "-std=c++17", "-fPIC", "-shared",
        "-fno-gpu-rdc",
        "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={tp['abi']}",
        "-Wno-deprecated-declarations",
        "-Wno-unused-result",
    ]
    for inc in tp["includes"]:
        cmd.append(f"-I{inc}")
    cmd += _sources()
    cmd += [
        f"-L{tp['lib']}",
        "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-ltorch_python",
        "-lc10", "-lc10_hip", "-lamdhip64",
        f"-Wl,-rpath,{tp['lib']}",
        "-o", OUT,
    ]
    if verbose:
        print("[ops.build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"[ops.build] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
