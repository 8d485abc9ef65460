"""Build the gfx950 HIP extension in-tree with hipcc (no hipify, no
CUDAExtension shim). Incremental: per-source .o caching + parallel compile.

Usage: python -m deepspeed_amd.ops.build  (or via setup.py / __graft_entry__)
"""
import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OBJ = os.path.join(HERE, "_build")
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
    return [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC))
            if f.endswith((".hip", ".cpp"))]


def _headers_mtime():
    hs = [os.path.join(CSRC, f) for f in os.listdir(CSRC)
          if f.endswith(".h")]
    return max((os.path.getmtime(h) for h in hs), default=0)


def _common_flags(tp):
    flags = [
        f"--offload-arch={ARCH}",
        "-O3", "-std=c++17", "-fPIC",
        "-fno-gpu-rdc",
        "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={tp['abi']}",
        "-Wno-deprecated-declarations",
        "-fopenmp",
        "-Wno-unused-result",
    ]
    for inc in tp["includes"]:
        flags.append(f"-I{inc}")
    return flags


def build(verbose=True, force=False):
    tp = _torch_paths()
    os.makedirs(OBJ, exist_ok=True)
    flags = _common_flags(tp)
    hmtime = _headers_mtime()
    objs, to_compile = [], []
    for src in _sources():
        obj = os.path.join(OBJ, os.path.basename(src) + ".o")
        objs.append(obj)
        if force or not os.path.exists(obj) or \
                os.path.getmtime(obj) < max(os.path.getmtime(src), hmtime):
            to_compile.append((src, obj))

    def compile_one(pair):
        src, obj = pair
        cmd = ["hipcc"] + flags + ["-c", src, "-o", obj]
        if verbose:
            print(f"[ops.build] compiling {os.path.basename(src)}")
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(
                f"hipcc failed for {src}:\n{r.stderr[-4000:]}")
        return obj

    if to_compile:
        with ThreadPoolExecutor(max_workers=min(8, len(to_compile))) as ex:
            list(ex.map(compile_one, to_compile))

    if to_compile or force or not os.path.exists(OUT):
        link = (["hipcc", "-shared", "-fopenmp"] + objs + [
            f"-L{tp['lib']}",
            "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-ltorch_python",
            "-lc10", "-lc10_hip", "-lamdhip64",
            f"-Wl,-rpath,{tp['lib']}",
            "-o", OUT,
        ])
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stderr[-4000:]}")
        if verbose:
            print(f"[ops.build] built {OUT}")
    elif verbose:
        print(f"[ops.build] up-to-date: {OUT}")
    return OUT


def needs_rebuild():
    return not os.path.exists(OUT)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
