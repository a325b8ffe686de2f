"""In-tree build of the HIP extension (gfx950).

Compiles every ``csrc/*.hip`` + ``csrc/bind.cpp`` with hipcc in parallel and
links ``automodel_amd/ops/libamd_ops.so``. No JIT cache: the .so lives in the
package so it travels with the repo snapshot to GPU boxes.

Usage: ``python -m automodel_amd.ops.build [--force]``
"""

from __future__ import annotations

import concurrent.futures
import hashlib
import json
import os
import subprocess
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
BUILD_DIR = os.path.join(PKG_DIR, "_build")
SO_PATH = os.path.join(PKG_DIR, "libamd_ops.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths() -> tuple[str, str]:
    import torch

    base = os.path.dirname(torch.__file__)
    return os.path.join(base, "include"), os.path.join(base, "lib")


def _sources() -> list[str]:
    out = []
    for f in sorted(os.listdir(CSRC)):
        if f.endswith((".hip", ".cpp")):
            out.append(os.path.join(CSRC, f))
    return out


def _hash_inputs(srcs: list[str]) -> str:
    h = hashlib.sha256()
    for f in srcs + [os.path.join(CSRC, x) for x in sorted(os.listdir(CSRC)) if x.endswith(".h")]:
        with open(f, "rb") as fh:
            h.update(fh.read())
        h.update(f.encode())
    h.update(ARCH.encode())
    return h.hexdigest()


def build(force: bool = False, verbose: bool = True) -> str:
    inc, lib = _torch_paths()
    srcs = _sources()
    stamp_path = os.path.join(BUILD_DIR, "stamp.json")
    digest = _hash_inputs(srcs)
    if not force and os.path.exists(SO_PATH) and os.path.exists(stamp_path):
        try:
            if json.load(open(stamp_path)).get("digest") == digest:
                if verbose:
                    print(f"[amd_ops] up to date: {SO_PATH}")
                return SO_PATH
        except Exception:
            pass

    os.makedirs(BUILD_DIR, exist_ok=True)
    cflags = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        f"-I{inc}", f"-I{os.path.join(inc, 'torch/csrc/api/include')}",
        "-D_GLIBCXX_USE_CXX11_ABI=1", "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__",
        "-Wno-unused-result",
    ]

    def compile_one(src: str) -> str:
        obj = os.path.join(BUILD_DIR, os.path.basename(src) + ".o")
        cmd = ["hipcc", *cflags, "-c", src, "-o", obj]
        if verbose:
            print(f"[amd_ops] hipcc -c {os.path.basename(src)}")
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed for {src}:\n{r.stdout}\n{r.stderr}")
        return obj

    with concurrent.futures.ThreadPoolExecutor(max_workers=min(8, len(srcs))) as ex:
        objs = list(ex.map(compile_one, srcs))

    link = [
        "hipcc", f"--offload-arch={ARCH}", "-shared", "-fPIC", *objs,
        f"-L{lib}", "-ltorch", "-ltorch_hip", "-ltorch_cpu", "-lc10", "-lc10_hip",
        "-lamdhip64", f"-Wl,-rpath,{lib}", "-o", SO_PATH,
    ]
    if verbose:
        print("[amd_ops] linking libamd_ops.so")
    r = subprocess.run(link, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    json.dump({"digest": digest}, open(stamp_path, "w"))
    if verbose:
        print(f"[amd_ops] built {SO_PATH}")
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
