"""In-tree build of the axon C++/HIP core (axon/_core.so) for gfx950.

Drives hipcc directly (no hipify, no CUDA shims): all host C++ is written
against PyTorch-ROCm's native c10::hip / ATen hip API, all kernels are
CDNA4 HIP. Content-hash caching keeps rebuilds incremental.
"""
import concurrent.futures as cf
import hashlib
import json
import os
import subprocess
import sys
import sysconfig

PKG = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG, "csrc")
OBJDIR = os.path.join(CSRC, ".obj")
OUT = os.path.join(PKG, "_core.so")

ROCM = os.environ.get("ROCM_HOME", "/opt/rocm")
HIPCC = os.path.join(ROCM, "bin", "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch

    t = os.path.dirname(torch.__file__)
    return (
        [os.path.join(t, "include"),
         os.path.join(t, "include", "torch", "csrc", "api", "include")],
        os.path.join(t, "lib"),
    )


def _sources():
    srcs = []
    for root, _dirs, files in os.walk(CSRC):
        if ".obj" in root:
            continue
        for f in sorted(files):
            if f.endswith((".cpp", ".hip")):
                srcs.append(os.path.join(root, f))
    return srcs


def _headers():
    hs = []
    for root, _dirs, files in os.walk(CSRC):
        for f in sorted(files):
            if f.endswith(".h"):
                hs.append(os.path.join(root, f))
    return hs


def build(verbose=False):
    inc, torch_lib = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    common = [
        "-fPIC", "-O3", "-std=c++17",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        f"--offload-arch={ARCH}",
        "-Wall", "-Wno-unused-function",
    ]
    torch_defs = [
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_core",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
    ]
    incs = [f"-I{p}" for p in inc + [py_inc, os.path.join(ROCM, "include"), CSRC]]

    os.makedirs(OBJDIR, exist_ok=True)
    hdr_hash = hashlib.sha256()
    for h in _headers():
        hdr_hash.update(open(h, "rb").read())
    hdr_digest = hdr_hash.hexdigest()

    cache_path = os.path.join(OBJDIR, "cache.json")
    cache = {}
    if os.path.exists(cache_path):
        try:
            cache = json.load(open(cache_path))
        except Exception:
            cache = {}

    def compile_one(src):
        rel = os.path.relpath(src, CSRC).replace(os.sep, "_")
        obj = os.path.join(OBJDIR, rel + ".o")
        key = hashlib.sha256(
            (open(src, "rb").read().hex() + hdr_digest + " ".join(common + torch_defs)).encode()
        ).hexdigest()
        if cache.get(rel) == key and os.path.exists(obj):
            return obj, rel, key, False
        flags = list(common)
        if src.endswith(".cpp"):
            flags += torch_defs
        cmd = [HIPCC, "-c", src, "-o", obj] + flags + incs
        if verbose:
            print(" ".join(cmd))
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed for {src}:\n{r.stdout}\n{r.stderr}")
        if r.stderr and verbose:
            print(r.stderr, file=sys.stderr)
        return obj, rel, key, True

    srcs = _sources()
    objs = [None] * len(srcs)
    built_any = False
    with cf.ThreadPoolExecutor(max_workers=os.cpu_count()) as ex:
        futs = {ex.submit(compile_one, s): i for i, s in enumerate(srcs)}
        for fut in cf.as_completed(futs):
            obj, rel, key, rebuilt = fut.result()
            objs[futs[fut]] = obj
            cache[rel] = key
            built_any |= rebuilt

    json.dump(cache, open(cache_path, "w"))

    if built_any or not os.path.exists(OUT):
        link = [HIPCC, "-shared", "-o", OUT] + objs + [
            f"-L{torch_lib}",
            "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-ltorch_python",
            "-lc10", "-lc10_hip", "-lrccl",
            f"-Wl,-rpath,{torch_lib}",
            f"-Wl,-rpath,{os.path.join(ROCM, 'lib')}",
        ]
        if verbose:
            print(" ".join(link))
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
        print(f"axon: built {OUT}")
    else:
        print("axon: _core.so up to date")
    return OUT


if __name__ == "__main__":
    build(verbose="-v" in sys.argv)
