"""In-tree build of the gfx950 HIP extension.

Direct ``hipcc`` invocation (no ninja, no JIT cache): the resulting
``_rlli_hip.so`` sits next to this file, ships to the GPU box with the
repo snapshot, and loads via ``torch.ops.load_library`` — so a fresh box
never recompiles.  Incremental: each translation unit recompiles only
when its mtime (or a header's) is newer than its object file.

Usage: python -m resilient_llm_amd.ops.build [--force]
"""

from __future__ import annotations

import concurrent.futures as cf
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
# Host-side sanitizers for the extension: RLLI_ASAN=1 /
# RLLI_TSAN=1 add -fsanitize to host code (device code unaffected).
# Sanitized builds go to their OWN objects + .so so they never clobber
# the production artifact that ships to GPU boxes.
_SAN = ("asan" if os.environ.get("RLLI_ASAN") == "1" else
        "tsan" if os.environ.get("RLLI_TSAN") == "1" else "")
SAN_FLAGS = {"asan": ["-fsanitize=address"],
             "tsan": ["-fsanitize=thread"], "": []}[_SAN]
BUILD = os.path.join(HERE, "build" + (f"-{_SAN}" if _SAN else ""))
OUT_SO = os.path.join(HERE, f"_rlli_hip{'_' + _SAN if _SAN else ''}.so")

SOURCES = [
    "ext.cpp",
    "lt_gemm.cpp",
    "rmsnorm.hip",
    "silu_mul.hip",
    "rope_kv.hip",
    "decode_attn.hip",
    "prefill_attn.hip",
    "prefill_mfma.hip",
    "prefill_paged.hip",
    "sampling.hip",
    "skinny_gemm.hip",
    "skinny2.hip",
    "streamprobe.hip",
]
HEADERS = ["common.h", "kernels.h"]


def _torch_flags() -> tuple[list[str], list[str]]:
    import torch
    import torch.utils.cpp_extension as ce
    includes = [f"-I{p}" for p in ce.include_paths()]
    libs = [f"-L{p}" for p in ce.library_paths()]
    libs += ["-ltorch", "-ltorch_cpu", "-lc10", "-ltorch_hip", "-lc10_hip",
             "-lamdhip64", "-lhipblaslt"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    includes.append(f"-D_GLIBCXX_USE_CXX11_ABI={abi}")
    return includes, libs


def _needs_build(src: str, obj: str) -> bool:
    if not os.path.exists(obj):
        return True
    obj_mtime = os.path.getmtime(obj)
    deps = [os.path.join(CSRC, src)] + [os.path.join(CSRC, h) for h in HEADERS]
    return any(os.path.getmtime(d) > obj_mtime for d in deps)


def _compile_one(src: str, includes: list[str], force: bool) -> str:
    obj = os.path.join(BUILD, src.replace("/", "_") + ".o")
    if not force and not _needs_build(src, obj):
        return obj
    cmd = (["hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
            "-DNDEBUG", "-c", os.path.join(CSRC, src), "-o", obj,
            f"-I{CSRC}"] + SAN_FLAGS + includes)
    print("  hipcc -c", src, flush=True)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc failed on {src}:\n{r.stderr[-8000:]}")
    return obj


def build(force: bool = False, verbose: bool = True) -> str:
    os.makedirs(BUILD, exist_ok=True)
    includes, libs = _torch_flags()
    objs: list[str] = []
    with cf.ThreadPoolExecutor(max_workers=min(8, len(SOURCES))) as ex:
        objs = list(ex.map(lambda s: _compile_one(s, includes, force), SOURCES))
    if (force or not os.path.exists(OUT_SO)
            or any(os.path.getmtime(o) > os.path.getmtime(OUT_SO) for o in objs)):
        cmd = (["hipcc", f"--offload-arch={ARCH}", "-shared", "-fPIC",
                "-o", OUT_SO] + SAN_FLAGS + objs + libs)
        if verbose:
            print("  hipcc -shared ->", os.path.basename(OUT_SO), flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stderr[-8000:]}")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("built", OUT_SO)
