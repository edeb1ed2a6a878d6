#!/usr/bin/env python3
"""Build the adapm_amd native extension in-tree with hipcc (gfx950).

No hipify, no torch cpp_extension JIT cache: sources are native HIP/C++,
compiled directly so the built .so lives in the repo and travels to GPU
boxes with the snapshot. Incremental: recompiles only TUs whose source (or
any header) is newer than the object.
"""
import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

REPO = Path(__file__).resolve().parent
CSRC = REPO / "adapm_amd" / "csrc"
OBJ = CSRC / ".obj"
TORCH_DIR = None


def torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    return ce.include_paths(), ce.library_paths()[0]


SOURCES = [
    "slab_host.cpp",
    "ops_cpu.cpp",
    "ops_hip.hip",
    "core.cpp",
    "kernels_hip.hip",
    "mfma_hip.hip",
    "kernels_cpu.cpp",
]

MODULE = "_C"


def build(verbose: bool = True, tsan: bool = False) -> Path:
    """tsan=True (or ADAPM_TSAN=1 / `python build.py --tsan`) compiles the
    HOST side of the extension with ThreadSanitizer — the race-hunt build
    the SURVEY prescribes next to the NaN-poison debug mode (the striped
    locks / lock-free metadata / PassPool are all host code). Expect
    noise from torch's own un-instrumented runtime; use a suppressions
    file for torch/* frames."""
    inc, libdir = torch_paths()
    obj_dir = OBJ if not tsan else (CSRC / ".obj_tsan")
    obj_dir.mkdir(exist_ok=True)
    py_inc = sysconfig.get_paths()["include"]
    soname = MODULE + sysconfig.get_config_var("EXT_SUFFIX")
    out = REPO / "adapm_amd" / soname

    cflags = [
        "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC", "-DNDEBUG", "-munsafe-fp-atomics",
        "-D_GLIBCXX_USE_CXX11_ABI=1", f"-DTORCH_EXTENSION_NAME={MODULE}",
        "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__", "-DADAPM_WITH_HIP",
        "-Wno-unused-result",
    ] + [f"-I{p}" for p in inc] + [f"-I{py_inc}", f"-I{CSRC}"]
    if tsan:
        cflags += ["-fsanitize=thread", "-g", "-fno-omit-frame-pointer"]

    headers = list(CSRC.glob("*.h"))
    hdr_mtime = max((h.stat().st_mtime for h in headers), default=0)

    objs = []
    jobs = []
    for src in SOURCES:
        sp = CSRC / src
        if not sp.exists():
            continue
        op = obj_dir / (src.replace(".", "_") + ".o")
        objs.append(op)
        if op.exists() and op.stat().st_mtime > max(sp.stat().st_mtime, hdr_mtime):
            continue
        cmd = ["hipcc", "-c", str(sp), "-o", str(op)] + cflags
        jobs.append((src, cmd))

    def run(job):
        name, cmd = job
        if verbose:
            print(f"[build] hipcc -c {name}", flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"compile failed: {name}\n{r.stdout}\n{r.stderr}")
        return name

    if jobs:
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(run, jobs))

    if jobs or not out.exists():
        link = (
            ["hipcc", "-shared", "-fPIC"] + (["-fsanitize=thread"] if tsan else [])
            + [str(o) for o in objs] + ["-o", str(out)]
            + [f"-L{libdir}", "-ltorch", "-ltorch_cpu", "-lc10", "-ltorch_python",
               "-ltorch_hip", "-lc10_hip", f"-Wl,-rpath,{libdir}"]
        )
        if verbose:
            print("[build] link", out.name, flush=True)
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    return out


if __name__ == "__main__":
    import os as _os
    import sys as _sys

    _tsan = "--tsan" in _sys.argv or _os.environ.get("ADAPM_TSAN", "0") == "1"
    p = build(tsan=_tsan)
    print(f"built {p}")
