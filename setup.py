"""Build hippt._C — the MI355X-native path tracing core.

Driven directly by hipcc (no torch cpp_extension dependency): host C++ TUs and
HIP device TUs are compiled for gfx950 and linked into hippt/_C.so in-tree, so
the built artifact travels to GPU boxes with the repo snapshot.

Usage: python setup.py build_ext --inplace    (or: python -m hippt.build)
"""
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("HIPPT_ARCH", "gfx950")

SOURCES = [
    "csrc/bind.cpp",
    "csrc/cpu/bvh_build.cpp",
    "csrc/cpu/sbvh_build.cpp",
    "csrc/cpu/cpu_render.cpp",
    "csrc/hip/pt_kernels.hip",
    "csrc/hip/wf_kernels.hip",
]


def pybind11_includes():
    import pybind11
    return [pybind11.get_include()]


def build(verbose=True):
    py_inc = sysconfig.get_paths()["include"]
    ext = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    out = ROOT / "hippt" / f"_C{ext}"
    objdir = ROOT / "build" / "obj"
    objdir.mkdir(parents=True, exist_ok=True)
    incs = [f"-I{py_inc}"] + [f"-I{p}" for p in pybind11_includes()]
    cflags = [
        "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
        "-ffast-math", "-fno-finite-math-only",
        "-Wno-unused-result",
    ]
    if os.environ.get("HIPPT_QBVH"):
        # quantized 64-byte BVH4 nodes as the traversal tree (A/B build)
        cflags.append("-DHIPPT_QBVH")
    ldflags = []
    if os.environ.get("HIPPT_DEBUG"):
        # debuggable build (reference parity: Debug -G -g device debug,
        # CMakeLists.txt:17-27): host -g + device line info, no fast math
        cflags = [f for f in cflags if f not in ("-O3", "-ffast-math")]
        cflags += ["-O1", "-g"]
    if os.environ.get("HIPPT_ASAN"):
        # host AddressSanitizer for the C++ builders/CPU renderer (beyond
        # the reference's -fsanitize=leak).  The extension then needs
        #   LD_PRELOAD=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so)
        # when imported into a stock (non-ASAN) python.
        cflags += ["-fsanitize=address", "-shared-libasan"]
        ldflags += ["-fsanitize=address", "-shared-libasan"]
    # the .o mtime cache is invalid when the flag set changes: stamp it
    stamp = objdir / ".flags"
    flags_now = " ".join(cflags)
    if not stamp.exists() or stamp.read_text() != flags_now:
        for o in objdir.glob("*.o"):
            o.unlink()
        stamp.write_text(flags_now)
    objs = []
    procs = []
    for src in SOURCES:
        sp = ROOT / src
        obj = objdir / (sp.stem + ".o")
        objs.append(str(obj))
        if obj.exists() and obj.stat().st_mtime > max(
            sp.stat().st_mtime,
            max((h.stat().st_mtime for h in (ROOT / "csrc").rglob("*.h")), default=0),
        ):
            continue
        cmd = [HIPCC, "-c", str(sp), "-o", str(obj)] + cflags + incs
        if src.endswith(".hip"):
            cmd += ["-x", "hip"]
        if verbose:
            print("[hippt build]", " ".join(cmd), flush=True)
        procs.append(subprocess.Popen(cmd))
    for p in procs:
        if p.wait() != 0:
            raise SystemExit(f"hipcc failed ({p.args[2]})")
    link = [HIPCC, "-shared", "-fPIC", f"--offload-arch={ARCH}", "-o", str(out)] + objs + ldflags
    if verbose:
        print("[hippt build]", " ".join(link), flush=True)
    subprocess.check_call(link)
    return out


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "clean":
        import shutil
        shutil.rmtree(ROOT / "build", ignore_errors=True)
        sys.exit(0)
    build()
