"""In-tree build of the hypha_amd native extensions.

hypha_amd/_C  — CDNA4 HIP kernels (attention, RMSNorm, RoPE, SwiGLU, CE,
                fused AdamW / Nesterov), compiled DIRECTLY with hipcc for
                gfx950 (no hipify, no CUDA-compat layer).
hypha_amd/_core — C++ control-plane library (messages, leases, scheduler
                  trackers/FSM, arbiter) bound with pybind11 (built when
                  cpp/ sources exist).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
HIP_DIR = ROOT / "hypha_amd" / "ops" / "hip"
CPP_DIR = ROOT / "cpp"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def torch_paths():
    import torch.utils.cpp_extension as ce

    return ce.include_paths(), ce.library_paths()


def _run(cmd):
    print("+", " ".join(str(c) for c in cmd), flush=True)
    subprocess.run([str(c) for c in cmd], check=True)


def _needs_rebuild(target: Path, sources) -> bool:
    if not target.exists():
        return True
    t = target.stat().st_mtime
    return any(Path(s).stat().st_mtime > t for s in sources)


def build_hip_extension():
    includes, libdirs = torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    sources = sorted(HIP_DIR.glob("*.hip")) + [HIP_DIR / "bindings.cpp"]
    out = ROOT / "hypha_amd" / "_C.so"
    headers = list(HIP_DIR.glob("*.h"))
    if not _needs_rebuild(out, sources + headers + [Path(__file__)]):
        print(f"{out} up to date")
        return
    objdir = ROOT / "build" / "hip"
    objdir.mkdir(parents=True, exist_ok=True)
    objs = []
    common = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DTORCH_EXTENSION_NAME=_C", "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-Wno-unused-result",
    ] + [f"-I{i}" for i in includes] + [f"-I{py_inc}", f"-I{HIP_DIR}"]
    procs = []
    for src in sources:
        obj = objdir / (src.stem + ".o")
        objs.append(obj)
        if _needs_rebuild(obj, [src] + headers):
            cmd = [HIPCC, *common, "-c", str(src), "-o", str(obj)]
            print("+", " ".join(cmd), flush=True)
            procs.append(subprocess.Popen(cmd))
    for p in procs:
        if p.wait() != 0:
            sys.exit(1)
    link = [HIPCC, "-shared", "-fPIC", *[str(o) for o in objs], "-o", str(out)]
    for d in libdirs:
        link += [f"-L{d}", f"-Wl,-rpath,{d}"]
    link += ["-ltorch", "-ltorch_python", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_cpu"]
    _run(link)
    print(f"built {out}")


def build_core_extension():
    """C++ control-plane pybind module (CPU-only, g++)."""
    srcs = sorted(CPP_DIR.glob("src/*.cpp")) + sorted(CPP_DIR.glob("bindings/*.cpp"))
    if not srcs:
        return
    import pybind11

    py_inc = sysconfig.get_paths()["include"]
    out = ROOT / "hypha_amd" / "_core.so"
    headers = list(CPP_DIR.glob("include/hypha/*.h")) + list(CPP_DIR.glob("include/hypha/*.hpp"))
    if not _needs_rebuild(out, srcs + headers + [Path(__file__)]):
        print(f"{out} up to date")
        return
    cmd = [
        "g++", "-O2", "-std=c++17", "-fPIC", "-shared", "-pthread",
        "-DTORCH_EXTENSION_NAME=_core",
        f"-I{CPP_DIR / 'include'}", f"-I{pybind11.get_include()}", f"-I{py_inc}",
        *[str(s) for s in srcs], "-lssl", "-lcrypto", "-o", str(out),
    ]
    _run(cmd)
    print(f"built {out}")


def build_daemons():
    """C++ control-plane daemon binaries -> bin/ (in-tree, travel with snapshot)."""
    srcs = sorted(CPP_DIR.glob("bin/*.cpp"))
    if not srcs:
        return
    bindir = ROOT / "bin"
    bindir.mkdir(exist_ok=True)
    lib_srcs = [str(s) for s in sorted(CPP_DIR.glob("src/*.cpp"))]
    headers = list(CPP_DIR.glob("include/hypha/*.h"))
    procs = []
    for src in srcs:
        exe = bindir / src.stem.replace("_", "-")
        if not _needs_rebuild(exe, [src] + lib_srcs + [str(h) for h in headers]):
            continue
        cmd = [
            "g++", "-O2", "-std=c++17", "-pthread",
            f"-I{CPP_DIR / 'include'}", str(src), *lib_srcs,
            "-lssl", "-lcrypto", "-o", str(exe),
        ]
        print("+", " ".join(cmd), flush=True)
        procs.append(subprocess.Popen(cmd))
    for p in procs:
        if p.wait() != 0:
            sys.exit(1)


if __name__ == "__main__":
    if "build_ext" in sys.argv or len(sys.argv) == 1:
        build_core_extension()
        build_daemons()
        build_hip_extension()
    else:
        print("usage: python setup.py build_ext --inplace")
