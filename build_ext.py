"""In-tree build of starway_amd._core with hipcc (gfx950).

Used by setup.py, __graft_entry__.build() and tests. Compiles each TU to
build/*.o with mtime-based incrementality, then links the extension .so into
starway_amd/ so it travels with the repo snapshot to GPU boxes.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent
CSRC = REPO / "csrc"
BUILD = REPO / "build"
PKG = REPO / "starway_amd"

SOURCES = ["engine.cpp", "gpu.cpp", "module.cpp", "kernels.hip", "smallmsg.hip"]
RCCL_SOURCES = ["rccl_group.cpp"]

HIPCC = os.environ.get("STARWAY_HIPCC", "hipcc")
ARCH = os.environ.get("STARWAY_OFFLOAD_ARCH", "gfx950")


def _pybind11_include() -> str:
    import pybind11

    return pybind11.get_include()


def ext_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return PKG / f"_core{suffix}"


def _needs_rebuild(obj: Path, src: Path, hdrs: list[Path]) -> bool:
    if not obj.exists():
        return True
    omt = obj.stat().st_mtime
    if src.stat().st_mtime > omt:
        return True
    return any(h.stat().st_mtime > omt for h in hdrs)


def build(verbose: bool = True, debug: bool = False) -> Path:
    BUILD.mkdir(exist_ok=True)
    hdrs = sorted(CSRC.glob("*.hpp"))
    py_inc = sysconfig.get_paths()["include"]
    common = [
        "-O3",
        "-std=c++20",
        f"--offload-arch={ARCH}",
        "-fPIC",
        "-I",
        str(CSRC),
        "-I",
        py_inc,
        "-I",
        _pybind11_include(),
        "-Wno-unused-result",
    ]
    if debug:
        common += ["-g", "-DSW_DEBUG"]
    else:
        common += ["-DNDEBUG"]

    objs: list[Path] = []
    linked_any = False
    for name in SOURCES:
        src = CSRC / name
        obj = BUILD / (name.replace(".", "_") + ".o")
        objs.append(obj)
        if _needs_rebuild(obj, src, hdrs):
            cmd = [HIPCC, *common, "-c", str(src), "-o", str(obj)]
            if verbose:
                print("[build_ext]", " ".join(cmd), flush=True)
            subprocess.run(cmd, check=True)
            linked_any = True

    out = ext_path()
    if linked_any or not out.exists():
        cmd = [HIPCC, "-shared", "-fPIC", *[str(o) for o in objs], "-o", str(out)]
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)

    # _rccl: separate module so importing starway_amd does not load librccl.
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    rccl_out = PKG / f"_rccl{suffix}"
    rccl_objs: list[Path] = []
    rccl_linked = False
    for name in RCCL_SOURCES:
        src = CSRC / name
        obj = BUILD / (name.replace(".", "_") + ".o")
        rccl_objs.append(obj)
        if _needs_rebuild(obj, src, hdrs):
            cmd = [HIPCC, *common, "-I", "/opt/rocm/include", "-c", str(src),
                   "-o", str(obj)]
            if verbose:
                print("[build_ext]", " ".join(cmd), flush=True)
            subprocess.run(cmd, check=True)
            rccl_linked = True
    if rccl_linked or not rccl_out.exists():
        cmd = [HIPCC, "-shared", "-fPIC", *[str(o) for o in rccl_objs],
               "-L/opt/rocm/lib", "-lrccl", "-o", str(rccl_out)]
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)

    # bin/copy_bench: standalone kernel microbench (PMC profiling target).
    bench_bin = REPO / "bin" / "copy_bench"
    bench_src = CSRC / "copy_bench_main.hip"
    bench_obj = BUILD / "copy_bench_main.o"
    if bench_src.exists():
        (REPO / "bin").mkdir(exist_ok=True)
        if _needs_rebuild(bench_obj, bench_src, hdrs) or not bench_bin.exists():
            subprocess.run([HIPCC, *common, "-c", str(bench_src), "-o",
                            str(bench_obj)], check=True)
            subprocess.run([HIPCC, str(bench_obj),
                            str(BUILD / "kernels_hip.o"), "-o",
                            str(bench_bin)], check=True)
    return out


def build_sanitizer(kind: str = "thread", verbose: bool = True) -> Path:
    """Sanitizer stress binary: the engine + shm + a GPU stub compiled with
    g++ -fsanitize={thread,address} and an embedded Python interpreter
    (csrc/sanitizer_stress_main.cpp). No HIP involved — the engine's
    cross-thread contracts are identical with or without a device."""
    BUILD.mkdir(exist_ok=True)
    (REPO / "bin").mkdir(exist_ok=True)
    py_inc = sysconfig.get_paths()["include"]
    ldlib = sysconfig.get_config_var("LDLIBRARY") or ""
    libdir = sysconfig.get_config_var("LIBDIR") or "/usr/lib"
    pyver = sysconfig.get_config_var("LDVERSION") or "3.10"
    out = REPO / "bin" / f"{kind[0]}san_stress"
    cmd = [
        "g++", "-O1", "-g", "-std=c++20", f"-fsanitize={kind}",
        "-fno-omit-frame-pointer",
        "-I", str(CSRC), "-I", py_inc, "-I", _pybind11_include(),
        str(CSRC / "engine.cpp"), str(CSRC / "gpu_stub.cpp"),
        str(CSRC / "sanitizer_stress_main.cpp"),
        f"-L{libdir}", f"-lpython{pyver}", "-lpthread", "-o", str(out),
    ]
    if verbose:
        print("[build_ext]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    if "--tsan" in sys.argv:
        print(f"built {build_sanitizer('thread')}")
    elif "--asan" in sys.argv:
        print(f"built {build_sanitizer('address')}")
    else:
        build(debug="--debug" in sys.argv)
        print(f"built {ext_path()}")
