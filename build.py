#!/usr/bin/env python3
"""In-tree build of the megba_amd native core.

Compiles all C++/HIP sources with hipcc (device code for gfx950 only) and
links one extension module megba_amd/_core.<abi>.so.  The .so lives in-tree
so it travels with the repo snapshot to GPU boxes.  Caching is mtime-based.

Usage: python build.py [--force] [--debug]
"""
import os
import subprocess
import sys
import sysconfig
import concurrent.futures as cf
from pathlib import Path

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "csrc"
BUILD = ROOT / "build" / "obj"
PKG = ROOT / "megba_amd"

HIPCC = os.environ.get("HIPCC", "hipcc")
GPU_ARCH = os.environ.get("MEGBA_GPU_ARCH", "gfx950")


def pybind_includes():
    import pybind11
    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


SOURCES = [
    # (path, is_device_code)
    ("megba/cpu_engine.cpp", False),
    ("megba/gpu/gpu_engine.hip", True),
    # one TU per block-dimension set so hipcc compiles them in parallel
    ("megba/gpu/gpu_dims_932.hip", True),
    ("megba/gpu/gpu_dims_632.hip", True),
    ("megba/gpu/gpu_dims_432.hip", True),
    ("megba/gpu/gpu_dims_933.hip", True),
    ("megba/gpu/gpu_dims_633.hip", True),
    ("megba/gpu/gpu_dims_433.hip", True),
    ("megba/jv/jetvector.hip", True),
    ("bindings.cpp", False),
]

COMMON_FLAGS = [
    "-O3", "-std=c++17", "-fPIC", "-fopenmp",
    "-I", str(CSRC),
    "-I", "/opt/rocm/include",
    "-DMEGBA_WITH_GPU",
]
DEVICE_FLAGS = [f"--offload-arch={GPU_ARCH}"]


def newest_header_mtime():
    mt = 0.0
    for p in CSRC.rglob("*.hpp"):
        mt = max(mt, p.stat().st_mtime)
    return mt


def compile_one(src, is_device, force, extra_flags, hdr_mtime):
    src_path = CSRC / src
    obj = BUILD / (src.replace("/", "_") + ".o")
    if (not force and obj.exists()
            and obj.stat().st_mtime > max(src_path.stat().st_mtime, hdr_mtime)):
        return obj
    obj.parent.mkdir(parents=True, exist_ok=True)
    cmd = [HIPCC, *COMMON_FLAGS, *extra_flags]
    if is_device:
        cmd += DEVICE_FLAGS + ["-x", "hip"]
    cmd += pybind_flags() + ["-c", str(src_path), "-o", str(obj)]
    print("  CC", src, flush=True)
    subprocess.run(cmd, check=True)
    return obj


def pybind_flags():
    out = []
    for inc in pybind_includes():
        out += ["-I", inc]
    return out


EXAMPLES = ["bal_solve_cpp", "bal_custom_edge_cpp"]


def build_examples(objs, force=False):
    """Native C++ CLI examples linked against the same objects."""
    hdr = newest_header_mtime()
    core = [o for o in objs if "bindings" not in o.name]
    for name in EXAMPLES:
        src = ROOT / "examples" / f"{name}.cpp"
        out = ROOT / "examples" / name
        obj = BUILD / f"examples_{name}.o"
        if force or not obj.exists() or obj.stat().st_mtime < max(
                src.stat().st_mtime, hdr):
            print(f"  CC examples/{name}.cpp", flush=True)
            subprocess.run([HIPCC, *COMMON_FLAGS, "-c", str(src),
                            "-o", str(obj)], check=True)
        if force or not out.exists() or any(
                o.stat().st_mtime > out.stat().st_mtime for o in core + [obj]):
            print(f"  LD examples/{name}", flush=True)
            subprocess.run([HIPCC, "-fopenmp", str(obj),
                            *[str(o) for o in core],
                            "-L/opt/rocm/lib", "-lrccl", "-lamdhip64",
                            "-o", str(out)], check=True)


def build(force=False, debug=False):
    BUILD.mkdir(parents=True, exist_ok=True)
    extra = ["-g"] if debug else []
    hdr = newest_header_mtime()
    with cf.ThreadPoolExecutor(max_workers=len(SOURCES)) as ex:
        objs = list(
            ex.map(lambda s: compile_one(s[0], s[1], force, extra, hdr), SOURCES))
    ext = sysconfig.get_config_var("EXT_SUFFIX")
    out = PKG / f"_core{ext}"
    need_link = force or not out.exists() or any(
        o.stat().st_mtime > out.stat().st_mtime for o in objs)
    if need_link:
        cmd = [HIPCC, "-shared", "-fopenmp", *[str(o) for o in objs],
               "-L/opt/rocm/lib", "-lrccl", "-lamdhip64", "-o", str(out)]
        print("  LD", out.name, flush=True)
        subprocess.run(cmd, check=True)
    build_examples(objs, force)
    print("built", out)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv, debug="--debug" in sys.argv)
