"""In-tree build of accl_amd._core.

Compiles every C++/HIP translation unit with hipcc for gfx950 (MI355X) and
links one Python extension. hipcc cross-compiles without a GPU, so this runs
on CPU-only machines; the built .so travels with the repo snapshot.
"""
import concurrent.futures
import os
import pathlib
import subprocess
import sysconfig

ROOT = pathlib.Path(__file__).resolve().parent
CSRC = ROOT / "csrc"
BUILD = ROOT / ".build"
HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(",")[0]

SOURCES = [
    "core/util.cpp",
    "core/accl.cpp",
    "emu/emudevice.cpp",
    "gpu/gpudevice.cpp",
    "gpu/engine.hip",
    "gpu/plugins.hip",
    "bindings/module.cpp",
]

HEADERS_GLOB = ["common/*.hpp", "core/*.hpp", "emu/*.hpp", "gpu/*.hpp"]


def _ext_suffix():
    return sysconfig.get_config_var("EXT_SUFFIX")


def so_path():
    return ROOT / ("_core" + _ext_suffix())


def _includes():
    import pybind11
    return [
        sysconfig.get_paths()["include"],
        pybind11.get_include(),
    ]


def _newest_header_mtime():
    newest = 0.0
    for pat in HEADERS_GLOB:
        for h in CSRC.glob(pat):
            newest = max(newest, h.stat().st_mtime)
    return newest


def build(force=False, verbose=False):
    BUILD.mkdir(exist_ok=True)
    out = so_path()
    hdr_m = _newest_header_mtime()
    inc = sum([["-I", i] for i in _includes()], [])
    cflags = ["--offload-arch=" + ARCH, "-O3", "-std=c++17", "-fPIC",
              "-DNDEBUG", "-Wno-unused-result"]

    objs = []
    jobs = []
    for src in SOURCES:
        sp = CSRC / src
        op = BUILD / (src.replace("/", "_") + ".o")
        objs.append(op)
        if (not force and op.exists()
                and op.stat().st_mtime > max(sp.stat().st_mtime, hdr_m)):
            continue
        cmd = [HIPCC, *cflags, *inc, "-c", str(sp), "-o", str(op)]
        jobs.append(cmd)

    def run(cmd):
        if verbose:
            print(" ".join(cmd))
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(
                f"compile failed: {' '.join(cmd)}\n{r.stdout}\n{r.stderr}")
        return cmd

    if jobs:
        with concurrent.futures.ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(run, jobs))

    if force or jobs or not out.exists() or any(
            o.stat().st_mtime > out.stat().st_mtime for o in objs):
        link = [HIPCC, "-shared", "-fPIC", *[str(o) for o in objs],
                "-o", str(out)]
        run(link)
    return out


if __name__ == "__main__":
    import sys
    build(force="--force" in sys.argv, verbose=True)
    print("built", so_path())
