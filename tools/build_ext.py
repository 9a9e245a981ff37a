#!/usr/bin/env python3
"""Builds the in-tree HIP/C++ extension (gfx950) with hipcc.

Produces ydf_amd/_ydf_ops.<abi>.so next to the package so the built artifact
travels to the GPU box with the repo snapshot. Incremental: per-source .o
files cached under build/ keyed on mtime.
"""
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
CC_DIR = REPO / "ydf_amd" / "ops" / "cc"
BUILD = REPO / "build" / "ops"
ARCH = os.environ.get("YDFA_OFFLOAD_ARCH", "gfx950")

SOURCES = [
    CC_DIR / "train_kernels.hip",
    CC_DIR / "infer_kernels.hip",
    CC_DIR / "cpu_ops.cpp",
    CC_DIR / "bindings.cpp",
]


def pybind_includes():
    import pybind11

    return [
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
    ]


def ext_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    return REPO / "ydf_amd" / f"_ydf_ops{suffix}"


def build(verbose: bool = True) -> Path:
    BUILD.mkdir(parents=True, exist_ok=True)
    out = ext_path()
    common = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-fvisibility=hidden",
        f"-I{CC_DIR}",
        *pybind_includes(),
    ]
    objs = []
    relink = not out.exists()
    headers = list(CC_DIR.glob("*.h"))
    hdr_mtime = max((h.stat().st_mtime for h in headers), default=0)
    for src in SOURCES:
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        if (
            obj.exists()
            and obj.stat().st_mtime > src.stat().st_mtime
            and obj.stat().st_mtime > hdr_mtime
        ):
            continue
        cmd = common + ["-x", "hip", "-c", str(src), "-o", str(obj)]
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
        relink = True
    if relink or any(o.stat().st_mtime > out.stat().st_mtime for o in objs):
        cmd = ["hipcc", "-shared", "-fPIC"] + [str(o) for o in objs] + [
            "-o",
            str(out),
        ]
        if verbose:
            print("[build_ext]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    p = build()
    print(f"built {p}")
    sys.exit(0)
