"""Build the in-tree HIP kernel library for gfx950.

Usage: ``python -m ddlw_amd.ops.build`` (also called by ``__graft_entry__.build``).
Raw ``hipcc --offload-arch=gfx950`` — no hipify, no CUDA path. The resulting
``libddlw_kernels.so`` sits next to this file so it travels with repo
snapshots to GPU boxes (it is git-ignored; history stays source-only).
"""
from __future__ import annotations

import hashlib
import json
import subprocess
import sys
from pathlib import Path

HERE = Path(__file__).resolve().parent
HIP_DIR = HERE / "hip"
OUT = HERE / "libddlw_kernels.so"
STAMP = HERE / ".build_stamp.json"

SOURCES = sorted(HIP_DIR.glob("*.hip"))
HIPCC = "/opt/rocm/bin/hipcc"

FLAGS = [
    "--offload-arch=gfx950",
    "-O3",
    "-std=c++17",
    "-fPIC",
    "-shared",
    "-ffast-math",
    "-fvisibility=hidden",
]


def _digest() -> str:
    h = hashlib.sha256()
    for f in SOURCES + sorted(HIP_DIR.glob("*.h")):
        h.update(f.name.encode())
        h.update(f.read_bytes())
    h.update(" ".join(FLAGS).encode())
    return h.hexdigest()


def build(force: bool = False, verbose: bool = True) -> Path:
    dig = _digest()
    if not force and OUT.exists() and STAMP.exists():
        try:
            if json.loads(STAMP.read_text()).get("digest") == dig:
                if verbose:
                    print(f"[ddlw.ops.build] up to date: {OUT}")
                return OUT
        except Exception:
            pass
    cmd = [HIPCC, *FLAGS, *[str(s) for s in SOURCES], "-o", str(OUT)]
    if verbose:
        print("[ddlw.ops.build]", " ".join(cmd), flush=True)
    res = subprocess.run(cmd, capture_output=True, text=True)
    if res.returncode != 0:
        sys.stderr.write(res.stdout + res.stderr)
        raise RuntimeError(f"hipcc failed ({res.returncode})")
    if res.stderr.strip() and verbose:
        sys.stderr.write(res.stderr)
    STAMP.write_text(json.dumps({"digest": dig}))
    if verbose:
        print(f"[ddlw.ops.build] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
