"""Build the native RCCL communicator library (``libddlw_rccl.so``) for the
Horovod-core-equivalent collective layer.

Usage: ``python -m ddlw_amd.parallel.build`` (also called by
``__graft_entry__.build``). Plain hipcc + ``-lrccl``; the ``.so`` sits
in-tree next to this file so it travels with repo snapshots to GPU boxes.
"""
from __future__ import annotations

import hashlib
import json
import subprocess
import sys
from pathlib import Path

HERE = Path(__file__).resolve().parent
SRC = HERE / "hip" / "rccl_comm.cpp"
OUT = HERE / "libddlw_rccl.so"
STAMP = HERE / ".build_stamp.json"

HIPCC = "/opt/rocm/bin/hipcc"
FLAGS = [
    "-O2",
    "-std=c++17",
    "-fPIC",
    "-shared",
    "-fvisibility=hidden",
    "-I/opt/rocm/include",
    "-L/opt/rocm/lib",
    "-lrccl",
]


def _digest() -> str:
    h = hashlib.sha256()
    h.update(SRC.read_bytes())
    h.update(" ".join(FLAGS).encode())
    return h.hexdigest()


def build(force: bool = False, verbose: bool = True) -> Path:
    dig = _digest()
    if not force and OUT.exists() and STAMP.exists():
        try:
            if json.loads(STAMP.read_text()).get("digest") == dig:
                if verbose:
                    print(f"[ddlw.parallel.build] up to date: {OUT}")
                return OUT
        except Exception:
            pass
    cmd = [HIPCC, *FLAGS, str(SRC), "-o", str(OUT)]
    if verbose:
        print("[ddlw.parallel.build]", " ".join(cmd), flush=True)
    res = subprocess.run(cmd, capture_output=True, text=True)
    if res.returncode != 0:
        sys.stderr.write(res.stdout + res.stderr)
        raise RuntimeError(f"hipcc failed ({res.returncode})")
    if res.stderr.strip() and verbose:
        sys.stderr.write(res.stderr)
    STAMP.write_text(json.dumps({"digest": dig}))
    if verbose:
        print(f"[ddlw.parallel.build] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
