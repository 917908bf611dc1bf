#!/usr/bin/env python3
"""Static kernel-resource verification (guide rule 20): compile every HIP
source with -Rpass-analysis=kernel-resource-usage and regenerate
profiles/kernel_resources.md. Fails (exit 1) if any kernel spills."""
import re
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent.parent
HIP = sorted((REPO / "ddlw_amd" / "ops" / "hip").glob("*.hip"))
OUT = REPO / "profiles" / "kernel_resources.md"


def demangle(name: str) -> str:
    try:
        return subprocess.run(["/opt/rocm/lib/llvm/bin/llvm-cxxfilt", name],
                              capture_output=True, text=True).stdout.strip()
    except Exception:
        return name


def main() -> int:
    rows = []
    for f in HIP:
        res = subprocess.run(
            ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-O3",
             "-std=c++17", "-fPIC", "-c", str(f), "-o", "/dev/null",
             "-Rpass-analysis=kernel-resource-usage"],
            capture_output=True, text=True)
        cur = {}
        for line in (res.stdout + res.stderr).splitlines():
            m = re.search(r"Function Name: (\S+)", line)
            if m:
                if cur.get("name"):
                    rows.append(cur)
                cur = {"name": m.group(1), "file": f.name}
                continue
            for key, pat in (
                ("vgpr", r"    VGPRs: (\d+)"),
                ("agpr", r"AGPRs: (\d+)"),
                ("spill", r"VGPRs Spill: (\d+)"),
                ("sspill", r"SGPRs Spill: (\d+)"),
                ("scratch", r"ScratchSize \[bytes/lane\]: (\d+)"),
                ("lds", r"LDS Size \[bytes/block\]: (\d+)"),
                ("occ", r"Occupancy \[waves/SIMD\]: (\d+)"),
            ):
                m = re.search(pat, line)
                if m:
                    cur[key] = int(m.group(1))
        if cur.get("name"):
            rows.append(cur)

    bad = [r for r in rows if r.get("spill", 0) or r.get("sspill", 0)
           or r.get("scratch", 0)]
    lines = [
        "# Kernel resource usage (gfx950, hipcc -Rpass-analysis=kernel-resource-usage)",
        "",
        "Static verification that NO kernel spills (guide rule 20): every kernel",
        "must report ScratchSize 0 and VGPRs/SGPRs Spill 0. Regenerate with",
        "`python bench/tools/kernel_resources.py`.",
        "",
        f"{len(rows)} kernels, {len(bad)} with spills.",
        "",
        "| kernel | file | VGPRs | AGPRs | spill | scratch B/lane | LDS B | waves/SIMD |",
        "|---|---|---|---|---|---|---|---|",
    ]
    for r in sorted(rows, key=lambda r: (r["file"], r["name"])):
        nm = demangle(r["name"]).split("(")[0]
        lines.append(
            f"| `{nm}` | {r['file']} | {r.get('vgpr','?')} | {r.get('agpr',0)} "
            f"| {r.get('spill',0)}+{r.get('sspill',0)} | {r.get('scratch','?')} "
            f"| {r.get('lds','?')} | {r.get('occ','?')} |")
    OUT.write_text("\n".join(lines) + "\n")
    print(f"wrote {OUT}: {len(rows)} kernels, {len(bad)} spilling")
    for r in bad:
        print("SPILL:", r)
    return 1 if bad else 0


if __name__ == "__main__":
    sys.exit(main())
