#!/usr/bin/env python3
"""Compile every .hip kernel for gfx950 and tabulate the compiler's
kernel-resource-usage report (VGPR/SGPR/LDS/spills/occupancy).

  python scripts/kernel_resources.py > profiles/rNN_kernel_resources.md

Runs on the CPU-only builder (hipcc cross-compiles); no GPU needed.
"""
import glob
import os
import re
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = sorted(glob.glob(os.path.join(ROOT, "bdbnn_amd/csrc/*.hip")))
SRC = [s for s in SRC if not s.endswith("_hip.hip")]

FIELDS = ["TotalSGPRs", "VGPRs", "AGPRs", "ScratchSize [bytes/lane]",
          "Occupancy [waves/SIMD]", "SGPRs Spill", "VGPRs Spill",
          "LDS Size [bytes/block]"]


def demangle(name):
    try:
        out = subprocess.run([ "/opt/rocm/lib/llvm/bin/llvm-cxxfilt", name],
                             capture_output=True, text=True).stdout.strip()
        return out or name
    except OSError:
        return name


def main():
    rows = []
    for src in SRC:
        p = subprocess.run(
            ["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-O3",
             "-std=c++17", "-c", src, "-o", "/dev/null",
             "-Rpass-analysis=kernel-resource-usage"],
            capture_output=True, text=True, cwd=ROOT)
        cur = None
        for line in p.stderr.splitlines():
            m = re.search(r"Function Name: (\S+)", line)
            if m:
                if cur:
                    rows.append(cur)
                cur = {"file": os.path.basename(src),
                       "kernel": demangle(m.group(1))}
                continue
            for f in FIELDS:
                m = re.search(rf"{re.escape(f)}: (\S+)", line)
                if m and cur is not None:
                    cur[f] = m.group(1)
        if cur:
            rows.append(cur)
    print("# Kernel resource usage (gfx950, hipcc "
          "-Rpass-analysis=kernel-resource-usage)\n")
    print("| file | kernel | VGPR | SGPR | LDS B | scratch | waves/SIMD "
          "| spills (s/v) |")
    print("|---|---|---|---|---|---|---|---|")
    for r in rows:
        k = r["kernel"]
        k = re.sub(r"\(.*\)$", "", k)
        if len(k) > 76:
            k = k[:73] + "..."
        print(f"| {r['file']} | `{k}` | {r.get('VGPRs','?')} "
              f"| {r.get('TotalSGPRs','?')} "
              f"| {r.get('LDS Size [bytes/block]','?')} "
              f"| {r.get('ScratchSize [bytes/lane]','?')} "
              f"| {r.get('Occupancy [waves/SIMD]','?')} "
              f"| {r.get('SGPRs Spill','?')}/{r.get('VGPRs Spill','?')} |")


if __name__ == "__main__":
    main()
