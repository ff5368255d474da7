#!/usr/bin/env python3
"""Static ISA audit of the shipped gfx950 code objects (no GPU needed).

Extracts the HIP fat binary from the in-tree extension, disassembles with
llvm-objdump, and reports per-kernel counts of the instruction classes the
design claims: MFMA matrix ops, LDS traffic (ds_*), wide global/buffer
loads, system-scope atomics, and wave-level DPP/permute shuffles.

Usage: python scripts/isa_audit.py [path/to/_hip_ops*.so]"""
import re
import subprocess
import sys
import tempfile
from collections import Counter, defaultdict
from pathlib import Path

OBJDUMP = "/opt/rocm/lib/llvm/bin/llvm-objdump"
EXTRACT = "/opt/rocm/lib/llvm/bin/clang-offload-bundler"

CLASSES = [
    ("mfma", re.compile(r"\bv_mfma_\S+")),
    ("lds(ds_*)", re.compile(r"\bds_(read|write|load|store)\S*")),
    ("global_load_wide", re.compile(r"\bglobal_load_(dwordx4|b128)\b")),
    ("global_load", re.compile(r"\bglobal_load_\S+")),
    ("buffer_load", re.compile(r"\bbuffer_load_\S+")),
    ("atomic", re.compile(r"\b(global|buffer|flat|ds)_atomic\S*")),
    ("shuffle(dpp/perm)", re.compile(r"\b(v_mov_b32_dpp|ds_bpermute\S*|ds_permute\S*|ds_swizzle\S*|v_permlane\S+)")),
]


def main():
    so = Path(sys.argv[1]) if len(sys.argv) > 1 else next(
        Path("harmony_amd/ops").glob("_hip_ops*.so"))
    so = so.resolve()
    with tempfile.TemporaryDirectory() as td:
        # llvm-objdump --offloading writes the extracted bundles NEXT TO
        # the input file, so work on a copy inside the temp dir
        import shutil
        cp = Path(td) / so.name
        shutil.copy2(so, cp)
        r = subprocess.run([OBJDUMP, "--offloading", str(cp)],
                           capture_output=True, text=True, cwd=td)
        hsacos = [p for p in Path(td).iterdir()
                  if "gfx950" in p.name and p != cp]
        if not hsacos:
            print("objdump --offloading produced no images:", r.stderr[-400:])
            sys.exit(1)
        per_kernel = defaultdict(Counter)
        for img in hsacos:
            d = subprocess.run([OBJDUMP, "-d", "--mcpu=gfx950", str(img)],
                               capture_output=True, text=True)
            kern = None
            for line in d.stdout.splitlines():
                m = re.match(r"^[0-9a-f]+ <(.+)>:", line)
                if m:
                    kern = m.group(1)
                    continue
                if kern is None:
                    continue
                for name, pat in CLASSES:
                    if pat.search(line):
                        per_kernel[kern][name] += 1
                        break
        rows = []
        for k, c in per_kernel.items():
            if not any(c.values()):
                continue
            short = re.sub(r"\(.*", "", k)
            short = short.replace("_ZN12_GLOBAL__N_1", "").strip()
            rows.append((short[:52], c))
        rows.sort(key=lambda r: -sum(r[1].values()))
        hdr = ["kernel"] + [n for n, _ in CLASSES]
        print("| " + " | ".join(hdr) + " |")
        print("|" + "---|" * len(hdr))
        for name, c in rows[:40]:
            print("| `" + name + "` | "
                  + " | ".join(str(c.get(n, 0)) for n, _ in CLASSES) + " |")


if __name__ == "__main__":
    main()
