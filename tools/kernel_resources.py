#!/usr/bin/env python3
"""Per-kernel resource usage (VGPR/SGPR/AGPR/LDS/spill/occupancy) from the
built extension's embedded gfx950 code object. Works without a GPU.

Occupancy on CDNA4: VGPRs+AGPRs come from one 512-register file per SIMD;
waves/SIMD ~= 512 // (vgpr+agpr rounded up to 8), capped at 8 by LDS
(160 KB/CU) and workgroup shape.

Usage: python tools/kernel_resources.py [path/to/lib.so]
"""

import re
import struct
import subprocess
import sys
import os

LLVM = "/opt/rocm/lib/llvm/bin"


def extract_hsaco(so_path: str) -> bytes:
    data = open(so_path, "rb").read()
    off = data.find(b"__CLANG_OFFLOAD_BUNDLE__")
    if off < 0:
        raise SystemExit(f"no offload bundle in {so_path}")
    p = off + 24
    n, = struct.unpack_from("<Q", data, p)
    p += 8
    for _ in range(n):
        eoff, esz, idl = struct.unpack_from("<QQQ", data, p)
        p += 24
        ident = data[p: p + idl].decode()
        p += idl
        if "gfx950" in ident:
            return data[off + eoff: off + eoff + esz]
    raise SystemExit("no gfx950 entry in bundle")


def demangle(names):
    out = subprocess.run(["c++filt"], input="\n".join(names),
                         capture_output=True, text=True)
    return out.stdout.splitlines()


def main():
    so = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        os.path.dirname(os.path.abspath(__file__)), "..",
        "dllama_amd", "ops", "_build", "dllama_hip.so")
    hsaco = extract_hsaco(so)
    tmp = "/tmp/_kres.hsaco"
    open(tmp, "wb").write(hsaco)
    notes = subprocess.run([f"{LLVM}/llvm-readelf", "--notes", tmp],
                           capture_output=True, text=True).stdout
    kernels = []
    cur = {}
    for line in notes.splitlines():
        m = re.match(r"\s*- \.agpr_count:\s*(\d+)", line)
        if m:
            if cur:
                kernels.append(cur)
            cur = {"agpr": int(m.group(1))}
        for key, pat in (("name", r"\.name:\s*(\S+)"),
                         ("sgpr", r"\.sgpr_count:\s*(\d+)"),
                         ("vgpr", r"\.vgpr_count:\s*(\d+)"),
                         ("spill", r"\.private_segment_fixed_size:\s*(\d+)"),
                         ("lds", r"\.group_segment_fixed_size:\s*(\d+)")):
            m = re.search(pat, line)
            if m:
                cur[key] = m.group(1) if key == "name" else int(m.group(1))
    if cur:
        kernels.append(cur)
    names = demangle([k.get("name", "?") for k in kernels])
    print(f"{'kernel':44s} {'VGPR':>5s} {'AGPR':>5s} {'SGPR':>5s} "
          f"{'LDS':>6s} {'spill':>5s} {'waves/SIMD':>10s}")
    for k, nm in sorted(zip(kernels, names),
                        key=lambda t: -(t[0].get("vgpr", 0) + t[0].get("agpr", 0))):
        short = nm.split("(")[0]
        tot = k.get("vgpr", 0) + k.get("agpr", 0)
        waves = min(8, 512 // max(8, (tot + 7) // 8 * 8)) if tot else 8
        print(f"{short:44s} {k.get('vgpr', 0):5d} {k.get('agpr', 0):5d} "
              f"{k.get('sgpr', 0):5d} {k.get('lds', 0):6d} "
              f"{k.get('spill', 0):5d} {waves:10d}")


if __name__ == "__main__":
    main()
