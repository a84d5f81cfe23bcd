#!/usr/bin/env python3
"""Static ISA resource report for the in-tree HIP extension.

Splits the .hip_fatbin section of agilerl_amd/ops/_hip_ops*.so into its
per-TU offload bundles, unbundles the gfx950 code objects and prints a
per-kernel table of VGPR/SGPR counts, LDS bytes and spill counts —
compile-time occupancy evidence to go with the rocprofv3 runtime profiles.
"""

import glob
import os
import re
import subprocess
import sys
import tempfile

LLVM = "/opt/rocm/lib/llvm/bin"
MAGIC = b"__CLANG_OFFLOAD_BUNDLE__"
TARGET = "hipv4-amdgcn-amd-amdhsa--gfx950"


def main():
    sos = glob.glob(os.path.join(os.path.dirname(__file__), "..",
                                 "agilerl_amd", "ops", "_hip_ops*.so"))
    if not sos:
        sys.exit("extension .so not found — build with setup.py build_ext --inplace")
    so = sos[0]
    with tempfile.TemporaryDirectory() as td:
        fat = os.path.join(td, "fatbin.bin")
        subprocess.run([f"{LLVM}/llvm-objcopy", "-O", "binary",
                        "--only-section=.hip_fatbin", so, fat], check=True)
        blob = open(fat, "rb").read()
        offsets = [m.start() for m in re.finditer(re.escape(MAGIC), blob)]
        rows = []
        for i, off in enumerate(offsets):
            end = offsets[i + 1] if i + 1 < len(offsets) else len(blob)
            part = os.path.join(td, f"part{i}.bin")
            open(part, "wb").write(blob[off:end])
            dev = os.path.join(td, f"dev{i}.o")
            r = subprocess.run([f"{LLVM}/clang-offload-bundler", "--unbundle",
                                "--type=o", f"--input={part}",
                                f"--targets={TARGET}", f"--output={dev}"],
                               capture_output=True)
            if r.returncode != 0:
                continue
            notes = subprocess.run([f"{LLVM}/llvm-readelf", "--notes", dev],
                                   capture_output=True, text=True).stdout
            for block in re.split(r"\.agpr_count:", notes)[1:]:
                name = re.search(r"\.name:\s+(\S+)", block)
                if not name:
                    continue
                demangled = subprocess.run(["c++filt", name.group(1)],
                                           capture_output=True, text=True).stdout.strip()
                short = demangled.split("(")[0]

                def grab(key, default="0"):
                    m = re.search(rf"\.{key}:\s+(\d+)", block)
                    return m.group(1) if m else default

                rows.append((short, grab("vgpr_count"), grab("sgpr_count"),
                             grab("group_segment_fixed_size"),
                             grab("vgpr_spill_count"), grab("sgpr_spill_count")))
    rows.sort()
    print(f"| kernel | VGPR | SGPR | LDS B | vspill | sspill |")
    print("|---|---|---|---|---|---|")
    for r in rows:
        print("| " + " | ".join(r) + " |")


if __name__ == "__main__":
    main()
