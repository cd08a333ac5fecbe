"""Compile-check the round-2 prototype and report its register/LDS budget.

NOT part of the shipped library: fei_amd/ops/build.py does not include
this TU, build() does not compile it, and nothing loads the output. Run:

    python experimental/build.py

Exit nonzero if hipcc rejects the kernel or the VGPR budget regresses
past 4-waves/SIMD occupancy (128 VGPRs).
"""

import os
import re
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "stream_engine_proto.hip")
OUT = os.path.join(HERE, "stream_engine_proto.o")
HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("FEI_AMD_ARCH", "gfx950")
READELF = "/opt/rocm/lib/llvm/bin/llvm-readelf"


def main() -> int:
    hsaco = os.path.join(HERE, "stream_engine_proto.hsaco")
    cmd = [HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "--genco",
           SRC, "-o", hsaco]
    print("[experimental]", " ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stderr[-4000:])
        return 1

    # the bundle wraps a device ELF; its msgpack notes carry the kernel
    # register/LDS budget
    data = open(hsaco, "rb").read()
    i = data.find(b"\x7fELF")
    elf = os.path.join(HERE, "stream_engine_proto.elf")
    with open(elf, "wb") as f:
        f.write(data[i:])
    d = subprocess.run([READELF, "--notes", elf],
                       capture_output=True, text=True)
    stats = {}
    for key in ("vgpr_count", "sgpr_count", "group_segment_fixed_size",
                "vgpr_spill_count", "sgpr_spill_count", "agpr_count"):
        m = re.search(rf"\.{key}:\s+(\d+)", d.stdout)
        if m:
            stats[key] = int(m.group(1))
    print("[experimental] kernel stats:", stats)
    if stats.get("vgpr_count", 999) > 128:
        print("[experimental] FAIL: VGPR budget blown (>128)")
        return 2
    if stats.get("vgpr_spill_count", 0) or stats.get("sgpr_spill_count", 0):
        print("[experimental] FAIL: register spills")
        return 3
    lds = stats.get("group_segment_fixed_size", 0)
    if not (128 * 1024 <= lds <= 160 * 1024):
        print(f"[experimental] FAIL: LDS {lds} B — ring missing or oversized"
              " (the compiler once eliminated it; see glds16_nt notes)")
        return 4
    print("[experimental] compile check OK "
          f"(VGPR {stats.get('vgpr_count')}, SGPR {stats.get('sgpr_count')}, "
          f"LDS {lds} B, no spills)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
