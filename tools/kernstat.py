#!/usr/bin/env python3
"""Print VGPR/AGPR/scratch/occupancy per kernel from a gfx950 object file
(reads the AMDGPU msgpack metadata note). Usage: kernstat.py obj.o [filter]"""
import subprocess, sys, re

def main():
    obj, filt = sys.argv[1], (sys.argv[2] if len(sys.argv) > 2 else "")
    out = subprocess.run(
        ["/opt/rocm/lib/llvm/bin/llvm-readobj", "--notes", obj],
        capture_output=True, text=True, check=True).stdout
    # llvm-readobj prints the metadata msgpack as YAML-ish text
    kernels = re.split(r"\n\s+- \.agpr_count:", out)
    for k in kernels[1:]:
        k = ".agpr_count:" + k
        def f(key, default="?"):
            m = re.search(rf"\.{key}:\s+(\S+)", k)
            return m.group(1) if m else default
        name = f("name")
        if filt and filt not in name:
            continue
        vgpr = int(f("vgpr_count", "0"))
        agpr = int(f("agpr_count", "0"))
        alloc = ((vgpr + agpr + 7) // 8) * 8
        waves = min(8, 512 // max(alloc, 1))
        print(f"{name}: vgpr={vgpr} agpr={agpr} sgpr={f('sgpr_count')} "
              f"scratch={f('private_segment_fixed_size')}B lds={f('group_segment_fixed_size')}B "
              f"spill_v={f('vgpr_spill_count', '0')} waves/SIMD={waves}")

if __name__ == "__main__":
    main()
