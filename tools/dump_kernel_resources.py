"""Extract per-kernel register/LDS usage from the built gfx950 code
objects inside quintnet_amd/_C*.so (no GPU needed — compiled facts for
occupancy analysis).

    python tools/dump_kernel_resources.py [name-filter-regex]

Walks every __CLANG_OFFLOAD_BUNDLE__ in the .hip_fatbin section,
unbundles the gfx950 object and reads the msgpack kernel notes via
llvm-readelf.  Occupancy math (MI355X / CDNA4): 512 VGPRs per SIMD at
wave64 → waves/SIMD = 512 // vgpr_alloc (granule 8); 160 KB LDS per CU
bounds workgroups/CU.
"""
import os
import re
import subprocess
import sys
import tempfile

LLVM = "/opt/rocm/lib/llvm/bin"
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def kernels_from_so(so_path):
    with tempfile.TemporaryDirectory() as td:
        fat = os.path.join(td, "fat.bin")
        subprocess.run(
            [f"{LLVM}/llvm-objcopy", f"--dump-section=.hip_fatbin={fat}",
             so_path, "/dev/null"], check=True)
        data = open(fat, "rb").read()
        magic = b"__CLANG_OFFLOAD_BUNDLE__"
        offs = [m.start() for m in re.finditer(re.escape(magic), data)]
        rows = []
        for i, o in enumerate(offs):
            end = offs[i + 1] if i + 1 < len(offs) else len(data)
            bpath = os.path.join(td, f"b{i}.bin")
            opath = os.path.join(td, f"b{i}.o")
            open(bpath, "wb").write(data[o:end])
            r = subprocess.run(
                [f"{LLVM}/clang-offload-bundler", "--unbundle", "--type=o",
                 f"--input={bpath}",
                 "--targets=hipv4-amdgcn-amd-amdhsa--gfx950",
                 f"--output={opath}"], capture_output=True)
            if r.returncode:
                continue
            notes = subprocess.run(
                [f"{LLVM}/llvm-readelf", "--notes", opath],
                capture_output=True, text=True).stdout
            cur = {}
            for line in notes.splitlines():
                line = line.strip().lstrip("- ")
                m = re.match(r"\.(\w+):\s*(.*)", line)
                if not m:
                    continue
                k, v = m.groups()
                if k == "agpr_count" and cur.get("name"):
                    rows.append(cur)
                    cur = {}
                cur[k] = v.strip()
            if cur.get("name"):
                rows.append(cur)
        return rows


def demangle(n):
    r = subprocess.run(["c++filt", n], capture_output=True, text=True)
    return (r.stdout.strip() or n).split("(")[0]


def main():
    patt = re.compile(sys.argv[1]) if len(sys.argv) > 1 else None
    so = None
    for f in os.listdir(os.path.join(REPO, "quintnet_amd")):
        if f.startswith("_C") and f.endswith(".so"):
            so = os.path.join(REPO, "quintnet_amd", f)
    assert so, "build the extension first (setup.py build_ext --inplace)"
    seen = set()
    print(f"{'kernel':64s} {'vgpr':>5} {'agpr':>4} {'sgpr':>4} "
          f"{'spill':>5} {'lds_B':>7} {'waves/SIMD':>10}")
    for r in kernels_from_so(so):
        name = demangle(r["name"].replace(".kd", ""))
        if name in seen or (patt and not patt.search(name)):
            continue
        seen.add(name)
        vgpr = int(r.get("vgpr_count", 0))
        agpr = int(r.get("agpr_count", 0))
        alloc = max(((vgpr + agpr + 7) // 8) * 8, 8)
        waves = 512 // alloc if alloc else 0
        print(f"{name[:64]:64s} {vgpr:5d} {agpr:4d} "
              f"{int(r.get('sgpr_count', 0)):4d} "
              f"{int(r.get('vgpr_spill_count', 0)):5d} "
              f"{int(r.get('group_segment_fixed_size', 0)):7d} {waves:10d}")


if __name__ == "__main__":
    main()
