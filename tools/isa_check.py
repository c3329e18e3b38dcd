"""Static ISA audit of the in-tree gfx950 code objects.

Extracts the .hip_fatbin bundles from the built extension, disassembles
each gfx950 code object, and reports per-kernel instruction mix +
resource metadata (the committed evidence in profiles/ comes from here).
Used by tests to guard two invariants:
  * matrix kernels (attention, skinny GEMM) issue v_mfma instructions
  * no hot kernel spills registers or allocates scratch
"""

from __future__ import annotations

import glob
import os
import re
import subprocess
import tempfile

LLVM = "/opt/rocm/lib/llvm/bin"
BUNDLER = "/opt/rocm/llvm/bin/clang-offload-bundler"
TARGET = "hipv4-amdgcn-amd-amdhsa--gfx950"
MAGIC = b"__CLANG_OFFLOAD_BUNDLE__"


def find_so() -> str | None:
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    hits = glob.glob(os.path.join(
        root, "quickstart_streaming_agents_amd", "qsa_hip*.so"))
    return hits[0] if hits else None


def extract_hsacos(so_path: str, workdir: str) -> list[str]:
    fat = os.path.join(workdir, "fatbin.bin")
    subprocess.run([os.path.join(LLVM, "llvm-objcopy"),
                    f"--dump-section=.hip_fatbin={fat}", so_path],
                   check=True, capture_output=True)
    data = open(fat, "rb").read()
    offs = []
    i = 0
    while True:
        j = data.find(MAGIC, i)
        if j < 0:
            break
        offs.append(j)
        i = j + 1
    out = []
    for n, o in enumerate(offs):
        end = offs[n + 1] if n + 1 < len(offs) else len(data)
        part = os.path.join(workdir, f"fb{n}.bin")
        open(part, "wb").write(data[o:end])
        hsaco = os.path.join(workdir, f"gfx950_{n}.hsaco")
        r = subprocess.run([BUNDLER, "--type=o", f"--input={part}",
                            f"--targets={TARGET}", f"--output={hsaco}",
                            "--unbundle"], capture_output=True)
        if r.returncode == 0 and os.path.exists(hsaco):
            out.append(hsaco)
    return out


def kernel_stats(hsacos: list[str]) -> dict[str, dict]:
    """kernel name -> {mfma, total} from disassembly + {vgpr_spill,
    sgpr_spill, scratch} from the metadata notes."""
    stats: dict[str, dict] = {}
    for h in hsacos:
        dis = subprocess.run(
            [os.path.join(LLVM, "llvm-objdump"), "-d", "--mcpu=gfx950", h],
            capture_output=True, text=True).stdout
        cur = None
        for line in dis.splitlines():
            m = re.match(r"^[0-9a-f]+ <(.+)>:", line)
            if m:
                cur = m.group(1)
                stats.setdefault(cur, {"mfma": 0, "total": 0})
                continue
            if cur and "\t" in line:
                stats[cur]["total"] += 1
                if "v_mfma" in line:
                    stats[cur]["mfma"] += 1
        notes = subprocess.run(
            [os.path.join(LLVM, "llvm-readelf"), "--notes", h],
            capture_output=True, text=True).stdout
        for blk in re.split(r"\n  - \.agpr_count:", notes)[1:]:
            name = re.search(r"\.name:\s*(\S+)", blk)
            if not name:
                continue
            st = stats.setdefault(name.group(1), {"mfma": 0, "total": 0})
            for field in ("vgpr_spill_count", "sgpr_spill_count",
                          "private_segment_fixed_size"):
                m = re.search(rf"\.{field}:\s*(\d+)", blk)
                st[field] = int(m.group(1)) if m else 0
    return stats


def main() -> int:
    so = find_so()
    if so is None:
        print("no built extension found")
        return 1
    with tempfile.TemporaryDirectory() as wd:
        stats = kernel_stats(extract_hsacos(so, wd))
    for name, st in sorted(stats.items()):
        if st["total"] > 60:
            print(f"{name[:64]:64s} instr={st['total']:5d} "
                  f"mfma={st['mfma']:3d} "
                  f"scratch={st.get('private_segment_fixed_size', 0)} "
                  f"spills={st.get('vgpr_spill_count', 0)}"
                  f"/{st.get('sgpr_spill_count', 0)}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
