"""Compile-time kernel resource audit (no GPU needed): hipcc cross-compiles
each gfx950 kernel with -Rpass-analysis=kernel-resource-usage and this script
enforces the budgets that kept round-1 performance:

- ZERO scratch and ZERO spills everywhere (guide rule #20: a runtime-indexed
  register array silently spills and costs 3x — measured on the decode
  kernel, 3.5 -> 1.0 TB/s),
- occupancy floors for the hot kernels (occupancy drops are how attention
  regressions sneak in: v7 runs 3 waves/SIMD at D=64, 2 at D=128-causal).

Usage: python scripts/check_kernel_resources.py [file.hip ...]
"""
import re
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent.parent / "modal_examples_amd" / "ops" / "csrc"

# kernel-name substring -> minimum waves/SIMD (from the shipped v7/K6 builds)
OCCUPANCY_FLOORS = {
    "fa32_kernelILi64ELb1": 3,   # D=64 causal
    "fa32_kernelILi64ELb0": 3,   # D=64
    "fa32_kernelILi128ELb1": 2,  # D=128 causal (llama prefill)
    "decode_kernel": 2,
    "decode_merge_kernel": 4,
}


def audit(path: Path) -> dict:
    r = subprocess.run(
        ["hipcc", "--offload-arch=gfx950", "-O3", "-c", str(path),
         "-o", "/dev/null", "-Rpass-analysis=kernel-resource-usage"],
        capture_output=True, text=True, timeout=600)
    if r.returncode != 0:
        raise SystemExit(f"{path.name}: compile failed\n{r.stderr[-2000:]}")
    kernels, cur = {}, None
    for line in r.stderr.splitlines():
        m = re.search(r"Function Name: (\S+)", line)
        if m:
            cur = m.group(1)
            kernels[cur] = {}
            continue
        m = re.search(r"remark:\s+([A-Za-z ]+?)(?: \[[^\]]+\])?: (\S+)", line)
        if m and cur:
            kernels[cur][m.group(1).strip()] = m.group(2)
    return kernels


def main(files):
    failures = []
    for f in files:
        for name, k in audit(f).items():
            scratch = int(k.get("ScratchSize", "0"))
            vspill = int(k.get("VGPRs Spill", "0"))
            sspill = int(k.get("SGPRs Spill", "0"))
            occ = int(k.get("Occupancy", "0"))
            line = (f"{f.name:24s} {name[:52]:52s} VGPR={k.get('VGPRs', '?'):>4s}"
                    f" AGPR={k.get('AGPRs', '?'):>4s} occ={occ} scratch={scratch}")
            print(line)
            if scratch or vspill or sspill:
                failures.append(f"{name}: scratch={scratch} spills={vspill}/{sspill}")
            for frag, floor in OCCUPANCY_FLOORS.items():
                if frag in name and occ < floor:
                    failures.append(f"{name}: occupancy {occ} < floor {floor}")
    if failures:
        print("\nBUDGET VIOLATIONS:")
        for x in failures:
            print(" -", x)
        return 1
    print("\nall kernels within budget (no scratch, no spills, occupancy floors hold)")
    return 0


if __name__ == "__main__":
    files = [Path(a) for a in sys.argv[1:]] or sorted(CSRC.glob("*.hip"))
    raise SystemExit(main(files))
