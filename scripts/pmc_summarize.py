"""Summarize a rocprofv3 --pmc counter CSV into per-kernel ratios
(mfma_busy_frac, wait_frac, lds_conflict_per_idx, wave_cycles) — the
profiles/pmc_summary_*.json format."""
import csv
import json
import re
import sys
from collections import defaultdict


def main(path, out):
    per = defaultdict(lambda: defaultdict(float))
    with open(path) as f:
        rd = csv.DictReader(f)
        for row in rd:
            name = row.get("Kernel_Name") or row.get("Kernel Name") or ""
            ctr = row.get("Counter_Name") or row.get("Counter Name") or ""
            val = float(row.get("Counter_Value") or row.get("Counter Value")
                        or 0)
            n = name.replace("void ", "").replace("(anonymous namespace)::",
                                                  "")
            cut = len(n)
            for ch in "<(":
                i = n.find(ch)
                if i != -1:
                    cut = min(cut, i)
            short = n[:cut].split("::")[-1].strip()
            per[short][ctr] += val
    out_d = {}
    for k, c in sorted(per.items()):
        wc = c.get("SQ_WAVE_CYCLES", 0.0)
        d = {"wave_cycles": wc}
        if wc:
            d["mfma_busy_frac"] = round(
                c.get("SQ_VALU_MFMA_BUSY_CYCLES", 0.0) / wc, 3)
            d["wait_frac"] = round(c.get("SQ_WAIT_ANY", 0.0) / wc, 3)
        idx = c.get("SQ_LDS_IDX_ACTIVE", 0.0)
        if idx:
            d["lds_conflict_per_idx"] = round(
                c.get("SQ_LDS_BANK_CONFLICT", 0.0) / idx, 4)
        out_d[k] = d
    with open(out, "w") as f:
        json.dump(out_d, f, indent=1, sort_keys=True)
    print("wrote", out, f"({len(out_d)} kernels)")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
