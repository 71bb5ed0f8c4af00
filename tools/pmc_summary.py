"""Aggregate a rocprofv3 per-dispatch counter CSV into a small per-kernel summary."""
import csv
import sys
from collections import defaultdict


def main(path, out_path):
    agg = defaultdict(lambda: defaultdict(float))
    calls = defaultdict(int)
    with open(path) as f:
        r = csv.DictReader(f)
        kname = "Kernel_Name" if "Kernel_Name" in (r.fieldnames or []) else "Kernel Name"
        for row in r:
            name = row.get(kname, "?").split("(")[0][:100]
            cname = row.get("Counter_Name") or row.get("Counter Name")
            val = float(row.get("Counter_Value") or row.get("Counter Value") or 0)
            agg[name][cname] += val
            if cname and cname.startswith("SQ_WAVE"):
                calls[name] += 1
    counters = sorted({c for v in agg.values() for c in v})
    rows = sorted(agg.items(), key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))
    with open(out_path, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["kernel", "dispatches"] + counters)
        for name, vals in rows:
            w.writerow([name, calls[name]] + [f"{vals.get(c, 0):.0f}" for c in counters])
    print(f"wrote {out_path}: {len(rows)} kernels")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
