"""Join rocprofv3 --stats kernel durations with --pmc FETCH_SIZE/WRITE_SIZE
counters into a per-kernel achieved-HBM-bandwidth table (roofline vs the
~6.3 TB/s achievable on MI355X; 8 TB/s peak).

Usage (on the GPU box):
  rocprofv3 --stats -d out/stats -- python bench.py ...
  rocprofv3 --pmc FETCH_SIZE -d out/fetch -- python bench.py ...
  python tools/roofline.py out/stats out/fetch [out/write] > roofline.md

FETCH_SIZE counts HBM read traffic in KB per dispatch (TCC block); the
256 MiB Infinity Cache absorbs re-reads, so numbers are true HBM traffic,
not L2/L3 hits.
"""
import csv
import glob
import os
import sys
from collections import defaultdict


def _find_csv(d, needle, prefer=None):
    cands = sorted(glob.glob(os.path.join(d, "**", "*.csv"),
                             recursive=True))
    if prefer:
        cands = [p for p in cands if prefer in os.path.basename(p)] + \
            [p for p in cands if prefer not in os.path.basename(p)]
    for p in cands:
        with open(p) as f:
            head = f.readline()
        if needle in head:
            return p
    return None


def load_stats(d):
    """kernel -> (total_ns, calls) from *_kernel_stats.csv or kernel
    trace."""
    out = {}
    p = _find_csv(d, "TotalDurationNs", prefer="kernel_stats") \
        or _find_csv(d, "DurationNs", prefer="kernel_trace")
    if p is None:
        return out
    with open(p) as f:
        rd = csv.DictReader(f)
        for row in rd:
            name = row.get("Name") or row.get("Kernel_Name") or ""
            name = name.strip('"')
            if "TotalDurationNs" in row:
                out[name] = (float(row["TotalDurationNs"]),
                             int(row.get("Calls", 1)))
            else:
                t, c = out.get(name, (0.0, 0))
                dur = float(row.get("DurationNs", 0) or
                            (float(row.get("End_Timestamp", 0)) -
                             float(row.get("Start_Timestamp", 0))))
                out[name] = (t + dur, c + 1)
    return out


def load_counter(d, counter):
    """kernel -> total counter value (KB for FETCH_SIZE/WRITE_SIZE)."""
    out = defaultdict(float)
    p = _find_csv(d, "Counter_Name") or _find_csv(d, counter)
    if p is None:
        return out
    with open(p) as f:
        rd = csv.DictReader(f)
        for row in rd:
            name = (row.get("Kernel_Name") or row.get("Name") or
                    "").strip('"')
            if "Counter_Name" in row:
                if row["Counter_Name"].strip() != counter:
                    continue
                out[name] += float(row.get("Counter_Value", 0))
            elif counter in row:
                out[name] += float(row[counter])
    return out


def main():
    stats_dir, fetch_dir = sys.argv[1], sys.argv[2]
    write_dir = sys.argv[3] if len(sys.argv) > 3 else None
    stats = load_stats(stats_dir)
    fetch = load_counter(fetch_dir, "FETCH_SIZE")
    write = load_counter(write_dir, "WRITE_SIZE") if write_dir else {}
    rows = []
    for name, (ns, calls) in stats.items():
        fkb = fetch.get(name, 0.0)
        wkb = write.get(name, 0.0)
        if ns <= 0:
            continue
        gbps = (fkb + wkb) * 1024 / ns  # KB/ns -> GB/s
        rows.append((ns, name, calls, fkb / 1048576, wkb / 1048576, gbps))
    rows.sort(reverse=True)
    total_ns = sum(r[0] for r in rows)
    print("| kernel | calls | time ms | %time | read GB | write GB | "
          "achieved GB/s | % of 6.3 TB/s |")
    print("|---|---|---|---|---|---|---|---|")
    for ns, name, calls, fgb, wgb, gbps in rows[:20]:
        short = name.split("(")[0][:60]
        print(f"| {short} | {calls} | {ns / 1e6:.2f} | "
              f"{100 * ns / total_ns:.1f}% | {fgb:.3f} | {wgb:.3f} | "
              f"{gbps:.0f} | {gbps / 63:.1f}% |")


if __name__ == "__main__":
    main()
