#!/usr/bin/env python3
"""Summarise a rocprofv3 rocpd results database (kernel-trace) and/or a
PMC counter CSV into the committed profiles/ evidence.

  python tools_dev/rocpd_stats.py kernel <results.db> [flops_per_launch]
  python tools_dev/rocpd_stats.py fetch  <counter_collection.csv> <workload> \
      [--write profiles/hbm_traffic.json]

FETCH_SIZE correction (gfx950, MI355X_MICROARCH.md §HBM): rocprofv3
FETCH_SIZE reports half the bytes of wide coalesced streaming reads —
the recorded traffic is 2 x FETCH_SIZE(KB) x 1024 per launch.
"""
import csv
import glob
import json
import os
import sqlite3
import sys


def kernel_stats(db_path, flops=None):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    u = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch_"))
    u = u[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
      SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
             AVG(kd.end-kd.start)/1e6
      FROM rocpd_kernel_dispatch_{u} kd
      JOIN rocpd_info_kernel_symbol_{u} ks ON kd.kernel_id = ks.id
      GROUP BY ks.display_name ORDER BY 3 DESC""").fetchall()
    out = []
    for name, cnt, tot, avg in rows:
        line = f"{tot:10.3f} ms total | {cnt:3d} x {avg:9.3f} ms | {name[:72]}"
        if flops and "gemm_mfma" in name:
            tf = flops / (avg / 1e3) / 1e12
            line += f"  -> {tf:.2f} TFLOP/s"
        out.append(line)
    return "\n".join(out)


def fetch_traffic(csv_path, workload, write=None):
    total = 0.0
    n = 0
    for row in csv.DictReader(open(csv_path)):
        if "gemm_mfma" in row.get("Kernel_Name", "") and \
                row["Counter_Name"] == "FETCH_SIZE":
            total += float(row["Counter_Value"])
            n += 1
    per_launch_kb = total / max(n, 1)
    corrected = 2.0 * per_launch_kb * 1024     # gfx950 wide-read correction
    rec = {
        "reads_bytes_per_launch": corrected,
        "fetch_size_kb_reported": per_launch_kb,
        "correction": "x2 gfx950 wide coalesced read undercount",
        "launches": n,
        "source": os.path.basename(csv_path),
    }
    print(workload, json.dumps(rec))
    if write:
        data = {}
        if os.path.exists(write):
            data = json.load(open(write))
        data[workload] = rec
        json.dump(data, open(write, "w"), indent=1)
    return rec


def traffic_by_kernel(csv_path):
    """Per-kernel FETCH_SIZE/WRITE_SIZE totals and per-launch averages
    from a counter_collection.csv (one counter per pass; gfx950 x2
    read correction applied to FETCH_SIZE per MI355X_MICROARCH.md §HBM;
    WRITE_SIZE left raw — calibrate against fill_random, which writes a
    known byte count and reads ~nothing)."""
    agg = {}
    for row in csv.DictReader(open(csv_path)):
        kn = row.get("Kernel_Name", "")
        cn = row.get("Counter_Name", "")
        if cn not in ("FETCH_SIZE", "WRITE_SIZE"):
            continue
        key = kn.split("(")[0][:60]
        a = agg.setdefault(key, {"FETCH_SIZE": [0.0, 0],
                                 "WRITE_SIZE": [0.0, 0]})
        a[cn][0] += float(row["Counter_Value"])
        a[cn][1] += 1
    out = {}
    for k, a in sorted(agg.items()):
        rec = {}
        if a["FETCH_SIZE"][1]:
            per = a["FETCH_SIZE"][0] / a["FETCH_SIZE"][1]
            rec["reads_bytes_per_launch"] = 2.0 * per * 1024
            rec["fetch_size_kb_reported"] = per
            rec["launches"] = a["FETCH_SIZE"][1]
        if a["WRITE_SIZE"][1]:
            per = a["WRITE_SIZE"][0] / a["WRITE_SIZE"][1]
            rec["write_size_kb_reported"] = per
            rec["writes_bytes_raw"] = per * 1024
            rec["launches_w"] = a["WRITE_SIZE"][1]
        out[k] = rec
        print(k, json.dumps(rec))
    return out


if __name__ == "__main__":
    if sys.argv[1] == "kernel":
        fl = float(sys.argv[3]) if len(sys.argv) > 3 else None
        print(kernel_stats(sys.argv[2], fl))
    elif sys.argv[1] == "fetch":
        w = None
        if "--write" in sys.argv:
            w = sys.argv[sys.argv.index("--write") + 1]
        fetch_traffic(sys.argv[2], sys.argv[3], w)
    elif sys.argv[1] == "traffic":
        traffic_by_kernel(sys.argv[2])
