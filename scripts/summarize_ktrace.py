"""Summarize a rocprofv3 --kernel-trace results.db into per-kernel time."""
import glob
import json
import sqlite3
import sys


def main(d, out):
    agg = {}
    for dbp in glob.glob(f"{d}/**/*_results.db", recursive=True):
        db = sqlite3.connect(dbp)
        cur = db.cursor()
        tabs = [r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        sfx = None
        for t in tabs:
            if t.startswith("rocpd_kernel_dispatch_"):
                sfx = t[len("rocpd_kernel_dispatch_"):]
        if not sfx:
            continue
        q = f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name"""
        for name, calls, ms in cur.execute(q):
            key = name.split("(")[0][:80]
            e = agg.setdefault(key, {"calls": 0, "total_ms": 0.0})
            e["calls"] += calls
            e["total_ms"] += ms or 0
    top = dict(sorted(agg.items(), key=lambda kv: -kv[1]["total_ms"])[:25])
    for v in top.values():
        v["total_ms"] = round(v["total_ms"], 2)
    json.dump(top, open(out, "w"), indent=1)
    print("wrote", out)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
