"""Summarize a rocprofv3 --pmc results.db into per-kernel counter sums."""
import glob
import json
import sqlite3
import sys


def main(d, out):
    dbs = glob.glob(f"{d}/**/*_results.db", recursive=True)
    agg = {}
    for dbp in dbs:
        db = sqlite3.connect(dbp)
        cur = db.cursor()
        tabs = [r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        sfx = None
        for t in tabs:
            if t.startswith("rocpd_pmc_event_"):
                sfx = t[len("rocpd_pmc_event_"):]
        if not sfx:
            continue
        cols = [r[1] for r in cur.execute(
            f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
        print("pmc cols:", cols)
        q = f"""
        SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
        JOIN rocpd_kernel_dispatch_{sfx} k ON pe.event_id = k.event_id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name, pi.name
        """
        try:
            for name, ctr, total, n in cur.execute(q):
                key = name.split("(")[0][:70]
                agg.setdefault(key, {"dispatches": 0})[ctr] = \
                    agg.get(key, {}).get(ctr, 0) + (total or 0)
                agg[key]["dispatches"] = max(agg[key]["dispatches"], n)
        except Exception as e:
            print("query failed:", e)
    top = dict(sorted(agg.items(),
                      key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))[:20])
    json.dump(top, open(out, "w"), indent=1)
    print("wrote", out, len(top), "kernels")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
