#!/usr/bin/env python3
"""Post-process rocprofv3 SQLite outputs from gpurun_out/ into committed
summaries under profiles/. Run in the dev container after a profiling run."""
import glob
import os
import sqlite3
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def table(conn, base):
    names = [r[0] for r in conn.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    for n in names:
        if n.startswith(base):
            return n
    raise KeyError(base)


def kernel_stats(db):
    c = sqlite3.connect(db)
    kd = table(c, "rocpd_kernel_dispatch")
    ks = table(c, "rocpd_info_kernel_symbol")
    rows = c.execute(f"""
      SELECT k.display_name, COUNT(*), AVG(d.end-d.start)/1e6,
             SUM(d.end-d.start)/1e6, MIN(d.end-d.start)/1e6,
             MAX(d.end-d.start)/1e6, MAX(k.arch_vgpr_count),
             MAX(k.sgpr_count), MAX(d.private_segment_size),
             MAX(d.group_segment_size)
      FROM {kd} d JOIN {ks} k ON d.kernel_id = k.id
      GROUP BY k.display_name ORDER BY 4 DESC""").fetchall()
    out = ["kernel | dispatches | avg_ms | total_ms | min_ms | max_ms | "
           "vgpr | sgpr | scratch_B | lds_B",
           "-" * 100]
    for r in rows:
        out.append(f"{r[0][:60]:62s} {r[1]:4d} {r[2]:9.3f} {r[3]:9.3f} "
                   f"{r[4]:8.3f} {r[5]:8.3f} {r[6]:4d} {r[7]:4d} {r[8]:6d} {r[9]:6d}")
    return "\n".join(out)


def pmc_stats(db):
    c = sqlite3.connect(db)
    try:
        kd = table(c, "rocpd_kernel_dispatch")
        ks = table(c, "rocpd_info_kernel_symbol")
        ev = table(c, "rocpd_pmc_event")
        pi = table(c, "rocpd_info_pmc")
        rows = c.execute(f"""
          SELECT k.display_name, p.name, COUNT(DISTINCT d.id), SUM(e.value)
          FROM {ev} e
          JOIN {kd} d ON e.event_id = d.event_id
          JOIN {ks} k ON d.kernel_id = k.id
          JOIN {pi} p ON e.pmc_id = p.id
          GROUP BY k.display_name, p.name ORDER BY 4 DESC""").fetchall()
    except Exception as exn:  # schema variant: counters CSV merged instead
        return f"(pmc schema mismatch: {exn})"
    out = ["kernel | counter | dispatches | total_value", "-" * 80]
    for r in rows:
        out.append(f"{r[0][:55]:57s} {r[1]:12s} {r[2]:4d} {r[3]:.6g}")
    return "\n".join(out)


def main():
    tag = sys.argv[1] if len(sys.argv) > 1 else "r01"
    os.makedirs(os.path.join(REPO, "profiles"), exist_ok=True)
    jobs = []
    for d in sorted(glob.glob(os.path.join(REPO, "gpurun_out", "*"))):
        if not os.path.isdir(d):
            continue
        dbs = glob.glob(os.path.join(d, "**", "*results.db"), recursive=True)
        csvs = glob.glob(os.path.join(d, "**", "*counter_collection.csv"),
                         recursive=True)
        if dbs or csvs:
            jobs.append((os.path.basename(d), dbs, csvs))
    for name, dbs, csvs in jobs:
        out = [f"# rocprofv3 summary: {name} (round {tag})", ""]
        for db in dbs:
            out.append(f"## {os.path.relpath(db, REPO)}")
            try:
                out.append(kernel_stats(db))
            except Exception as e:
                out.append(f"(kernel stats unavailable: {e})")
            out.append("")
            if "pmc" in name:
                out.append(pmc_stats(db))
                out.append("")
        for csv in csvs:
            import csv as csvmod
            import collections
            agg = collections.defaultdict(lambda: collections.defaultdict(float))
            for row in csvmod.DictReader(open(csv)):
                k = row["Kernel_Name"].split("(")[0].replace("void ", "")
                agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
            out.append(f"## {os.path.relpath(csv, REPO)} (counter totals)")
            for k, d2 in agg.items():
                for cname, v in sorted(d2.items()):
                    out.append(f"{k[:55]:57s} {cname:24s} {v:.6g}")
            out.append("")
        path = os.path.join(REPO, "profiles", f"{tag}_{name}.txt")
        with open(path, "w") as f:
            f.write("\n".join(out) + "\n")
        print("wrote", path)


if __name__ == "__main__":
    main()
