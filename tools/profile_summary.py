"""Summarize a rocprofv3 rocpd SQLite DB into a markdown kernel table.

python tools/profile_summary.py gpurun_out/prof/xxx_results.db > profiles/foo.md
"""
import sqlite3
import sys


def main(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
      SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
             AVG(k.end-k.start)/1e3, ks.arch_vgpr_count, ks.sgpr_count,
             MAX(k.group_segment_size)
      FROM {kd} k JOIN {ks} ks ON k.kernel_id = ks.id
      GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30""").fetchall()
    total = sum(r[2] for r in rows)
    print(f"| kernel | calls | total ms | avg us | % | vgpr | lds B |")
    print(f"|---|---|---|---|---|---|---|")
    for r in rows:
        name = r[0].replace("void ", "").replace("(anonymous namespace)::", "")
        name = name.split("(")[0].split("<")[0][:70] or name[:70]
        print(f"| {name} | {r[1]} | {r[2]:.2f} | {r[3]:.1f} | "
              f"{100*r[2]/total:.1f} | {r[4]} | {r[6]} |")
    print(f"\nTotal GPU kernel time: {total:.2f} ms")


if __name__ == "__main__":
    main(sys.argv[1])
