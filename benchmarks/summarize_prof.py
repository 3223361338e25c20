"""Summarize a rocprofv3 rocpd .db into a per-kernel hotlist (markdown).

Usage: python benchmarks/summarize_prof.py <results.db> [out.md]
"""

import sqlite3
import sys


def summarize(dbf, out=None):
    db = sqlite3.connect(dbf)
    cur = db.cursor()
    tables = [
        r[0]
        for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table' "
            "AND name LIKE 'rocpd_kernel_dispatch%'"
        )
    ]
    sfx = tables[0].replace("rocpd_kernel_dispatch_", "")
    rows = cur.execute(
        f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
               AVG(kd.end-kd.start)/1e3,
               MAX(ks.arch_vgpr_count), MAX(ks.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30
        """
    ).fetchall()
    total = sum(r[2] for r in rows)
    lines = [
        "| total ms | % | calls | avg us | VGPR | LDS B | kernel |",
        "|---:|---:|---:|---:|---:|---:|---|",
    ]
    for name, cnt, ms, us, vgpr, lds in rows:
        short = str(name).split("(")[0][:72]
        lines.append(
            f"| {ms:.2f} | {100*ms/total:.1f} | {cnt} | {us:.1f} "
            f"| {vgpr} | {lds} | `{short}` |"
        )
    lines.append(f"\ntotal GPU kernel time: {total:.1f} ms")
    text = "\n".join(lines)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")
    print(text)


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
