#!/usr/bin/env python3
"""Summarize a rocprofv3 results db (kernel dispatch + memcpy stats) into a
text table for profiles/."""

import glob
import sqlite3
import sys


def main(path_glob, out_path):
    dbs = glob.glob(path_glob)
    if not dbs:
        print(f"no db at {path_glob}", file=sys.stderr)
        return 1
    db = sqlite3.connect(dbs[0])
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    sfx = t[len("rocpd_kernel_dispatch_"):]
    lines = [f"# rocprofv3 kernel summary ({dbs[0]})", ""]
    lines.append(f"{'kernel':<52} {'grid':>9} {'cnt':>5} {'total_ms':>9} "
                 f"{'avg_us':>8} {'max_us':>8}")
    q = f"""
SELECT ks.display_name, kd.grid_size_x, COUNT(*),
       SUM(kd.end-kd.start)/1e6, AVG(kd.end-kd.start)/1e3, MAX(kd.end-kd.start)/1e3
FROM rocpd_kernel_dispatch_{sfx} kd
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
GROUP BY ks.display_name, kd.grid_size_x ORDER BY 4 DESC LIMIT 20"""
    for r in cur.execute(q):
        lines.append(f"{r[0][:52]:<52} {r[1]:>9} {r[2]:>5} {r[3]:>9.3f} "
                     f"{r[4]:>8.1f} {r[5]:>8.1f}")
    lines.append("")
    lines.append(f"{'memcpy kind':<52} {'cnt':>5} {'total_ms':>9} {'avg_us':>8} {'MB':>10}")
    q2 = f"""
SELECT s.string, COUNT(*), SUM(mc.end-mc.start)/1e6, AVG(mc.end-mc.start)/1e3,
       SUM(mc.size)/1e6
FROM rocpd_memory_copy_{sfx} mc JOIN rocpd_string_{sfx} s ON mc.name_id=s.id
GROUP BY s.string"""
    for r in cur.execute(q2):
        lines.append(f"{r[0][:52]:<52} {r[1]:>5} {r[2]:>9.3f} {r[3]:>8.1f} {r[4]:>10.1f}")
    text = "\n".join(lines) + "\n"
    with open(out_path, "w") as f:
        f.write(text)
    print(text)
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1], sys.argv[2]))
