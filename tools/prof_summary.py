"""Summarize a rocprofv3 rocpd .db into compact text (run on the GPU box;
only the summary travels back, not the multi-MB database)."""

import sqlite3
import sys


def main(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    names = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")]
    sfx = names[0].replace("rocpd_kernel_dispatch_", "")
    print(f"== kernel totals ({path}) ==")
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
           AVG(k.end-k.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY SUM(k.end-k.start) DESC LIMIT 12
    """
    tot = 0.0
    for name, cnt, ms, us in cur.execute(q):
        tot += ms
        print(f"{ms:9.1f} ms {cnt:6d} x {us:8.1f} us  {name[:66]}")
    print(f"total kernel {tot:.1f} ms")
    rows = list(cur.execute(
        f"SELECT start, end FROM rocpd_kernel_dispatch_{sfx} ORDER BY start"))
    idle = 0
    last = None
    for s, e in rows:
        if last is not None and s > last:
            idle += s - last
        last = max(last or e, e)
    print(f"span {(rows[-1][1]-rows[0][0])/1e9:.2f} s, idle {idle/1e6:.0f} ms, "
          f"{len(rows)} dispatches")
    # per-shape breakdown of our kernels
    q2 = f"""
    SELECT ks.display_name, k.grid_size_x/k.workgroup_size_x, k.grid_size_y,
           k.grid_size_z, COUNT(*), AVG(k.end-k.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id
    WHERE ks.display_name LIKE 'k_%' GROUP BY 1,2,3,4 ORDER BY 5 DESC LIMIT 10
    """
    print("== shapes (kernel, blocks_x, gy, gz, count, avg us) ==")
    for r in cur.execute(q2):
        print("  ", r)
    # PMC counters if present
    try:
        pmc_info = {r[0]: r[1] for r in cur.execute(
            f"SELECT id, name FROM rocpd_info_pmc_{sfx}")}
        if pmc_info:
            print("== PMC sums per kernel ==")
            q3 = f"""
            SELECT ks.display_name, p.pmc_id, SUM(p.value), COUNT(*)
            FROM rocpd_pmc_event_{sfx} p
            JOIN rocpd_kernel_dispatch_{sfx} k ON p.event_id = k.event_id
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
            WHERE ks.display_name LIKE 'k_%'
            GROUP BY 1, 2
            """
            for name, pid, val, cnt in cur.execute(q3):
                print(f"  {name[:40]:40s} {pmc_info.get(pid, pid):24s} "
                      f"{val:.3e} over {cnt}")
    except sqlite3.OperationalError as e:
        print("pmc read failed:", e)


if __name__ == "__main__":
    main(sys.argv[1])
