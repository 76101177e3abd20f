"""Render a rocprofv3 results.db into a markdown kernel-time table."""

import glob
import sqlite3
import sys


def summarize(db_path: str, out_path: str, title: str,
              n_iters: int = None) -> None:
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")
        if r[0].startswith("rocpd_kernel_dispatch")][0]
    sfx = t[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms,
               AVG(k.end-k.start)/1e3 avg_us,
               MAX(ks.arch_vgpr_count) vgpr, MAX(k.group_segment_size) lds
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY ms DESC LIMIT 30""").fetchall()
    total, cnt = cur.execute(
        f"SELECT SUM(end-start)/1e6, COUNT(*) "
        f"FROM rocpd_kernel_dispatch_{sfx}").fetchone()
    with open(out_path, "w") as f:
        f.write(f"# {title}\n\n")
        f.write(f"Total kernel time: {total:.1f} ms over {cnt} dispatches")
        if n_iters:
            f.write(f" ({n_iters} iterations -> "
                    f"{total / n_iters:.2f} ms kernel time/iter)")
        f.write("\n\n")
        f.write("| kernel | calls | total ms | avg us | VGPR | LDS B |\n")
        f.write("|---|---|---|---|---|---|\n")
        for name, n, ms, avg, vgpr, lds in rows:
            short = name[:200].replace("|", "\\|")
            f.write(f"| `{short}` | {n} | {ms:.2f} | {avg:.1f} "
                    f"| {vgpr or ''} | {lds or ''} |\n")
    print(f"wrote {out_path}")


if __name__ == "__main__":
    db = sys.argv[1] if len(sys.argv) > 1 else \
        glob.glob("gpurun_out/prof*/**_results.db")[-1]
    out = sys.argv[2] if len(sys.argv) > 2 else "profiles/summary.md"
    title = sys.argv[3] if len(sys.argv) > 3 else "Kernel profile"
    iters = int(sys.argv[4]) if len(sys.argv) > 4 else None
    summarize(db, out, title, iters)
