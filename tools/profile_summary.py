"""Summarize a rocprofv3 results DB into a markdown kernel-time table."""
import glob
import sqlite3
import sys


def summarize(path: str, out_path: str, top: int = 25) -> None:
    db = sqlite3.connect(path)
    cur = db.cursor()
    kd = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    sfx = kd.replace("rocpd_kernel_dispatch_", "")
    rows = list(
        cur.execute(
            f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6
            FROM rocpd_kernel_dispatch_{sfx} kd
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}"""
        )
    )
    total_ms, n = list(cur.execute(f"SELECT SUM(end-start)/1e6, COUNT(*) FROM rocpd_kernel_dispatch_{sfx}"))[0]
    with open(out_path, "w") as f:
        f.write(f"# rocprofv3 kernel summary: {path}\n\n")
        f.write(f"Total GPU kernel time: {total_ms:.2f} ms across {n} dispatches\n\n")
        f.write("| time (ms) | calls | kernel |\n|---|---|---|\n")
        for name, cnt, ms in rows:
            f.write(f"| {ms:.3f} | {cnt} | `{name[:110]}` |\n")
    print(f"wrote {out_path}")


if __name__ == "__main__":
    pattern = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof*/runc/*_results.db"
    out = sys.argv[2] if len(sys.argv) > 2 else "profiles/kernel_summary.md"
    dbs = sorted(glob.glob(pattern))
    assert dbs, f"no results DB matching {pattern}"
    summarize(dbs[-1], out)
