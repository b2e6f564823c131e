"""On-box analysis: group kernels in the steady-state tail of a rocprof DB."""
import glob
import sqlite3
import sys

path = sorted(glob.glob(sys.argv[1]))[-1]
out_path = sys.argv[2]
tail_ms = float(sys.argv[3]) if len(sys.argv) > 3 else 300.0
db = sqlite3.connect(path)
cur = db.cursor()
kd = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0]
sfx = kd.replace("rocpd_kernel_dispatch_", "")
t0, t1 = list(cur.execute(f"SELECT MIN(start), MAX(end) FROM {kd}"))[0]
lo = t1 - tail_ms * 1e6
rows = list(cur.execute(
    f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6
    FROM {kd} kd JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    WHERE kd.start > {lo}
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30"""))
total_ms, n = list(cur.execute(f"SELECT SUM(end-start)/1e6, COUNT(*) FROM {kd} WHERE start > {lo}"))[0]
span_ms = (t1 - lo) / 1e6
with open(out_path, "w") as f:
    f.write(f"# steady-state kernel summary (last {tail_ms:.0f} ms of timeline)\n\n")
    f.write(f"busy {total_ms:.1f} ms / span {span_ms:.1f} ms ({100*total_ms/span_ms:.0f}% busy), {n} dispatches\n\n")
    f.write("| time (ms) | calls | kernel |\n|---|---|---|\n")
    for name, cnt, ms in rows:
        f.write(f"| {ms:.3f} | {cnt} | `{name[:100]}` |\n")
print(open(out_path).read()[:3000])
