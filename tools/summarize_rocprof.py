"""Summarize a rocprofv3 sqlite results DB into a small kernel-stats table.

Usage: python tools/summarize_rocprof.py results.db out.txt [--tail-frac 0.3]
Writes top kernels by total time over the trailing window (steady state).
"""
import sqlite3
import sys


def main():
    db_path, out_path = sys.argv[1], sys.argv[2]
    tail_frac = float(sys.argv[3]) if len(sys.argv) > 3 else 0.3
    db = sqlite3.connect(db_path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    tmin, tmax = db.execute(f"SELECT MIN(start), MAX(end) FROM {kd}").fetchone()
    cut = tmax - int((tmax - tmin) * tail_frac)
    rows = db.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM {kd} k JOIN {ks} s ON k.kernel_id = s.id
        WHERE k.start > {cut}
        GROUP BY s.display_name ORDER BY 3 DESC LIMIT 40""").fetchall()
    tot, n = db.execute(
        f"SELECT SUM(end-start)/1e6, COUNT(*) FROM {kd} WHERE start > {cut}"
    ).fetchone()
    with open(out_path, "w") as f:
        f.write(f"# rocprofv3 kernel stats, trailing {tail_frac:.0%} window "
                f"of {(tmax-tmin)/1e9:.2f}s\n")
        f.write(f"# total kernel time {tot:.1f} ms across {n} dispatches\n")
        f.write(f"{'total_ms':>10} {'calls':>7} {'avg_us':>9}  kernel\n")
        for name, calls, ms, avg in rows:
            f.write(f"{ms:10.3f} {calls:7d} {avg:9.2f}  {str(name)[:130]}\n")
    print(f"wrote {out_path}")


if __name__ == "__main__":
    main()
