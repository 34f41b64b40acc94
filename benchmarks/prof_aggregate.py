"""Aggregate a rocprofv3 SQLite results DB into a per-kernel stats table
(top-N by total GPU time) — runs on the GPU box so only the small text
summary rides back through gpurun_out."""

import glob
import sqlite3
import sys


def main(db_glob: str, top: int = 40, tail_frac: float = 0.0):
    """tail_frac > 0 keeps only dispatches that START in the last fraction
    of the trace timeline — strips the MIOpen find phase / warmup so the
    table shows the steady-state round."""
    paths = sorted(glob.glob(db_glob))
    if not paths:
        print(f"no DB matches {db_glob}")
        return
    for path in paths:
        db = sqlite3.connect(path)
        cur = db.cursor()
        tables = [r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        sym_t = [t for t in tables if t.startswith("rocpd_info_kernel_symbol")]
        disp_t = [t for t in tables if t.startswith("rocpd_kernel_dispatch")]
        if not sym_t or not disp_t:
            continue
        syms = {r[0]: r[1] for r in cur.execute(
            f"SELECT id, display_name FROM {sym_t[0]}")}
        lo, hi = next(cur.execute(
            f"SELECT MIN(start), MAX(end) FROM {disp_t[0]}"))
        cutoff = lo + (hi - lo) * tail_frac
        agg = {}
        for kid, start, end in cur.execute(
                f"SELECT kernel_id, start, end FROM {disp_t[0]}"):
            if start < cutoff:
                continue
            name = syms.get(kid, str(kid))
            ent = agg.setdefault(name, [0, 0.0])
            ent[0] += 1
            ent[1] += (end - start) / 1e3   # ns -> us
        total = sum(v[1] for v in agg.values())
        print(f"== {path}: {len(agg)} kernels, total {total/1e3:.2f} ms ==")
        print(f"{'kernel':<72s} {'calls':>7s} {'total_ms':>9s} "
              f"{'avg_us':>8s} {'%':>5s}")
        for name, (n, us) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:top]:
            short = name.split("(")[0][:72]
            print(f"{short:<72s} {n:>7d} {us/1e3:>9.2f} {us/n:>8.1f} "
                  f"{100*us/total:>5.1f}")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "/tmp/prof2/**/*_results.db",
         int(sys.argv[2]) if len(sys.argv) > 2 else 40,
         float(sys.argv[3]) if len(sys.argv) > 3 else 0.0)
