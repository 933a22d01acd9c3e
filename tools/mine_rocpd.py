"""Mine a rocprofv3 --output-format rocpd SQLite DB into a per-kernel stats
table (name, calls, total ms, mean us, % of kernel time), schema-agnostically:
finds the kernel-dispatch table by column inspection so it works across
rocprofv3 minor versions.

Usage: python tools/mine_rocpd.py <db-path-or-dir> [top_n]
"""

import glob
import os
import sqlite3
import sys


def find_db(path):
    if os.path.isfile(path):
        return path
    hits = sorted(glob.glob(os.path.join(path, "**", "*.db"), recursive=True))
    if not hits:
        raise SystemExit(f"no .db under {path}")
    return hits[0]


def main():
    db = find_db(sys.argv[1])
    top_n = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]

    # Prefer the ready-made `kernels` view (rocprofv3 ships one that joins the
    # dispatch rows to the symbol names); fall back to anything kernel-shaped.
    cands = sorted((t for t in tables if "kernel" in t.lower()),
                   key=lambda t: (t.lower() != "kernels", len(t)))
    for t in cands:
        cols = [c[1].lower() for c in cur.execute(f"PRAGMA table_info('{t}')")]
        name_col = next((c for c in cols if c in ("name", "display_name", "kernel_name")), None)
        start = next((c for c in cols if c in ("start", "start_timestamp", "begin")), None)
        end = next((c for c in cols if c in ("end", "end_timestamp")), None)
        dur = next((c for c in cols if "duration" in c), None)
        if not name_col or not (dur or (start and end)):
            continue
        sample = cur.execute(f'SELECT "{name_col}" FROM `{t}` LIMIT 1').fetchone()
        if not sample or not isinstance(sample[0], str) or not sample[0]:
            continue  # empty/interned names: this is not the joined view
        expr = f'"{dur}"' if dur else f'("{end}" - "{start}")'
        rows = cur.execute(
            f'SELECT "{name_col}", COUNT(*), SUM({expr}), AVG({expr}) '
            f"FROM `{t}` GROUP BY \"{name_col}\" ORDER BY SUM({expr}) DESC"
        ).fetchall()
        if rows and rows[0][2]:
            report(rows, top_n, t)
            return
    # Last resort: join the raw dispatch rows to the symbol table ourselves.
    if "rocpd_kernel_dispatch" in tables and "rocpd_info_kernel_symbol" in tables:
        rows = cur.execute(
            'SELECT S.display_name, COUNT(*), SUM(K."end" - K."start"), '
            'AVG(K."end" - K."start") FROM rocpd_kernel_dispatch K '
            "JOIN rocpd_info_kernel_symbol S ON S.id = K.kernel_id "
            'GROUP BY S.display_name ORDER BY SUM(K."end" - K."start") DESC'
        ).fetchall()
        if rows and rows[0][2]:
            report(rows, top_n, "dispatch x symbol join")
            return
    raise SystemExit(f"no kernel table with names found; candidates: {cands}")


def string_table(cur, tables):
    """Map id -> string from whichever interning table this DB carries."""
    for t in tables:
        if "string" not in t.lower() and "symbol" not in t.lower() and "info" not in t.lower():
            continue
        cols = [c[1].lower() for c in cur.execute(f"PRAGMA table_info('{t}')")]
        idc = next((c for c in cols if c in ("id", "string_id", "rowid")), None)
        sc = next((c for c in cols if c in ("string", "value", "name", "display_name",
                                            "kernel_name", "formatted_kernel_name")), None)
        if idc and sc:
            lut = {r[0]: r[1] for r in cur.execute(f"SELECT {idc}, {sc} FROM '{t}'")
                   if isinstance(r[1], str)}
            if lut:
                return lut
    return {}


def short(name, width=86):
    name = name.split("(")[0].strip()
    for p in ("void ", "at::native::", "dinov3::"):
        name = name.replace(p, "")
    return name[:width]


def report(rows, top_n, table):
    total = sum(r[2] for r in rows)
    print(f"# kernel table: {table}; {sum(r[1] for r in rows)} dispatches, "
          f"{total / 1e6:.3f} ms total kernel time")
    print(f"{'kernel':<88} {'calls':>6} {'total_ms':>9} {'mean_us':>8} {'%':>6}")
    for name, calls, tot, avg in rows[:top_n]:
        print(f"{short(name):<88} {calls:>6} {tot / 1e6:>9.3f} {avg / 1e3:>8.1f} "
              f"{100.0 * tot / total:>6.2f}")


if __name__ == "__main__":
    main()
