"""Summarize a rocprofv3 rocpd SQLite database: per-kernel count / total /
mean duration, sorted by total. Usage: python rocpd_summary.py <db-or-dir>"""

import glob
import os
import sqlite3
import sys


def find_db(path):
    if os.path.isfile(path):
        return path
    cands = sorted(
        glob.glob(os.path.join(path, "**", "*.db"), recursive=True),
        key=os.path.getmtime,
    )
    if not cands:
        raise SystemExit(f"no .db under {path}")
    return cands[-1]


def main(path):
    db = find_db(path)
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')"
    )]
    print(f"# {db}")
    kd = next((t for t in tables if "kernel_dispatch" in t), None)
    if kd is None:
        print("tables:", tables)
        return
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    print("dispatch table:", kd, cols)
    name_col = next(
        (c for c in cols
         if "name" in c.lower() and not c.lower().endswith("_id")),
        None,
    )
    if name_col is None:
        # names live in a kernel-info table; find the join key
        ki = next((t for t in tables if "kernel" in t and "info" in t), None)
        kcols = [r[1] for r in cur.execute(f"PRAGMA table_info({ki})")] if ki else []
        print("dispatch cols:", cols)
        print("info table:", ki, kcols)
        # common rocpd layout: kernel_dispatch.kernel_id -> kernel info id
        join_left = next((c for c in cols if c in ("kernel_id", "kernel_info_id")), None)
        kname = next((c for c in kcols if "name" in c.lower()), None)
        kid = next((c for c in kcols if c in ("id", "kernel_id")), None)
        q = f"""
            SELECT ki.{kname}, COUNT(*), SUM(kd.end - kd.start),
                   AVG(kd.end - kd.start)
            FROM {kd} kd JOIN {ki} ki ON kd.{join_left} = ki.{kid}
            GROUP BY ki.{kname} ORDER BY 3 DESC LIMIT 25
        """
    else:
        q = f"""
            SELECT {name_col}, COUNT(*), SUM(end - start), AVG(end - start)
            FROM {kd} GROUP BY {name_col} ORDER BY 3 DESC LIMIT 25
        """
    total = 0
    rows = list(cur.execute(q))
    if rows and isinstance(rows[0][0], int):
        # names are ids into the string table
        st = next((t for t in tables if "string" in t), None)
        scols = [r[1] for r in cur.execute(f"PRAGMA table_info({st})")]
        sid = next((c for c in scols if "id" in c.lower()), scols[0])
        sval = next(
            (c for c in scols if c.lower() in ("string", "value", "str")),
            scols[-1],
        )
        lut = dict(cur.execute(f"SELECT {sid}, {sval} FROM {st}"))
        if rows and rows[0][0] not in lut:
            print("string table:", st, scols,
                  list(cur.execute(f"SELECT * FROM {st} LIMIT 3")))
        rows = [(str(lut.get(r[0], r[0])), *r[1:]) for r in rows]
    total = sum(r[2] for r in rows if r[2])
    print(f"{'kernel':70s} {'count':>7s} {'total_ms':>10s} {'avg_us':>9s} {'%':>5s}")
    for name, cnt, tot, avg in rows:
        short = (name or "?")[:70]
        print(f"{short:70s} {cnt:7d} {tot/1e6:10.3f} {avg/1e3:9.2f} "
              f"{100.0*tot/total:5.1f}")


def pmc(path):
    """Aggregate PMC counter sums per kernel (rocprofv3 --pmc runs)."""
    db = find_db(path)
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')"
    )]
    pe = next((t for t in tables if "pmc_event" in t), None)
    pi = next((t for t in tables if "info_pmc" in t), None)
    kd = next((t for t in tables if "kernel_dispatch" in t), None)
    ki = next((t for t in tables if "kernel" in t and "info" in t), None)
    if not (pe and pi and kd and ki):
        print("pmc tables missing:", tables)
        return
    for t in (pe, pi):
        cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
        print(t, cols)
    q = f"""
        SELECT ki.kernel_name, ip.name, SUM(p.value), COUNT(*)
        FROM {pe} p
        JOIN {kd} kd ON p.event_id = kd.event_id
        JOIN {ki} ki ON kd.kernel_id = ki.id
        JOIN {pi} ip ON p.pmc_id = ip.id
        GROUP BY ki.kernel_name, ip.name ORDER BY 3 DESC LIMIT 40
    """
    try:
        for kname, cname, total, cnt in cur.execute(q):
            print(f"{str(kname)[:56]:56s} {str(cname):34s} "
                  f"{total:16.0f} n={cnt}")
    except sqlite3.OperationalError as e:
        print("query failed:", e)


if __name__ == "__main__":
    if len(sys.argv) > 2 and sys.argv[1] == "--pmc":
        pmc(sys.argv[2])
    else:
        main(sys.argv[1] if len(sys.argv) > 1 else ".")
