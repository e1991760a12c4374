"""Parse a rocprofv3 rocpd SQLite db into a per-kernel stats table.
Usage: python profparse.py <results.db> (defensive about table names)."""
import sqlite3
import sys

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
disp = next(t for t in tables if "kernel_dispatch" in t)
cols = [c[1] for c in cur.execute(f"PRAGMA table_info({disp})")]
# find the kernel-name source: either a name column or a symbol table
kinfo = [t for t in tables if "kernel" in t and t != disp]
rows = cur.execute(f"SELECT * FROM {disp}").fetchall()
ci = {c: i for i, c in enumerate(cols)}


def col(r, *names):
    for n in names:
        if n in ci:
            return r[ci[n]]
    return None


# kernel symbol metadata
sym = {}
for t in kinfo:
    tc = [c[1] for c in cur.execute(f"PRAGMA table_info({t})")]
    if any("name" in c for c in tc) and any("id" in c for c in tc):
        idc = "id" if "id" in tc else next(c for c in tc if c.endswith("id"))
        namec = next(c for c in tc if "display_name" in c or c == "name"
                     or "kernel_name" in c)
        extra = [c for c in tc if any(k in c for k in
                 ("vgpr", "sgpr", "scratch", "private", "group"))]
        q = f"SELECT {idc}, {namec}" + (
            ", " + ", ".join(extra) if extra else "") + f" FROM {t}"
        for r in cur.execute(q):
            sym[r[0]] = (r[1], r[2:] if extra else ())
        if sym:
            extra_names = extra
            break

agg = {}
for r in rows:
    kid = col(r, "kernel_id", "kernel_symbol_id", "symbol_id")
    st = col(r, "start", "start_timestamp", "begin_ns")
    en = col(r, "end", "end_timestamp", "end_ns")
    name, extra = sym.get(kid, (str(kid), ()))
    if "(" in str(name):
        name = str(name).split("(")[0]
    a = agg.setdefault(name, [0, 0.0, extra])
    a[0] += 1
    a[1] += (en - st) / 1e6

print(f"{'kernel':<30} {'n':>5} {'total_ms':>10} {'avg_ms':>9}  extras")
for name, (n, tot, extra) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
    print(f"{name:<30} {n:>5} {tot:>10.2f} {tot/n:>9.3f}  {list(extra)}")
try:
    print("extra columns:", extra_names)
except NameError:
    pass
