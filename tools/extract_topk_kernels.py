"""Dump the rocprofv3 'top' / 'top_kernels' view from a results db."""
import sqlite3, sys

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
for view in ("top", "top_kernels", "kernels", "busy"):
    try:
        cur.execute(f"SELECT * FROM {view} LIMIT 30")
    except Exception as e:
        print(f"-- {view}: {e}", file=sys.stderr)
        continue
    cols = [d[0] for d in cur.description]
    print(f"== {view} == {cols}")
    for row in cur.fetchall():
        print(" | ".join(str(x)[:110] for x in row))
    print()
