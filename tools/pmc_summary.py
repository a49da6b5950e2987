import sqlite3, sys, collections
db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
cols = [c[1] for c in cur.execute("PRAGMA table_info(counters_collection)")]
agg = collections.defaultdict(lambda: collections.defaultdict(float))
n = collections.defaultdict(lambda: collections.Counter())
for row in cur.execute("SELECT * FROM counters_collection"):
    d = dict(zip(cols, row))
    kn = d.get('kernel_name') or ''
    for tag in ("ce_linear_bwd", "ce_linear_lse", "ce_linear_fwd", "attn_bwd", "ln_bwd"):
        if tag in kn:
            agg[tag][d['counter_name']] += float(d['value'])
            n[tag][d['counter_name']] += 1
for tag, cs in agg.items():
    print(f"== {tag} ==")
    for k, v in sorted(cs.items()):
        print(f"  {k}: {v:.4e} (n={n[tag][k]}, per-dispatch {v/max(1,n[tag][k]):.3e})")
