"""Aggregate rocprofv3 PMC SQLite output per kernel name."""
import glob
import sqlite3
import sys
from collections import defaultdict

db = glob.glob(sys.argv[1] + "/**/*.db", recursive=True)
assert db, "no results db"
con = sqlite3.connect(db[0])
cur = con.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
ctab = [t for t in tabs if "counter" in t.lower()]
ktab = [t for t in tabs if "kernel_dispatch" in t.lower()]
stab = [t for t in tabs if "kernel_symbol" in t.lower()]
print("# tables:", ctab, ktab, stab, file=sys.stderr)
sym = {}
for t in stab:
    for r in cur.execute(f"SELECT id, display_name FROM {t}"):
        sym[r[0]] = r[1]
disp = {}
for t in ktab:
    cols = [c[1] for c in cur.execute(f"PRAGMA table_info({t})")]
    kcol = "kernel_id" if "kernel_id" in cols else [
        c for c in cols if "symbol" in c or "kernel" in c][0]
    for r in cur.execute(f"SELECT id, {kcol}, (end-start) FROM {t}"):
        disp[r[0]] = (sym.get(r[1], str(r[1])), r[2])
agg = defaultdict(lambda: defaultdict(float))
cnt = defaultdict(int)
dur = defaultdict(float)
for t in ctab:
    cols = [c[1] for c in cur.execute(f"PRAGMA table_info({t})")]
    if "counter_name" in cols:
        q = f"SELECT dispatch_id, counter_name, value FROM {t}"
    elif "name" in cols and "value" in cols and "dispatch_id" in cols:
        q = f"SELECT dispatch_id, name, value FROM {t}"
    else:
        print("# skip", t, cols, file=sys.stderr)
        continue
    for did, cn, v in cur.execute(q):
        name = disp.get(did, ("?", 0))[0]
        agg[name][cn] += v
for did, (name, d) in disp.items():
    cnt[name] += 1
    dur[name] += d
for name in sorted(agg, key=lambda n: -dur[n]):
    short = name.split("(")[0][:60]
    print(f"== {short}  n={cnt[name]}  total_ms={dur[name]/1e6:.2f}")
    tot = agg[name].get("SQ_WAVE_CYCLES", 0) or 1
    for cn, v in sorted(agg[name].items()):
        print(f"   {cn:,s}: {v:.3e}  ({v/tot*100:.1f}% of WAVE_CYCLES)"
              if cn.startswith("SQ_") else f"   {cn}: {v:.3e}")
