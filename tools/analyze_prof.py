"""Summarize a rocprofv3 kernel-trace DB: per-step busy fraction and
per-kernel totals for one steady-state step (delimited by sgd bursts)."""
import sqlite3, glob, re, collections, sys

import os
arg = sys.argv[1]
cands = [arg] if os.path.isfile(arg) else (
    sorted(glob.glob(arg + '/runc/*_results.db'))
    or sorted(glob.glob(arg + '/*_results.db'))
    or sorted(glob.glob(arg + '/**/*_results.db', recursive=True)))
db_path = cands[0]
db = sqlite3.connect(db_path)
cur = db.cursor()
t = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")
    if r[0].startswith('rocpd_kernel_dispatch')][0]
sfx = t[len('rocpd_kernel_dispatch_'):]
rows = list(cur.execute(
    f"""SELECT k.start, k.end, ks.display_name
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        ORDER BY k.start"""))
opt = [r for r in rows if 'sgd' in str(r[2]).lower() or 'adamw' in str(r[2]).lower()]
steps, cs = [], [opt[0]]
for r in opt[1:]:
    if r[0] - cs[-1][0] > 1e6:
        steps.append(cs)
        cs = [r]
    else:
        cs.append(r)
steps.append(cs)
bounds = [s[-1][1] for s in steps]
w0, w1 = bounds[-3], bounds[-2]
ks = [r for r in rows if r[0] >= w0 and r[1] <= w1]
merged, ce = 0, 0
for r in ks:
    if r[0] > ce:
        merged += r[1] - r[0]
        ce = r[1]
    elif r[1] > ce:
        merged += r[1] - ce
        ce = r[1]
print(f"one step: wall {(w1-w0)/1e6:.3f} ms busy {merged/1e6:.3f} "
      f"({100*merged/(w1-w0):.0f}%) n={len(ks)}")
agg = collections.Counter()
for r in ks:
    n = str(r[2])
    m = re.search(r'(\w+_kernel|Cijk_\w{1,12})', n)
    key = m.group(1) if m else n[:36]
    if key.endswith(('patch_kernel', 'smallk_kernel')):
        tpl = re.search(r'true, \d|false, \d', n)
        key += ' ' + (tpl.group(0) if tpl else '')
    agg[key] += r[1] - r[0]
for k, v in agg.most_common(20):
    print(f"  {v/1e3:8.1f} us  {k}")
