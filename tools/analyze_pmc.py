import sqlite3, glob, re, collections, sys
import os
arg = sys.argv[1]
cands = [arg] if os.path.isfile(arg) else (
    sorted(glob.glob(arg + '/runc/*_results.db'))
    or sorted(glob.glob(arg + '/*_results.db'))
    or sorted(glob.glob(arg + '/**/*_results.db', recursive=True)))
db_path = cands[0]
db = sqlite3.connect(db_path); cur = db.cursor()
t = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'") if r[0].startswith('rocpd_kernel_dispatch')][0]
sfx = t[len('rocpd_kernel_dispatch_'):]
pmc_names = {r[0]: r[1] for r in cur.execute(f"SELECT id, name FROM rocpd_info_pmc_{sfx}")}
data = collections.defaultdict(dict)
for name, pid, val in cur.execute(f"""SELECT ks.display_name, p.pmc_id, SUM(p.value)
  FROM rocpd_pmc_event_{sfx} p JOIN rocpd_kernel_dispatch_{sfx} k ON p.event_id = k.event_id
  JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id GROUP BY 1,2"""):
    m = re.search(r'(\w+_kernel)', str(name))
    key = m.group(1) if m else str(name)[:30]
    tpl = re.search(r'(true|false), \d', str(name))
    if key.endswith(('patch_kernel','smallk_kernel')) and tpl: key += ' ' + tpl.group(0)
    nm = pmc_names.get(pid, pid)
    data[key][nm] = data[key].get(nm, 0) + val
for k, d in sorted(data.items(), key=lambda kv: -kv[1].get('SQ_WAVE_CYCLES',0))[:9]:
    wc = d.get('SQ_WAVE_CYCLES', 1)
    print(f"{k}: WC {wc:.2e} WAIT {100*d.get('SQ_WAIT_ANY',0)/wc:.0f}% WAITINST {100*d.get('SQ_WAIT_INST_ANY',0)/wc:.0f}% VALU {d.get('SQ_INSTS_VALU',0):.2e} MFMA {d.get('SQ_INSTS_MFMA',0):.2e} LDSCONF {d.get('SQ_LDS_BANK_CONFLICT',0):.2e}")
