"""Parse rocprofv3 --pmc results.db -> per-kernel counter ratios.

Usage: python scripts/pmc2md.py <dir-with-results.db> [title]
Counters expected: SQ_INSTS_MFMA SQ_BUSY_CYCLES SQ_LDS_BANK_CONFLICT
SQ_WAIT_ANY (one --pmc run; kernel linkage via event_id).
"""
import collections
import glob
import sqlite3
import sys

d = sys.argv[1]
title = sys.argv[2] if len(sys.argv) > 2 else d
db = sqlite3.connect(glob.glob(f"{d}/*_results.db")[0])
suf = [r[0] for r in db.execute(
    "SELECT name FROM sqlite_master WHERE name LIKE "
    "'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch", "")
q = f"""SELECT s.display_name, c.name, SUM(p.value)
        FROM rocpd_pmc_event{suf} p
        JOIN rocpd_kernel_dispatch{suf} d ON d.event_id = p.event_id
        JOIN rocpd_info_kernel_symbol{suf} s ON s.id = d.kernel_id
        JOIN rocpd_info_pmc{suf} c ON c.id = p.pmc_id
        GROUP BY 1, 2"""
agg = collections.defaultdict(dict)
for name, cname, v in db.execute(q):
    agg[name[:62]][cname] = v
rows = []
for k, v in agg.items():
    busy = v.get("SQ_BUSY_CYCLES", 0) or 1
    rows.append((busy, k, v.get("SQ_INSTS_MFMA", 0) / busy,
                 v.get("SQ_LDS_BANK_CONFLICT", 0) / busy,
                 v.get("SQ_WAIT_ANY", 0) / busy))
rows.sort(reverse=True)
print(f"# {title}")
print("| kernel | mfma/busy | lds-conflict/busy | wait/busy |")
print("|---|---|---|---|")
for b, k, m, l, w in rows[:14]:
    print(f"| {k} | {m:.3f} | {l:.3f} | {w:.1f} |")
