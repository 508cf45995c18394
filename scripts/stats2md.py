"""Parse rocprofv3 results.db -> per-kernel stats markdown."""
import glob, json, os, sqlite3, sys

hdr = {"alexnet": "AlexNet bs=256", "cifar10_quick": "CIFAR10-quick bs=100",
       "googlenet": "GoogLeNet bs=128", "lrcn": "LRCN bs=64 T=21"}
for m, title in hdr.items():
    dbs = glob.glob(f"gpurun_out/prof_{m}/*_results.db")
    if not dbs:
        print("no db for", m)
        continue
    db = sqlite3.connect(dbs[0])
    suf = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE "
        "'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch", "")
    q = f"""SELECT s.display_name, COUNT(*), SUM(d.end - d.start)
            FROM rocpd_kernel_dispatch{suf} d
            JOIN rocpd_info_kernel_symbol{suf} s ON s.id = d.kernel_id
            GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"""
    rows = list(db.execute(q))
    tot = sum(r[2] for r in rows)
    js = {}
    for line in open(f"gpurun_out/prof_{m}.log"):
        line = line.strip()
        if line.startswith('{"metric"'):
            js = json.loads(line)
    out = [f"# {title} bf16 1xMI355X — round-2 final kernel profile "
           f"(hipGraph stepping)",
           f"{js.get('ms_per_step', '?')} ms/step under rocprofv3 "
           f"--kernel-trace (adds overhead; unprofiled numbers in "
           f"BASELINE.md), value={js.get('value', '?')} "
           f"{js.get('unit', '')}.",
           "", "| kernel | calls | total us | avg us | % |",
           "|---|---|---|---|---|"]
    for name, calls, ns in rows[:22]:
        us = ns / 1e3
        out.append(f"| {name[:80]} | {calls} | {us:.0f} | "
                   f"{us / max(1, calls):.1f} | {100 * ns / tot:.1f} |")
    open(f"gpurun_out/{m}_profile.md", "w").write("\n".join(out) + "\n")
    print("wrote", m, f"total_gpu_us={tot/1e3:.0f}")
