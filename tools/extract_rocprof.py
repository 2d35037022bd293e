"""Extract compact kernel/PMC summaries from rocprofv3 sqlite DBs."""
import collections
import glob
import sqlite3
import sys


def kernel_stats(root, out):
    with open(out, "w") as fh:
        for db in glob.glob(f"{root}/**/*.db", recursive=True):
            conn = sqlite3.connect(db)
            cur = conn.cursor()
            tabs = [r[0] for r in cur.execute(
                "SELECT name FROM sqlite_master WHERE type='table'")]
            kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch_")]
            if not kd:
                continue
            suf = kd[0][len("rocpd_kernel_dispatch_"):]
            q = (f"SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, "
                 f"AVG(k.end-k.start)/1e3, MAX(ks.arch_vgpr_count), "
                 f"MAX(ks.accum_vgpr_count) "
                 f"FROM rocpd_kernel_dispatch_{suf} k "
                 f"JOIN rocpd_info_kernel_symbol_{suf} ks ON k.kernel_id=ks.id "
                 f"GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 45")
            hdr = "total_ms calls avg_us vgpr agpr kernel"
            fh.write(hdr + "\n")
            for name, n, tot, avg, vg, ag in cur.execute(q):
                fh.write(f"{tot:10.3f} {n:6d} {avg:9.1f} {vg or 0:5d} "
                         f"{ag or 0:5d}  {name[:100]}\n")


def pmc_stats(root, out):
    with open(out, "w") as fh:
        for db in glob.glob(f"{root}/**/*.db", recursive=True):
            conn = sqlite3.connect(db)
            cur = conn.cursor()
            tabs = [r[0] for r in cur.execute(
                "SELECT name FROM sqlite_master WHERE type='table'")]
            kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch_")]
            if not kd:
                continue
            suf = kd[0][len("rocpd_kernel_dispatch_"):]
            try:
                q = (f"SELECT ks.display_name, pi.name, SUM(pe.value) "
                     f"FROM rocpd_pmc_event_{suf} pe "
                     f"JOIN rocpd_info_pmc_{suf} pi ON pe.pmc_id=pi.id "
                     f"JOIN rocpd_kernel_dispatch_{suf} k "
                     f"  ON pe.event_id=k.event_id "
                     f"JOIN rocpd_info_kernel_symbol_{suf} ks "
                     f"  ON k.kernel_id=ks.id "
                     f"GROUP BY ks.display_name, pi.name")
                agg = collections.defaultdict(dict)
                for kname, cname, v in cur.execute(q):
                    agg[kname][cname] = v
                cols = sorted({c for d in agg.values() for c in d})
                fh.write("kernel | " + " | ".join(cols) + "\n")
                for kname, d in sorted(
                        agg.items(),
                        key=lambda kv: -kv[1].get("SQ_INSTS_MFMA", 0))[:35]:
                    fh.write(kname[:80] + " | " +
                             " | ".join(f"{d.get(c, 0):.4g}" for c in cols) +
                             "\n")
            except Exception as ex:  # fall back: dump schema for debugging
                fh.write(f"pmc extraction failed: {ex}\ntables: {tabs}\n")
                for t in tabs:
                    if "pmc" in t:
                        fh.write(f"-- {t}\n")
                        for r in cur.execute(f"PRAGMA table_info({t})"):
                            fh.write(f"   {r}\n")


if __name__ == "__main__":
    mode, root, out = sys.argv[1], sys.argv[2], sys.argv[3]
    if mode == "kernels":
        kernel_stats(root, out)
    else:
        pmc_stats(root, out)
