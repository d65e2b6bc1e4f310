"""Extract per-kernel PMC sums from a rocprofv3 rocpd .db (counter-collection
run). Usage: python tools/pmc_extract.py <results.db>"""
import sqlite3
import sys


def main(path):
    con = sqlite3.connect(path)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(prefix):
        return [t for t in tabs if t.startswith(prefix)][0]

    kd, ks, pe, pi, st = (tab("rocpd_kernel_dispatch"),
                          tab("rocpd_info_kernel_symbol"),
                          tab("rocpd_pmc_event"), tab("rocpd_info_pmc"),
                          tab("rocpd_string"))
    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(DISTINCT kd.id),
           SUM(kd.end - kd.start)
    FROM {pe} pe
    JOIN {kd} kd ON pe.event_id = kd.event_id
    JOIN {ks} ks ON kd.kernel_id = ks.id
    JOIN {pi} pi ON pe.pmc_id = pi.id
    GROUP BY ks.display_name, pi.name
    """
    agg = {}
    for kname, cname, val, calls, ns in cur.execute(q):
        k = kname.split("(")[0][:48]
        agg.setdefault(k, {})[cname] = val
        agg[k]["_calls"] = calls
        agg[k]["_ms"] = ns / 1e6
    for k, d in sorted(agg.items(), key=lambda kv: -kv[1].get("_ms", 0)):
        if d.get("_ms", 0) < 1.0:
            continue
        print(f"== {k}  ({d['_calls']} calls, {d['_ms']:.2f} ms total)")
        busy = d.get("SQ_BUSY_CYCLES", 0)
        act = d.get("SQ_ACTIVE_INST_ANY", 0)
        for name, v in sorted(d.items()):
            if name.startswith("_"):
                continue
            extra = ""
            if busy and name in ("SQ_ACTIVE_INST_ANY", "SQ_WAIT_ANY",
                                 "SQ_WAIT_INST_ANY"):
                extra = f"  ({v / busy * 100:.0f}% of SQ_BUSY_CYCLES)"
            if act and name == "SQ_LDS_BANK_CONFLICT":
                extra = f"  ({v / act * 100:.0f}% of active-inst cycles)"
            print(f"   {name:24s} {v:.3e}{extra}")


if __name__ == "__main__":
    main(sys.argv[1])
