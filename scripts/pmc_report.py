#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc rocpd output: per-kernel mean counter value
per dispatch.  Usage: python scripts/pmc_report.py 'gpurun_out/pmc_x/**/*results.db'
"""
import glob
import sqlite3
import sys
from collections import defaultdict


def main():
    pat = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/pmc/**/*results.db"
    dbs = glob.glob(pat, recursive=True)
    if not dbs:
        print(f"no db matches {pat}")
        return 1
    for db in dbs:
        con = sqlite3.connect(db)
        tabs = [r[0] for r in con.execute(
            "select name from sqlite_master where type='table'")]
        uuid = None
        for t in tabs:
            if t.startswith("rocpd_pmc_event_"):
                uuid = t[len("rocpd_pmc_event_"):]
        if uuid is None:
            continue
        q = f"""
        select s.string, pi.name, sum(pe.value), count(distinct kd.id)
        from rocpd_pmc_event_{uuid} pe
        join rocpd_kernel_dispatch_{uuid} kd on pe.event_id = kd.event_id
        join rocpd_info_kernel_symbol_{uuid} ks on kd.kernel_id = ks.id
        join rocpd_string_{uuid} s on ks.display_name_id = s.id
        join rocpd_info_pmc_{uuid} pi on pe.pmc_id = pi.id
        group by 1, 2
        """
        rows = None
        try:
            rows = con.execute(q).fetchall()
        except sqlite3.OperationalError:
            # schema variant: kernel symbol name may be direct
            q2 = f"""
            select ks.display_name, pi.name, sum(pe.value),
                   count(distinct kd.id)
            from rocpd_pmc_event_{uuid} pe
            join rocpd_kernel_dispatch_{uuid} kd on pe.event_id = kd.event_id
            join rocpd_info_kernel_symbol_{uuid} ks on kd.kernel_id = ks.id
            join rocpd_info_pmc_{uuid} pi on pe.pmc_id = pi.id
            group by 1, 2
            """
            rows = con.execute(q2).fetchall()
        per = defaultdict(dict)
        for kname, counter, total, ndisp in rows:
            per[kname[:70]][counter] = (total, ndisp)
        print(f"== {db}")
        for kname, counters in sorted(per.items()):
            print(f"  {kname}")
            for counter, (total, nd) in sorted(counters.items()):
                print(f"    {counter:<28s} total {total:.3e}  "
                      f"/dispatch {total / max(1, nd):.3e}  ({nd} disp)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
