#!/usr/bin/env python3
"""Extracts per-launch PMC averages for scan_program_kernel from rocprofv3
rocpd SQLite databases and writes profiles/r02/pmc_traffic.json + a
human-readable table (the committed evidence bench.py cites for
roofline.traffic).

Usage: python tools/pmc_extract.py <dir-prefix> <out-json>
  e.g. python tools/pmc_extract.py gpurun_out/pmcf profiles/r02/pmc_traffic.json

Correction (MI355X_MICROARCH.md, HBM section): gfx950 FETCH_SIZE reports
half the bytes of wide coalesced streaming reads -> corrected bytes =
raw KB x 1024 x 2.  WRITE_SIZE is uncalibrated on gfx950; raw KB x 1024.
"""

import glob
import json
import sqlite3
import sys

# algorithmic bytes per launch (DESIGN.md data-layout table x config rows),
# cross-checked against vql_stage_algo_bytes in the bench lines
ALGO = {"phrase_regex": 29.65e9, "phrase": 26.81e9, "dict_time": 56.7e6}


def kernel_avg(db_glob, kernel="scan_program_kernel"):
    for db in glob.glob(db_glob):
        c = sqlite3.connect(db)
        tabs = [r[0] for r in c.execute(
            "select name from sqlite_master where type='table'")]
        pmc = [t for t in tabs if t.startswith("rocpd_pmc_event")]
        if not pmc:
            continue
        u = pmc[0].replace("rocpd_pmc_event_", "")
        q = (f"select ks.display_name, count(*), avg(pe.value) "
             f"from rocpd_pmc_event_{u} pe "
             f"join rocpd_kernel_dispatch_{u} kd on kd.event_id = pe.event_id "
             f"join rocpd_info_kernel_symbol_{u} ks on ks.id = kd.kernel_id "
             f"group by 1")
        for name, n, avg in c.execute(q):
            if kernel in name:
                return n, avg
    return None, None


def main():
    prefix, out_path = sys.argv[1], sys.argv[2]
    res = {}
    for wl, algo in ALGO.items():
        nf, fetch_kb = kernel_avg(f"{prefix}_{wl}_FETCH_SIZE/runc/*_results.db")
        nw, write_kb = kernel_avg(f"{prefix}_{wl}_WRITE_SIZE/runc/*_results.db")
        if fetch_kb is None or write_kb is None:
            print(f"{wl}: missing PMC data", file=sys.stderr)
            continue
        fetch_b = fetch_kb * 1024 * 2
        write_b = write_kb * 1024
        res[wl] = {
            "fetch_bytes_per_launch": fetch_b,
            "write_bytes_per_launch": write_b,
            "fetch_raw_kb": fetch_kb,
            "write_raw_kb": write_kb,
            "launches_sampled": nf,
            "algorithmic_bytes_per_launch": algo,
            "fetch_over_algorithmic": fetch_b / algo,
            "correction": "FETCH_SIZE KB x1024 x2 (gfx950 reports half of "
                          "wide coalesced reads, MI355X_MICROARCH.md HBM "
                          "section); WRITE_SIZE KB x1024, uncalibrated per "
                          "the same guide",
            "source": "profiles/r02/pmc_summary.txt",
        }
        print(f"{wl:14s} FETCH {fetch_kb:14.1f} KB x2 = {fetch_b / 1e9:8.3f} GB"
              f"  vs algorithmic {algo / 1e9:8.3f} GB "
              f"(ratio {fetch_b / algo:.3f})   WRITE {write_b / 1e6:9.2f} MB"
              f"  (n={nf})")
    with open(out_path, "w") as f:
        json.dump(res, f, indent=1)


if __name__ == "__main__":
    main()
