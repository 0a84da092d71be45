#!/usr/bin/env python3
"""Ad-hoc filter microbench over the bench.py data cache: stages the given
filter JSON on the same parts bench.py generates and reports per-launch
kernel time + achieved bandwidth.  For decomposing multi-leaf program cost
(e.g. the or8 config) leaf family by leaf family on a GPU box.

Usage: python tools/microbench.py '<filter_json>' [--rows N] [--steps K]
"""

import argparse
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("filter")
    ap.add_argument("--rows", type=int,
                    default=int(os.environ.get("VQL_BENCH_ROWS", 100_000_000)))
    ap.add_argument("--msg-len", type=int, default=256)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--data-dir", default=os.environ.get(
        "VQL_DATA_DIR", "/tmp/vql_bench_data"))
    args = ap.parse_args()

    from bench import prepare_parts
    from victorialogs_amd import Filter, Part, Stage

    nparts = max(1, min(64, (os.cpu_count() or 8),
                        max(8, args.rows // 16_000_000)))
    dirs = prepare_parts(os.path.join(args.data_dir, "rank0"), args.rows,
                         nparts, args.msg_len, seed_base=1)
    parts = [Part(d) for d in dirs]
    filt = Filter(args.filter)
    t0 = time.time()
    st = Stage(parts, filt, device=0)
    stage_s = time.time() - t0
    hits = 0
    for _ in range(args.warmup):
        hits = st.scan()
    kms = 0.0
    for _ in range(args.steps):
        st.scan()
        kms += st.last_kernel_ms
    kms /= args.steps
    out = {
        "filter": args.filter,
        "rows": st.rows,
        "live_rows": st.live_rows,
        "hits": hits,
        "staged_gb": st.staged_bytes / 1e9,
        "algo_gb": st.algo_bytes / 1e9,
        "kernel_ms": kms,
        "achieved_tb_s": st.algo_bytes / (kms / 1e3) / 1e12 if kms else 0,
        "rows_per_s": st.rows / (kms / 1e3) if kms else 0,
        "stage_s": stage_s,
    }
    print(json.dumps(out))
    st.close()
    filt.close()
    for p in parts:
        p.close()


if __name__ == "__main__":
    main()
