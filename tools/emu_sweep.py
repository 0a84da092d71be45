#!/usr/bin/env python3
"""Long-running randomized parity sweep: seeded random filter trees through
the EMULATED product pipeline (real staging + real per-row device code,
VQL_LIB=tools/host_emu/libvlogsql_emu.so) against the CPU oracle,
bit-identical bitmaps required.  CPU-only hardening between GPU rounds —
round 1 ran ~131k trees this way; run with a seed range to extend.

Usage: VQL_LIB=tools/host_emu/libvlogsql_emu.so \
           python tools/emu_sweep.py [start_seed] [end_seed]
"""

import os
import random
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    start = int(sys.argv[1]) if len(sys.argv) > 1 else 10_000
    end = int(sys.argv[2]) if len(sys.argv) > 2 else 20_000
    assert "libvlogsql_emu" in os.environ.get("VQL_LIB", ""), \
        "run with VQL_LIB pointing at the emu build"

    import tests.conftest as cf
    from tests.test_gpu_fuzz import (GEN_FIELDS_NUM, GEN_FIELDS_STR,
                                     GEN_PHRASES, GEN_REGEXES, TYPED_FIELDS_NUM,
                                     TYPED_FIELDS_STR, TYPED_PHRASES,
                                     TYPED_REGEXES, random_tree)
    from tests.test_gpu_parity import assert_parity

    class TF:
        def mktemp(self, x):
            import pathlib
            return pathlib.Path(tempfile.mkdtemp())

    gen = cf.gen_part.__wrapped__(TF())
    typed = cf.typed_part.__wrapped__(TF())

    import json
    t0 = time.time()
    done = 0
    for seed in range(start, end):
        rng = random.Random(seed)
        for part_dir, ph, fs, fn, res in (
                (gen, GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM, GEN_REGEXES),
                (typed, TYPED_PHRASES, TYPED_FIELDS_STR, TYPED_FIELDS_NUM,
                 TYPED_REGEXES)):
            tree = random_tree(rng, ph, fs, fn, res, depth=3)
            fjson = json.dumps(tree)
            try:
                assert_parity(part_dir, fjson)
            except RuntimeError as e:
                if "regex" in str(e):
                    continue  # loud unsupported-construct reject: fine
                raise
            done += 1
        if (seed - start) % 200 == 199:
            dt = time.time() - t0
            print(f"seed {seed}: {done} trees OK ({done / dt:.1f}/s)",
                  flush=True)
    print(f"SWEEP CLEAN: {done} trees, seeds [{start}, {end})")


if __name__ == "__main__":
    main()
