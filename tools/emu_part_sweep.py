#!/usr/bin/env python3
"""Randomized-PART parity sweep: random column shapes/types/contents are
written through the part writer, then random filter trees scan them through
the EMULATED product pipeline (VQL_LIB=emu) against the oracle,
bit-identical bitmaps required.  Complements tools/emu_sweep.py (fixed
parts, random trees) by randomizing the staging/codec inputs too.

Usage: VQL_LIB=tools/host_emu/libvlogsql_emu.so \
           python tools/emu_part_sweep.py [start_seed] [end_seed]
"""

import json
import os
import random
import shutil
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

WORDS = ["error", "warn", "info", "the", "stream", "foo", "bar", "x1",
         "готово", "значение", "", "a_b", "0", "-7", "3.5", "ip", "row"]
DICTS = [["debug", "info", "warn"], ["a", "b"], ["x", "y", "z", "w"],
         ["один", "два"]]


def rand_column(rng, rows):
    kind = rng.randrange(8)
    name = f"c{rng.randrange(6)}"
    if kind == 0:    # random words (string/dict/const depending on shape)
        vals = [" ".join(rng.choice(WORDS)
                         for _ in range(rng.randrange(0, 4)))
                for _ in range(rows)]
    elif kind == 1:  # dict-shaped
        d = rng.choice(DICTS)
        vals = [rng.choice(d) for _ in range(rows)]
    elif kind == 2:  # const
        vals = [rng.choice(WORDS)] * rows
    elif kind == 3:  # uints of random width
        hi = rng.choice([9, 250, 60000, 4_000_000, 10**12])
        vals = [str(rng.randrange(hi)) for _ in range(rows)]
    elif kind == 4:  # signed ints
        vals = [str(rng.randrange(-1000, 1000)) for _ in range(rows)]
    elif kind == 5:  # floats
        vals = [f"{rng.randrange(-50, 50)}.{rng.randrange(100)}"
                for _ in range(rows)]
    elif kind == 6:  # ipv4
        vals = [f"10.{rng.randrange(4)}.{rng.randrange(4)}.{rng.randrange(9)}"
                for _ in range(rows)]
    else:            # iso8601
        vals = [f"2024-0{rng.randrange(1, 9)}-1{rng.randrange(9)}"
                f"T0{rng.randrange(9)}:00:0{rng.randrange(9)}.000Z"
                for _ in range(rows)]
    return {"name": name, "values": vals}


def rand_part(rng, tmpdir, idx):
    blocks = []
    base = 1700000000000000000 + rng.randrange(10**6) * 1000
    for b in range(rng.randrange(1, 4)):
        rows = rng.randrange(1, 500)
        ncols = rng.randrange(1, 5)
        cols = {}
        for _ in range(ncols):
            c = rand_column(rng, rows)
            cols[c["name"]] = c  # unique names within the block
        if rng.random() < 0.7:
            cols["_msg"] = {"name": "_msg",
                            "values": rand_column(rng, rows)["values"]}
            cols["_msg"]["name"] = "_msg"
        blocks.append({
            "stream": rng.randrange(2),
            "timestamps": sorted(base + rng.randrange(10**9)
                                 for _ in range(rows)),
            "columns": [{"name": (n if n != "_msg" else "_msg"),
                         "values": c["values"]} for n, c in cols.items()],
        })
    blocks.sort(key=lambda blk: (blk["stream"], blk["timestamps"][0]))
    d = os.path.join(tmpdir, f"p{idx}")
    from victorialogs_amd import write_custom_part
    write_custom_part(d, {"blocks": blocks})
    fields = sorted({c["name"] for blk in blocks for c in blk["columns"]})
    return d, fields


def main():
    start = int(sys.argv[1]) if len(sys.argv) > 1 else 0
    end = int(sys.argv[2]) if len(sys.argv) > 2 else 50_000
    assert "libvlogsql_emu" in os.environ.get("VQL_LIB", "")

    from tests.test_gpu_fuzz import random_tree
    from tests.test_gpu_parity import assert_parity

    phrases = WORDS + ["10.1", "2024", "00", "deb"]
    regexes = ["err(or|)", "in.o", "\\d+\\.\\d", "^готово", "a_b|x",
               "\\bwarn\\b", "(?i)INFO", "10\\.[0-3]", "w?arn$"]
    t0 = time.time()
    done = 0
    tmp = tempfile.mkdtemp(prefix="part_sweep_")
    try:
        for seed in range(start, end):
            rng = random.Random(seed ^ 0x5eed)
            part, fields = rand_part(rng, tmp, seed % 64)
            fnum = [f for f in fields if f.startswith("c")] or ["c0"]
            for _ in range(4):
                tree = random_tree(rng, phrases, fields + ["absent"], fnum,
                                   regexes, depth=2)
                try:
                    assert_parity(part, json.dumps(tree))
                except RuntimeError as e:
                    if "regex" in str(e):
                        continue
                    raise
                done += 1
            shutil.rmtree(part, ignore_errors=True)
            if (seed - start) % 100 == 99:
                dt = time.time() - t0
                print(f"seed {seed}: {done} scans OK ({done / dt:.1f}/s)",
                      flush=True)
    finally:
        shutil.rmtree(tmp, ignore_errors=True)
    print(f"PART SWEEP CLEAN: {done} scans, seeds [{start}, {end})")


if __name__ == "__main__":
    main()
