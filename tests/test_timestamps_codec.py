"""Timestamps codec round-trip fuzz (the VictoriaMetrics int64 array codec:
Const / DeltaConst / NearestDelta +-zstd, vendor/.../encoding/encoding.go):
random timestamp shapes written through the part writer must scan back
exactly under point time filters."""

import json
import random

import pytest

from victorialogs_amd import OracleScanner, write_custom_part

SHAPES = [
    ("const", lambda rng, n: [1700000000000000000] * n),
    ("delta_const", lambda rng, n: [1700000000000000000 + i * 12345
                                    for i in range(n)]),
    ("random_small", lambda rng, n: sorted(
        1700000000000000000 + rng.randrange(0, 10**6) for _ in range(n))),
    ("random_wide", lambda rng, n: sorted(
        rng.randrange(-2**62, 2**62) for _ in range(n))),
    ("negative", lambda rng, n: sorted(
        -10**18 + rng.randrange(0, 10**9) for _ in range(n))),
    ("jumps", lambda rng, n: sorted(
        rng.choice([0, 10**18, -10**18, 1, -1]) + rng.randrange(0, 100)
        for _ in range(n))),
]


@pytest.mark.parametrize("name,gen", SHAPES, ids=[s[0] for s in SHAPES])
def test_timestamps_roundtrip(tmp_path, name, gen):
    rng = random.Random(sum(name.encode()))  # deterministic across runs
    n = 257
    ts = gen(rng, n)
    d = str(tmp_path / name)
    write_custom_part(d, {"blocks": [{
        "stream": 0,
        "timestamps": ts,
        "columns": [{"name": "v", "values": [str(i) for i in range(n)]}],
    }]})
    sc = OracleScanner(d)
    try:
        # point filters on a sample of timestamps must hit exactly the rows
        # holding that timestamp (filter_time is inclusive on both ends)
        for idx in [0, 1, n // 2, n - 2, n - 1] + [rng.randrange(n)
                                                   for _ in range(10)]:
            t = ts[idx]
            want = sum(1 for x in ts if x == t)
            hits, _ = sc.scan(json.dumps({"type": "time", "min": t, "max": t}))
            assert hits == want, f"{name}: ts={t} hits={hits} want={want}"
        # full-range filter hits everything
        hits, _ = sc.scan(json.dumps(
            {"type": "time", "min": min(ts), "max": max(ts)}))
        assert hits == n
    finally:
        sc.close()
