"""Randomized differential parity: seeded random filter trees evaluated by
the HIP kernels and the CPU oracle must produce bit-identical bitmaps."""

import json
import random

import pytest

from tests.test_gpu_parity import assert_parity

pytestmark = pytest.mark.gpu


GEN_PHRASES = [
    "message", "stream", "the stream 1 and", "ip=", "uuid", "u64",
    "message for the stream", "pad", "zzz_absent", "host_0", "error", "info",
    "some value", "0", "1", "255", "",
]
GEN_FIELDS_STR = ["_msg", "var_0", "host", "run_id", "missing"]
GEN_FIELDS_NUM = ["u8_0", "u16_0", "u32_0", "u64_0", "i64_0", "float_0"]
GEN_REGEXES = [
    "stream (0|1)", "uuid=[0-9a-f]", "ip=1.*uuid", "message.+", "err(or|)x|info",
    "pad", ".*u64.*", "host_.+",
    # NFA class
    "stream \\d+ and", "u64=\\d?\\d", "ip=[0-9]+\\.[0-9]+", "pa?d=",
    "[^;]*uuid", "w\\w+ker",
]

TYPED_PHRASES = [
    "level=error", "took", "row", "warn", "ERROR", "13", "два", "value",
    "1.5", "200ms", "absent", "",
]
TYPED_FIELDS_STR = ["_msg", "lvl", "uni", "mix", "constcol", "nope"]
TYPED_FIELDS_NUM = ["u8", "u16", "u32", "u64", "i64", "f64", "ip", "iso"]
TYPED_REGEXES = [
    "level=(error|warn)", "took 1.*ms", "row [0-9]", "два|foo", "19(2|3)",
    "-3", "2024-01", "fixed.+",
    # NFA class
    "took \\d+ms", "l[a-z]+e\\d", "ро?w", "[0-9]+\\.[0-9]+", "\\dms|GiB",
    "2024-\\d+-0[1-5]T",
]


def random_tree(rng, phrases, fields_str, fields_num, regexes, depth):
    leaf_kinds = ["phrase", "phrase", "exact", "regexp", "time", "range",
                  "prefix", "exact_prefix", "sequence"]
    kind = rng.choice(
        leaf_kinds + ["and", "or", "not"] if depth > 0 else leaf_kinds)
    if kind == "phrase":
        return {"type": "phrase",
                "field": rng.choice(fields_str + fields_num),
                "phrase": rng.choice(phrases)}
    if kind == "exact":
        return {"type": "exact",
                "field": rng.choice(fields_str + fields_num),
                "value": rng.choice(phrases)}
    if kind in ("prefix", "exact_prefix"):
        return {"type": kind,
                "field": rng.choice(fields_str + fields_num),
                "prefix": rng.choice(phrases)}
    if kind == "sequence":
        return {"type": "sequence",
                "field": rng.choice(fields_str + fields_num),
                "phrases": [rng.choice(phrases)
                            for _ in range(rng.randrange(1, 4))]}
    if kind == "regexp":
        return {"type": "regexp", "field": rng.choice(fields_str),
                "re": rng.choice(regexes)}
    if kind == "time":
        base = 1700000000000000000
        a = base + rng.randrange(-10**9, 5 * 10**13)
        b = a + rng.randrange(-10**9, 10**13)
        return {"type": "time", "min": a, "max": b}
    if kind == "range":
        a = rng.uniform(-1e9, 1e9) * rng.choice([1, 1e-6, 1e9])
        b = a + rng.uniform(-10, 1e9)
        return {"type": "range", "field": rng.choice(fields_num + fields_str),
                "min": a, "max": b}
    if kind == "not":
        return {"type": "not",
                "filter": random_tree(rng, phrases, fields_str, fields_num,
                                      regexes, depth - 1)}
    n = rng.randrange(2, 4)
    return {"type": kind, "filters": [
        random_tree(rng, phrases, fields_str, fields_num, regexes, depth - 1)
        for _ in range(n)]}


@pytest.mark.parametrize("seed", range(8))
def test_fuzz_generated_part(gen_part, seed):
    rng = random.Random(1000 + seed)
    for _ in range(10):
        tree = random_tree(rng, GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM,
                           GEN_REGEXES, depth=3)
        assert_parity(gen_part, json.dumps(tree))


@pytest.mark.parametrize("seed", range(8))
def test_fuzz_typed_part(typed_part, seed):
    rng = random.Random(2000 + seed)
    for _ in range(10):
        tree = random_tree(rng, TYPED_PHRASES, TYPED_FIELDS_STR,
                           TYPED_FIELDS_NUM, TYPED_REGEXES, depth=3)
        assert_parity(typed_part, json.dumps(tree))
