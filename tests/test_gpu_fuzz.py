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
    "^message", "worker 0;$", "^message.*uuid", "s{1,2}am", "e{2}",
    "^[a-z ]{10,30}for", "o{1,3}",
]

TYPED_PHRASES = [
    "level=error", "took", "row", "warn", "ERROR", "13", "два", "value",
    "1.5", "200ms", "absent", "", "ДВА", "Foo", "LEVEL",
]
TYPED_FIELDS_STR = ["_msg", "lvl", "uni", "mix", "constcol", "nope", "lvl2",
                    "ipstr", "mix2"]
TYPED_FIELDS_NUM = ["u8", "u16", "u32", "u64", "i64", "f64", "ip", "iso",
                    "u8b", "i64b", "f64b", "ipb", "isob"]
TYPED_REGEXES = [
    "level=(error|warn)", "took 1.*ms", "row [0-9]", "два|foo", "19(2|3)",
    "-3", "2024-01", "fixed.+",
    # NFA class
    "took \\d+ms", "l[a-z]+e\\d", "ро?w", "[0-9]+\\.[0-9]+", "\\dms|GiB",
    "2024-\\d+-0[1-5]T",
    "^log", "ms$", "^other stream row \\d{1,2};", "l{2}", "[0-9]{3}",
    "^(debug|info|warn|error)$",
]


def random_tree(rng, phrases, fields_str, fields_num, regexes, depth):
    leaf_kinds = ["phrase", "phrase", "exact", "regexp", "time", "range",
                  "prefix", "exact_prefix", "sequence",
                  "in", "contains_any", "contains_all", "string_range",
                  "ipv4_range", "len_range", "day_range", "week_range",
                  "value_type", "any_case_phrase", "any_case_prefix",
                  "eq_field", "le_field"]
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
    if kind in ("eq_field", "le_field"):
        d = {"type": kind,
             "field": rng.choice(fields_str + fields_num),
             "other_field": rng.choice(fields_str + fields_num)}
        if kind == "le_field" and rng.random() < 0.4:
            d["exclude_equal"] = True
        return d
    if kind in ("any_case_phrase",):
        return {"type": kind,
                "field": rng.choice(fields_str + fields_num),
                "phrase": rng.choice(phrases + [p.upper() for p in phrases])}
    if kind == "any_case_prefix":
        return {"type": kind,
                "field": rng.choice(fields_str + fields_num),
                "prefix": rng.choice(phrases + [p.title() for p in phrases])}
    if kind in ("prefix", "exact_prefix"):
        return {"type": kind,
                "field": rng.choice(fields_str + fields_num),
                "prefix": rng.choice(phrases)}
    if kind in ("in", "contains_any", "contains_all"):
        return {"type": kind,
                "field": rng.choice(fields_str + fields_num),
                "values": [rng.choice(phrases)
                           for _ in range(rng.randrange(0, 4))]}
    if kind == "string_range":
        a, b = rng.choice(phrases), rng.choice(phrases)
        return {"type": "string_range",
                "field": rng.choice(fields_str + fields_num),
                "min": a, "max": b}
    if kind == "ipv4_range":
        a = rng.randrange(0, 2**32)
        b = a + rng.randrange(-100, 2**24)
        return {"type": "ipv4_range",
                "field": rng.choice(fields_str + fields_num),
                "min": a, "max": min(b, 2**32 - 1)}
    if kind == "len_range":
        a = rng.randrange(0, 30)
        return {"type": "len_range",
                "field": rng.choice(fields_str + fields_num),
                "min": a, "max": a + rng.randrange(0, 30)}
    if kind == "day_range":
        a = rng.randrange(0, 86400 * 10**9)
        d = {"type": "day_range", "start": a,
             "end": min(a + rng.randrange(0, 10**13), 86400 * 10**9 - 1)}
        if rng.random() < 0.5:
            d["offset"] = rng.randrange(-14, 14) * 3600 * 10**9
        return d
    if kind == "week_range":
        a = rng.randrange(0, 7)
        d = {"type": "week_range", "start": a, "end": rng.randrange(a, 7)}
        if rng.random() < 0.5:
            d["offset"] = rng.randrange(-14, 14) * 3600 * 10**9
        return d
    if kind == "value_type":
        return {"type": "value_type",
                "field": rng.choice(fields_str + fields_num),
                "value_type": rng.choice(
                    ["string", "dict", "uint8", "uint16", "uint32", "uint64",
                     "int64", "float64", "ipv4", "iso8601", "const"])}
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
