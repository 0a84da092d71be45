"""Algebraic properties every row-local filter must satisfy exactly —
an oracle self-check independent of the GPU and of hand-computed fixtures:

- complement: hits(f) + hits(not f) == rows
- intersection: hits(and(f,g)) <= min(hits(f), hits(g))
- union: hits(or(f,g)) >= max and == hits(f)+hits(g)-hits(and(f,g))
- determinism: two scans agree
"""

import json
import random

import pytest

from tests.test_gpu_fuzz import (GEN_FIELDS_NUM, GEN_FIELDS_STR, GEN_PHRASES,
                                 GEN_REGEXES, random_tree)
from victorialogs_amd import OracleScanner


@pytest.fixture(scope="module")
def scanner(gen_part):
    sc = OracleScanner(gen_part)
    yield sc
    sc.close()


def rows_of(sc):
    return sum(sc.block_rows(i) for i in range(sc.blocks))


def test_filter_algebra(scanner):
    rng = random.Random(4242)
    rows = rows_of(scanner)
    for _ in range(60):
        f = random_tree(rng, GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM,
                        GEN_REGEXES, depth=2)
        g = random_tree(rng, GEN_PHRASES, GEN_FIELDS_STR, GEN_FIELDS_NUM,
                        GEN_REGEXES, depth=2)
        hf, bf = scanner.scan(json.dumps(f), with_bitmaps=True)
        hg, _ = scanner.scan(json.dumps(g))
        hnf, _ = scanner.scan(json.dumps({"type": "not", "filter": f}))
        hand, _ = scanner.scan(json.dumps({"type": "and", "filters": [f, g]}))
        hor, _ = scanner.scan(json.dumps({"type": "or", "filters": [f, g]}))
        assert hf + hnf == rows, f"complement broken for {f}"
        assert hand <= min(hf, hg), f"and not intersective for {f} & {g}"
        assert hor >= max(hf, hg), f"or not unioning for {f} | {g}"
        assert hor == hf + hg - hand, f"inclusion-exclusion broken {f} | {g}"
        hf2, bf2 = scanner.scan(json.dumps(f), with_bitmaps=True)
        assert (hf2, bf2) == (hf, bf), "scan not deterministic"
