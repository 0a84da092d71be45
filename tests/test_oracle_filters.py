"""Oracle filter-evaluation tests on custom fixtures with hand-computed
expected row sets (the filter_test.go:158-277 fixture pattern)."""

import pytest

from victorialogs_amd import OracleScanner, write_custom_part


def decode_bitmaps(orc, words: bytes):
    """Maps the concatenated per-block padded bitmap words back to global row
    indices (bitmap.go:113-125 layout, blocks padded to word boundaries)."""
    out = set()
    word_idx = 0
    row_base = 0
    for b in range(orc.blocks):
        rows = orc.block_rows(b)
        nwords = (rows + 63) // 64
        chunk = words[word_idx * 8:(word_idx + 0 + nwords) * 8]
        v = int.from_bytes(chunk, "little")
        i = 0
        while v:
            if v & 1:
                out.add(row_base + i)
            v >>= 1
            i += 1
        word_idx += nwords
        row_base += rows
    return out


def scan_rows(part_dir, fjson, rows):
    orc = OracleScanner(part_dir)
    try:
        hits, words = orc.scan(fjson, with_bitmaps=True)
        got = decode_bitmaps(orc, words)
        assert len(got) == hits
        assert max(got, default=0) < rows
        return got
    finally:
        orc.close()


@pytest.fixture(scope="module")
def small_part(tmp_path_factory):
    ts = [100 + i for i in range(10)]
    spec = {
        "blocks": [
            {
                "stream": 0,
                "timestamps": ts,
                "columns": [
                    {"name": "_msg", "values": [
                        "abc def",        # 0
                        "abc. def",       # 1
                        "abcdef",         # 2
                        "",               # 3
                        "x abc y",        # 4
                        "ABC def",        # 5
                        "abc",            # 6
                        "def abc",        # 7
                        "раз два",        # 8
                        "abc-def",        # 9
                    ]},
                    {"name": "n", "values": [str(i * 10) for i in range(10)]},
                ],
            }
        ]
    }
    d = str(tmp_path_factory.mktemp("p") / "small")
    write_custom_part(d, spec)
    return d


def test_phrase_semantics(small_part):
    f = '{"type":"phrase","field":"_msg","phrase":"abc"}'
    assert scan_rows(small_part, f, 10) == {0, 1, 4, 6, 7, 9}
    f = '{"type":"phrase","field":"_msg","phrase":"abc def"}'
    assert scan_rows(small_part, f, 10) == {0}
    f = '{"type":"phrase","field":"_msg","phrase":""}'
    assert scan_rows(small_part, f, 10) == {3}
    f = '{"type":"phrase","field":"_msg","phrase":"два"}'
    assert scan_rows(small_part, f, 10) == {8}
    f = '{"type":"phrase","field":"_msg","phrase":"ABC"}'
    assert scan_rows(small_part, f, 10) == {5}  # case-sensitive


def test_uint_column_exact_and_range(small_part):
    # "n" encodes as uint8 (0..90)
    f = '{"type":"phrase","field":"n","phrase":"30"}'
    assert scan_rows(small_part, f, 10) == {3}
    f = '{"type":"exact","field":"n","value":"90"}'
    assert scan_rows(small_part, f, 10) == {9}
    f = '{"type":"range","field":"n","min":25,"max":55}'
    assert scan_rows(small_part, f, 10) == {3, 4, 5}
    f = '{"type":"range","field":"n","min":-5,"max":0}'
    assert scan_rows(small_part, f, 10) == {0}
    # out-of-range prune
    f = '{"type":"range","field":"n","min":1000,"max":2000}'
    assert scan_rows(small_part, f, 10) == set()


def test_missing_column(small_part):
    f = '{"type":"phrase","field":"nosuch","phrase":"x"}'
    assert scan_rows(small_part, f, 10) == set()
    # empty phrase on a missing column matches everything
    f = '{"type":"phrase","field":"nosuch","phrase":""}'
    assert scan_rows(small_part, f, 10) == set(range(10))


def test_bool_combinators(small_part):
    f = ('{"type":"and","filters":['
         '{"type":"phrase","field":"_msg","phrase":"abc"},'
         '{"type":"phrase","field":"_msg","phrase":"def"}]}')
    assert scan_rows(small_part, f, 10) == {0, 1, 7, 9}
    f = ('{"type":"or","filters":['
         '{"type":"phrase","field":"_msg","phrase":"abcdef"},'
         '{"type":"phrase","field":"_msg","phrase":"два"}]}')
    assert scan_rows(small_part, f, 10) == {2, 8}
    f = ('{"type":"not","filter":'
         '{"type":"phrase","field":"_msg","phrase":"abc"}}')
    assert scan_rows(small_part, f, 10) == {2, 3, 5, 8}
    # nested: not(and(abc, def)) and time subset
    f = ('{"type":"and","filters":['
         '{"type":"time","min":100,"max":104},'
         '{"type":"not","filter":{"type":"and","filters":['
         '{"type":"phrase","field":"_msg","phrase":"abc"},'
         '{"type":"phrase","field":"_msg","phrase":"def"}]}}]}')
    assert scan_rows(small_part, f, 10) == {2, 3, 4}


def test_time_filter(small_part):
    f = '{"type":"time","min":103,"max":105}'
    assert scan_rows(small_part, f, 10) == {3, 4, 5}
    f = '{"type":"time","min":0,"max":1000}'
    assert scan_rows(small_part, f, 10) == set(range(10))
    f = '{"type":"time","min":200,"max":100}'  # min > max
    assert scan_rows(small_part, f, 10) == set()


def test_regex_classes(small_part):
    f = '{"type":"regexp","field":"_msg","re":"abc"}'
    # literal => contains
    assert scan_rows(small_part, f, 10) == {0, 1, 2, 4, 6, 7, 9}
    f = '{"type":"regexp","field":"_msg","re":"abc|два"}'
    assert scan_rows(small_part, f, 10) == {0, 1, 2, 4, 6, 7, 8, 9}
    f = '{"type":"regexp","field":"_msg","re":"abc.*def"}'
    # prefix "abc" + dotstar... SimplifyRegex: prefix="abc", suffix="def" =>
    # orValues path with prefix loop
    assert scan_rows(small_part, f, 10) == {0, 1, 2, 9}
    f = '{"type":"regexp","field":"_msg","re":"abc.+"}'
    # rows where something follows the FIRST "abc" occurrence
    assert scan_rows(small_part, f, 10) == {0, 1, 2, 4, 9}
    f = '{"type":"regexp","field":"_msg","re":".*def.*"}'
    assert scan_rows(small_part, f, 10) == {0, 1, 2, 5, 7, 9}


def test_unsupported_regex_rejected(small_part):
    from victorialogs_amd import oracle_helpers
    lib = oracle_helpers()
    assert lib.orc_compile_filter(
        b'{"type":"regexp","field":"_msg","re":"a\\\\p{L}b"}') in (None, 0)
    err = lib.orc_errstr().decode()
    assert "not supported" in err or "fast-path" in err


def test_typed_part_sanity(typed_part):
    # sanity counts on the typed fixture (hand-computed)
    rows = 600  # 2 blocks x 300
    f = '{"type":"phrase","field":"lvl","phrase":"error"}'
    assert len(scan_rows(typed_part, f, rows)) == 75  # block 1: i%4==3
    f = '{"type":"exact","field":"lvl","value":"ERROR"}'
    assert len(scan_rows(typed_part, f, rows)) == 150  # block 2: i%2==1
    f = '{"type":"phrase","field":"u8","phrase":"13"}'
    # block1: i%250==13 -> i in {13,263} (2 rows); block2: i%7==13 never
    assert len(scan_rows(typed_part, f, rows)) == 2
    f = '{"type":"phrase","field":"constcol","phrase":"value"}'
    assert len(scan_rows(typed_part, f, rows)) == 300  # const col block 1 only
    f = '{"type":"phrase","field":"uni","phrase":"два"}'
    assert len(scan_rows(typed_part, f, rows)) == 100  # i%3==0 in block 1


def test_oracle_multithreaded_equals_single(gen_part):
    orc = OracleScanner(gen_part)
    try:
        f = '{"type":"phrase","field":"_msg","phrase":"the stream 1 and"}'
        h1, w1 = orc.scan(f, with_bitmaps=True, threads=1)
        h4, w4 = orc.scan(f, with_bitmaps=True, threads=4)
        assert h1 == h4 and w1 == w4
    finally:
        orc.close()


def test_complex_filters_reference_fixture(tmp_path):
    """TestComplexFilters ported verbatim (filter_test.go:14-156): the
    reference's own column fixture and expected row sets."""
    import json

    from victorialogs_amd import OracleScanner, write_custom_part

    values = [
        "a foo",
        "a foobar",
        "aa abc a",
        "ca afdf a,foobar baz",
        "a fddf foobarbaz",
        "a",
        "a foobar abcdef",
        "a kjlkjf dfff",
        "a ТЕСТЙЦУК НГКШ ",
        "a !!,23.(!1)",
    ]
    d = str(tmp_path / "complex")
    write_custom_part(d, {"blocks": [{
        "stream": 0,
        "timestamps": [1700000000000000000 + i for i in range(len(values))],
        "columns": [{"name": "foo", "values": values}],
    }]})

    def ph(p):
        return {"type": "phrase", "field": "foo", "phrase": p}

    cases = [
        # (foobar AND NOT baz AND (abcdef OR xyz)) -> [6]
        ({"type": "and", "filters": [
            ph("foobar"),
            {"type": "not", "filter": ph("baz")},
            {"type": "or", "filters": [ph("abcdef"), ph("xyz")]}]}, [6]),
        # (foobaz AND NOT baz AND (abcdef OR xyz)) -> []
        ({"type": "and", "filters": [
            ph("foobaz"),
            {"type": "not", "filter": ph("baz")},
            {"type": "or", "filters": [ph("abcdef"), ph("xyz")]}]}, []),
        # (foobar AND NOT baz AND (abcdef OR xyz OR a)) -> [1, 6]
        ({"type": "and", "filters": [
            ph("foobar"),
            {"type": "not", "filter": ph("baz")},
            {"type": "or", "filters": [ph("abcdef"), ph("xyz"), ph("a")]}]},
         [1, 6]),
        # (foobar AND NOT qwert AND (abcdef OR xyz OR a)) -> [1, 3, 6]
        ({"type": "and", "filters": [
            ph("foobar"),
            {"type": "not", "filter": ph("qwert")},
            {"type": "or", "filters": [ph("abcdef"), ph("xyz"), ph("a")]}]},
         [1, 3, 6]),
    ]
    sc = OracleScanner(d)
    try:
        for f, want in cases:
            hits, bits = sc.scan(json.dumps(f), with_bitmaps=True)
            word = int.from_bytes(bits[:8], "little")
            rows = [i for i in range(len(values)) if (word >> i) & 1]
            assert rows == want, f"{f}: got {rows} want {want}"
            assert hits == len(want)
    finally:
        sc.close()
