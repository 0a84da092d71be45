"""Golden-vector tests pinning the oracle against the reference's own
known-answer tests:

- XXH64 public vectors (pins vendor/github.com/cespare/xxhash/v2)
- bloom hex goldens        lib/logstorage/bloomfilter_test.go:105-119
- TestMatchPhrase table    lib/logstorage/filter_phrase_test.go:9-60
- tokenizer behavior       lib/logstorage/tokenizer.go:12-117 semantics
- float/iso8601 formatting Go strconv.AppendFloat / time layout semantics
"""

import ctypes

import pytest

from victorialogs_amd import oracle_helpers


@pytest.fixture(scope="module")
def lib():
    return oracle_helpers()


def xxh(lib, data: bytes) -> int:
    return lib.orc_xxhash64(data, len(data))


def test_xxhash64_known_answers(lib):
    # canonical XXH64 vectors, seed 0
    assert xxh(lib, b"") == 0xEF46DB3751D8E999
    assert xxh(lib, b"a") == 0xD24EC4F1A98C6E5B
    assert xxh(lib, b"abc") == 0x44BC2CF5AD770999
    assert xxh(lib, b"Nobody inspects the spammish repetition") == 0xFBCEA83C8A378BF1
    # >32-byte path
    assert xxh(lib, b"The quick brown fox jumps over the lazy dog") == 0x0B242D361FDA71BC


def bloom_tokens(lib, tokens):
    buf = ctypes.create_string_buffer(4096)
    n = lib.orc_bloom_marshal_tokens("\n".join(tokens).encode(), buf, 4096)
    return buf.raw[:n]


def test_bloom_marshal_tokens_golden(lib):
    # bloomfilter_test.go:115-118 hex known answers
    assert bloom_tokens(lib, []) == b""
    assert bloom_tokens(lib, ["foo"]) == bytes.fromhex("0000008240180004")
    assert bloom_tokens(lib, ["foo", "bar", "baz"]) == bytes.fromhex(
        "000081A3485C1026"
    )
    assert bloom_tokens(lib, ["foo", "bar", "baz", "foo"]) == bytes.fromhex(
        "000081A3485C1026"
    )


MATCH_PHRASE_TABLE = [
    # filter_phrase_test.go:20-48 verbatim
    ("", "", True),
    ("foo", "", False),
    ("", "foo", False),
    ("foo", "foo", True),
    ("foo bar", "foo", True),
    ("foo bar", "bar", True),
    ("a foo bar", "foo", True),
    ("a foo bar", "fo", False),
    ("a foo bar", "oo", False),
    ("foobar", "foo", False),
    ("foobar", "bar", False),
    ("foobar", "oob", False),
    ("afoobar foo", "foo", True),
    ("раз два (три!)", "три", True),
    ("", "foo bar", False),
    ("foo bar", "foo bar", True),
    ("(foo bar)", "foo bar", True),
    ("afoo bar", "foo bar", False),
    ("afoo bar", "afoo ba", False),
    ("foo bar! baz", "foo bar!", True),
    ("a.foo bar! baz", ".foo bar! ", True),
    ("foo bar! baz", "foo bar! b", False),
    ("255.255.255.255", "5", False),
    ("255.255.255.255", "55", False),
    ("255.255.255.255", "255", True),
    ("255.255.255.255", "5.255", False),
    ("255.255.255.255", "255.25", False),
    ("255.255.255.255", "255.255", True),
]


def test_match_phrase_truth_table(lib):
    for s, phrase, want in MATCH_PHRASE_TABLE:
        sb, pb = s.encode(), phrase.encode()
        got = lib.orc_match_phrase(sb, len(sb), pb, len(pb)) == 1
        assert got == want, f"matchPhrase({s!r}, {phrase!r}) = {got}, want {want}"


def tokenize(lib, s):
    buf = ctypes.create_string_buffer(65536)
    sb = s.encode()
    n = lib.orc_tokenize(sb, len(sb), buf, 65536)
    out = buf.raw[:n].decode()
    return out.split("\n") if out else []


def test_tokenizer(lib):
    assert tokenize(lib, "foo bar baz") == ["foo", "bar", "baz"]
    assert tokenize(lib, "foo bar foo") == ["foo", "bar"]  # dedup
    assert tokenize(lib, "ip=1.2.3.4; x_y=9") == ["ip", "1", "2", "3", "4", "x_y", "9"]
    assert tokenize(lib, "") == []
    assert tokenize(lib, "!!!") == []
    assert tokenize(lib, "раз два（три)") == ["раз", "два", "три"]
    assert tokenize(lib, "_lead tail_") == ["_lead", "tail_"]


def fmt_float(lib, f):
    buf = ctypes.create_string_buffer(2048)
    n = lib.orc_format_float64(f, buf, 2048)
    return buf.raw[:n].decode()


def test_float_formatting_matches_go_strconv(lib):
    # Go strconv.AppendFloat(dst, f, 'f', -1, 64) shortest round-trip
    cases = [
        (0.0, "0"),
        (1.0, "1"),
        (-1.5, "-1.5"),
        (0.3, "0.3"),
        (123.456, "123.456"),
        (0.1 + 0.2, "0.30000000000000004"),
        (1e20, "100000000000000000000"),
        (2.5e-5, "0.000025"),
        (1234.5678, "1234.5678"),
        (-0.0, "-0"),
    ]
    for f, want in cases:
        assert fmt_float(lib, f) == want, f"format({f}) != {want}"
    # round-trip property on assorted values
    for f in [3.14159, 1 / 3, 9007199254740991.0, 5e-324, 1.7976931348623157e308]:
        s = fmt_float(lib, f)
        assert float(s) == f, f"{s} does not round-trip to {f}"


def test_iso8601_format_parse_roundtrip(lib):
    buf = ctypes.create_string_buffer(64)
    out = ctypes.c_longlong()
    for nsecs in [0, 1700000000_123000000, 946684800_000000000,
                  4102444799_999000000]:
        n = lib.orc_format_iso8601(nsecs, buf, 64)
        s = buf.raw[:n]
        assert len(s) == 24 and s.endswith(b"Z")
        assert lib.orc_parse_iso8601(s, len(s), ctypes.byref(out)) == 1
        # formatting truncates to ms; parse returns the ms-truncated value
        assert out.value == nsecs // 1000000 * 1000000
    # known rendering
    n = lib.orc_format_iso8601(1700000000_123000000, buf, 64)
    assert buf.raw[:n] == b"2023-11-14T22:13:20.123Z"


def test_parse_iso8601_rejects(lib):
    out = ctypes.c_longlong()
    # NB: a ' ' delimiter instead of 'T' IS accepted (values_encoder.go:495-501)
    for bad in [b"2023-11-14T22:13:20Z", b"2023/11/14T22:13:20.123Z",
                b"1500-01-01T00:00:00.000Z", b"2023-13-40T22:13:20.12Z"]:
        assert lib.orc_parse_iso8601(bad, len(bad), ctypes.byref(out)) == 0


def test_rfc3339_parse_table(lib):
    """TryParseTimestampRFC3339Nano success/failure tables
    (values_encoder_test.go:151-236), expected nsecs via Python datetime."""
    import ctypes
    from datetime import datetime, timezone

    lib.orc_parse_rfc3339.restype = ctypes.c_longlong

    def parse(s):
        out = ctypes.c_longlong()
        r = lib.orc_parse_rfc3339(s.encode(), len(s), ctypes.byref(out))
        return out.value if r == 0 else None

    def py_ns(iso, frac_ns=0):
        dt = datetime.strptime(iso, "%Y-%m-%dT%H:%M:%S").replace(
            tzinfo=timezone.utc)
        return int(dt.timestamp()) * 10**9 + frac_ns

    ok = [
        ("2023-01-15T23:45:51Z", py_ns("2023-01-15T23:45:51")),
        ("2023-01-15T23:45:51.1Z", py_ns("2023-01-15T23:45:51", 100000000)),
        ("2023-01-15T23:45:51.123456789Z",
         py_ns("2023-01-15T23:45:51", 123456789)),
        ("1677-09-21T00:12:44Z", py_ns("1677-09-21T00:12:44")),
        ("2262-04-11T23:47:15.999999999Z",
         py_ns("2262-04-11T23:47:15", 999999999)),
        # timezone offsets (values_encoder_test.go:186-188)
        ("2023-01-16T00:45:51+01:00", py_ns("2023-01-15T23:45:51")),
        ("2023-01-16T00:45:51.123-01:00",
         py_ns("2023-01-16T01:45:51", 123000000)),
        # SQL datetime delimiter
        ("2023-01-16 00:45:51+01:00", py_ns("2023-01-15T23:45:51")),
        ("2023-01-16 00:45:51.123-01:00",
         py_ns("2023-01-16T01:45:51", 123000000)),
    ]
    for s, want in ok:
        assert parse(s) == want, s

    bad = ["", "foobar", "2023-01-15T22:15:51.Z", "1676-09-21T00:12:43Z",
           "2263-04-11T23:47:17Z", "1677-09-21T00:12:43.999999999Z",
           "2262-04-11T23:47:16Z", "YYYY-04-11T23:47:17Z",
           "2023-MM-11T23:47:17Z", "2023-01-DDT23:47:17Z"]
    for s in bad:
        assert parse(s) is None, s


def test_parse_math_number_legs(lib):
    """parseMathNumber legs (pipe_math.go:1066-1080, block_result.go:2710-2752)."""
    import ctypes
    import math

    lib.orc_parse_math_number.restype = ctypes.c_double

    def pm(s):
        return lib.orc_parse_math_number(s.encode(), len(s))

    cases = [
        ("123", 123.0), ("-1.5", -1.5), ("1e5", 1e5), ("1.5e-3", 0.0015),
        ("0x1F", 31.0), ("0o17", 15.0), ("0b101", 5.0), ("017", 17.0),
        ("1_000", 1000.0), ("inf", math.inf), ("-Inf", -math.inf),
        ("1.5KB", 1500.0), ("2h30m", 9e12), ("100ms", 1e8),
        ("10.0.0.1", 167772161.0),
        ("2024-01-01T00:00:00Z", 1704067200e9),
        ("0x1.8p1", 3.0),
        ("9007199254740993", 9007199254740992.0),  # tryParseFloat64 leg, lossy
    ]
    for s, want in cases:
        got = pm(s)
        assert got == want, f"{s}: got {got} want {want}"
    for s in ["abc", "", "1.2.3", "12:34", "nan?"]:
        assert math.isnan(pm(s)), s
