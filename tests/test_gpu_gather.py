"""GPU gather (blockResult materialization): matched
rows' values compacted on-device must equal the fixture's expected values."""

import pytest

from victorialogs_amd import Filter, Part, Stage

pytestmark = pytest.mark.gpu

DICT_VALS = ["debug", "info", "warn", "error"]


def expected_rows(rows=300):
    """Rows of the typed_part fixture matching lvl == 'error' (block 1)."""
    return [i for i in range(rows) if i % 4 == 3]


@pytest.fixture(scope="module")
def stage(typed_part):
    part = Part(typed_part)
    filt = Filter('{"type":"phrase","field":"lvl","phrase":"error"}')
    st = Stage(part, filt, device=0)
    st.scan()
    yield st
    st.close()
    filt.close()
    part.close()


def test_gather_rowids(stage):
    values, rowids = stage.gather("lvl")
    exp = expected_rows()
    assert rowids == exp
    assert all(v == b"error" for v in values)


def test_gather_string_column(stage):
    values, rowids = stage.gather("_msg")
    exp = expected_rows()
    assert len(values) == len(exp)
    for v, i in zip(values, exp):
        want = f"log line {i} level={DICT_VALS[i % 4]} took {i % 50}ms".encode()
        assert v == want


def test_gather_const_column(stage):
    values, _ = stage.gather("constcol")
    assert all(v == b"fixed value 42" for v in values)


def test_gather_numeric_columns(stage):
    exp = expected_rows()
    values, _ = stage.gather("u8")
    assert values == [str(i % 250).encode() for i in exp]
    values, _ = stage.gather("u32")
    assert values == [str(70000 + i * 1009).encode() for i in exp]
    values, _ = stage.gather("i64")
    assert values == [str((i - 150) * 37).encode() for i in exp]
    values, _ = stage.gather("f64")
    # (i-150)/8 is exactly representable; check round-trip + shortest form
    for v, i in zip(values, exp):
        assert float(v) == (i - 150) / 8, f"{v} != {(i-150)/8}"
        assert b"e" not in v and len(v) <= 10
    values, _ = stage.gather("ip")
    assert values == [f"10.{i % 256}.{(i * 3) % 256}.{(i * 7) % 256}".encode()
                      for i in exp]
    values, _ = stage.gather("iso")
    assert values == [
        b"2024-01-%02dT%02d:%02d:%02d.%03dZ"
        % (1 + i % 28, i % 24, i % 60, (i * 3) % 60, i % 1000)
        for i in exp
    ]


def test_gather_missing_column(stage):
    values, rowids = stage.gather("no_such_column")
    exp = expected_rows()
    assert rowids == exp
    assert all(v == b"" for v in values)


def test_gather_multiblock(typed_part):
    """Gather across blocks with matches in both (global rowids)."""
    part = Part(typed_part)
    filt = Filter('{"type":"noop"}')  # all rows match
    st = Stage(part, filt, device=0)
    hits = st.scan()
    assert hits == 600
    values, rowids = st.gather("u8")
    assert rowids == list(range(600))
    want = [str(i % 250).encode() for i in range(300)] + \
           [str(i % 7).encode() for i in range(300)]
    assert values == want
    st.close()
    filt.close()
    part.close()
