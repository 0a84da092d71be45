"""Format-compatibility fixture: a FormatVersion-3 part written byte-by-byte
by tools/gen_handmade_part.py straight from the reference's on-disk format
spec (NOT by this repo's C++ PartWriter) and committed under tests/golden.
Drift between our writer and reader would pass writer->reader round trips
but fail here.  Expected row sets are hand-computed
from the fixture data in the generator.

The part deliberately mixes per-block encodings of the same column:
_msg is string-typed (bloom) in block 1 and dict-typed (no bloom) in
block 2; "code" is uint8 in block 1 and dict in block 2; "host" is a const
column; "level"/"code" live in the values.bin0/bloom.bin0 shard mapped by
column_idxs.bin; timestamps use the DeltaConst codec."""

import os

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
PART = os.path.join(HERE, "golden", "handmade_part")

# filter -> expected matching rows per block (hand-computed from the
# generator's fixture data)
CASES = [
    ('{"type":"phrase","field":"_msg","phrase":"alpha"}',
     [[0, 1], [0, 2]]),
    ('{"type":"phrase","field":"_msg","phrase":"beta"}',
     [[0, 1, 3], [2]]),
    ('{"type":"phrase","field":"_msg","phrase":"omega"}',
     [[], [0, 1]]),
    ('{"type":"exact","field":"_msg","value":"alpha omega"}',
     [[], [0]]),
    ('{"type":"phrase","field":"level","phrase":"error"}',
     [[1, 4, 7], []]),
    ('{"type":"phrase","field":"level","phrase":"info"}',
     [[0, 2, 5, 8], [1]]),
    ('{"type":"phrase","field":"host","phrase":"h1"}',
     [[0, 1, 2, 3, 4, 5, 6, 7, 8], [0, 1, 2]]),
    ('{"type":"phrase","field":"host","phrase":"h2"}',
     [[], []]),
    ('{"type":"range","field":"code","min":3,"max":7}',
     [[2, 3, 4, 5, 6], [0, 1]]),
    ('{"type":"exact","field":"code","value":"9"}',
     [[8], [2]]),
    ('{"type":"time","min":1700000001000000000,"max":1700000001000001000}',
     [[], [0, 1, 2]]),
    ('{"type":"time","min":1700000000000000000,"max":1700000000000002000}',
     [[0, 1, 2], []]),
    ('{"type":"and","filters":['
     '{"type":"phrase","field":"_msg","phrase":"alpha"},'
     '{"type":"phrase","field":"level","phrase":"error"}]}',
     [[1], []]),
    ('{"type":"regexp","field":"_msg","re":"(gamma|omega)"}',
     [[2, 3], [0, 1]]),
    ('{"type":"not","filter":'
     '{"type":"phrase","field":"_msg","phrase":"alpha"}}',
     [[2, 3, 4, 5, 6, 7, 8], [1]]),
    # typed columns (block 1 only; missing in block 2)
    ('{"type":"range","field":"u16c","min":302,"max":305}',
     [[2, 3, 4, 5], []]),
    ('{"type":"exact","field":"u32c","value":"70007"}', [[7], []]),
    ('{"type":"range","field":"u64c","min":5000000001,"max":5000000002}',
     [[1, 2], []]),
    ('{"type":"exact","field":"i64c","value":"-3"}', [[1], []]),
    ('{"type":"range","field":"i64c","min":-2,"max":1}',
     [[2, 3, 4, 5], []]),
    ('{"type":"range","field":"f64c","min":1.0,"max":3.0}', [[1, 2], []]),
    ('{"type":"phrase","field":"ipc","phrase":"10.0.0.4"}', [[4], []]),
    ('{"type":"ipv4_range","field":"ipc","min":167772162,"max":167772165}',
     [[2, 3, 4, 5], []]),
    ('{"type":"exact","field":"isoc","value":"2024-01-01T00:00:03.000Z"}',
     [[3], []]),
    ('{"type":"phrase","field":"isoc","phrase":"2024"}',
     [[0, 1, 2, 3, 4, 5, 6, 7, 8], []]),
]

BLOCK_ROWS = [9, 3]


def expected_bits(per_block):
    words = b""
    for bi, rows in enumerate(per_block):
        w = 0
        for r in rows:
            w |= 1 << r
        words += w.to_bytes(8, "little")
    return words


def test_handmade_part_oracle():
    from victorialogs_amd import OracleScanner

    orc = OracleScanner(PART)
    try:
        assert orc.blocks == 2
        assert [orc.block_rows(i) for i in range(2)] == BLOCK_ROWS
        for fjson, per_block in CASES:
            hits, bits = orc.scan(fjson, with_bitmaps=True)
            assert hits == sum(len(r) for r in per_block), fjson
            assert bits == expected_bits(per_block), fjson
    finally:
        orc.close()


@pytest.mark.gpu
def test_handmade_part_gpu():
    from victorialogs_amd import Filter, Part, Stage

    part = Part(PART)
    try:
        assert part.blocks == 2
        for fjson, per_block in CASES:
            filt = Filter(fjson)
            st = Stage(part, filt, device=0)
            try:
                hits = st.scan()
                bits = st.fetch_bitmaps(2)
                assert hits == sum(len(r) for r in per_block), fjson
                assert bits == expected_bits(per_block), fjson
            finally:
                st.close()
                filt.close()
    finally:
        part.close()
