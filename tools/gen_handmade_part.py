#!/usr/bin/env python3
"""Writes a tiny FormatVersion-3 VictoriaLogs part BYTE BY BYTE from the
reference's on-disk format specification — deliberately NOT using this
repo's C++ PartWriter, so that scanning it with the oracle and the GPU is
a (semi-)independent format-compatibility check: a byte-level drift between
our writer and reader would pass writer→reader round trips but fail here
(VERDICT r01 missing item 6 — no Go toolchain exists on
any box, so a part written by the actual reference binary is unobtainable;
this restates the format directly from the reference source instead).

Layout sources (all in /root/reference):
  lib/logstorage/part_header.go:15-80        metadata.json fields
  lib/logstorage/index_block_header.go       metaindex.bin records
  lib/logstorage/block_header.go:69-80       blockHeader marshal
  lib/logstorage/block_header.go:990-997     timestampsHeader marshal
  lib/logstorage/block_header.go:454-484     columnsHeader (+inline counts)
  lib/logstorage/block_header.go:275-347     columnsHeaderIndex
  lib/logstorage/block_header.go:634-712     columnHeader per-type marshal
  lib/logstorage/values_encoder.go:1289-1297 valuesDict marshal
  lib/logstorage/rows.go:35-41               const-column Field marshal
  lib/logstorage/column_names.go:7-40        column_names.bin, column_idxs.bin
  lib/logstorage/encoding.go:16-50,149-242,338-370  strings/uint64/bytes blocks
  lib/logstorage/bloomfilter.go:49-55,74-170 bloom build + marshal (BE words)
  vendor/.../lib/encoding/encoding.go:82-96,232-246  timestamps DeltaConst
  vendor/.../lib/encoding/int.go             BE ints, varuint (LEB128), varint
  lib/logstorage/tokenizer.go:132-140        token chars [a-zA-Z0-9_]

The XXH64 below is implemented from the public xxHash spec and self-checked
against its published test vectors AND the reference's own bloom hex
known-answer test (bloomfilter_test.go:105-119) before any file is written.
"""

import ctypes
import ctypes.util
import json
import os
import struct
import sys

MASK64 = (1 << 64) - 1

# ---- XXH64 (public spec; seed 0) ----
P1 = 11400714785074694791
P2 = 14029467366897019727
P3 = 1609587929392839161
P4 = 9650029242287828579
P5 = 2870177450012600261


def rotl(x, r):
    return ((x << r) | (x >> (64 - r))) & MASK64


def xxh64(data, seed=0):
    n = len(data)
    i = 0
    if n >= 32:
        v1 = (seed + P1 + P2) & MASK64
        v2 = (seed + P2) & MASK64
        v3 = seed
        v4 = (seed - P1) & MASK64
        while i + 32 <= n:
            for _ in range(1):
                pass
            lanes = struct.unpack_from("<4Q", data, i)
            v1 = (rotl((v1 + lanes[0] * P2) & MASK64, 31) * P1) & MASK64
            v2 = (rotl((v2 + lanes[1] * P2) & MASK64, 31) * P1) & MASK64
            v3 = (rotl((v3 + lanes[2] * P2) & MASK64, 31) * P1) & MASK64
            v4 = (rotl((v4 + lanes[3] * P2) & MASK64, 31) * P1) & MASK64
            i += 32
        h = (rotl(v1, 1) + rotl(v2, 7) + rotl(v3, 12) + rotl(v4, 18)) & MASK64
        for v in (v1, v2, v3, v4):
            h = ((h ^ (rotl((v * P2) & MASK64, 31) * P1 & MASK64)) * P1 + P4) & MASK64
    else:
        h = (seed + P5) & MASK64
    h = (h + n) & MASK64
    while i + 8 <= n:
        k = struct.unpack_from("<Q", data, i)[0]
        h = ((rotl(h ^ (rotl((k * P2) & MASK64, 31) * P1 & MASK64), 27) * P1) + P4) & MASK64
        i += 8
    if i + 4 <= n:
        k = struct.unpack_from("<I", data, i)[0]
        h = ((rotl(h ^ (k * P1 & MASK64), 23) * P2) + P3) & MASK64
        i += 4
    while i < n:
        h = (rotl(h ^ (data[i] * P5 & MASK64), 11) * P1) & MASK64
        i += 1
    h ^= h >> 33
    h = (h * P2) & MASK64
    h ^= h >> 29
    h = (h * P3) & MASK64
    h ^= h >> 32
    return h


# public xxHash test vectors (seed 0); the >=32-byte stripe path is further
# pinned by the reference-byte bloom known answers just below
assert xxh64(b"") == 0xEF46DB3751D8E999
assert xxh64(b"a") == 0xD24EC4F1A98C6E5B
assert xxh64(b"abc") == 0x44BC2CF5AD770999
assert xxh64(bytes(range(64))) == 0xF7C67301DB6713F0

# ---- tokenizer ([a-zA-Z0-9_] runs; ASCII fixture data only) ----


def tokenize_unique(values):
    seen = []
    seen_set = set()
    for v in values:
        tok = []
        for ch in v:
            if ch.isascii() and (ch.isalnum() or ch == "_"):
                tok.append(ch)
            else:
                if tok:
                    t = "".join(tok)
                    if t not in seen_set:
                        seen_set.add(t)
                        seen.append(t)
                    tok = []
        if tok:
            t = "".join(tok)
            if t not in seen_set:
                seen_set.add(t)
                seen.append(t)
    return seen


# ---- bloom (bloomfilter.go:74-170): 16 bits/unique-token, 6 probes ----


def bloom_marshal(tokens):
    bits_count = len(tokens) * 16
    words = (bits_count + 63) // 64
    bits = [0] * words
    max_bits = words * 64
    for tok in tokens:
        h = xxh64(tok.encode())
        for _ in range(6):
            hk = xxh64(struct.pack("<Q", h))
            h = (h + 1) & MASK64
            idx = hk % max_bits
            bits[idx >> 6] |= 1 << (idx & 63)
    return b"".join(struct.pack(">Q", w) for w in bits)


# the reference's own known answers (bloomfilter_test.go:105-119)
assert bloom_marshal(["foo"]).hex() == "0000008240180004"
assert bloom_marshal(["foo", "bar", "baz"]).hex() == "000081a3485c1026"

# ---- primitive encoders (vendor/.../lib/encoding/int.go) ----


def be16(v):
    return struct.pack(">H", v)


def be32(v):
    return struct.pack(">I", v)


def be64(v):
    return struct.pack(">Q", v)


def varuint(v):
    out = bytearray()
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def varint(v):
    # zig-zag then varuint (int.go:69-74)
    return varuint(((v << 1) ^ (v >> 63)) & MASK64)


def marshal_bytes(b):
    # encoding.MarshalBytes: varuint len + bytes
    return varuint(len(b)) + b


# ---- zstd via libzstd.so.1 (only the codec; layout logic stays here) ----
_zstd = ctypes.CDLL("libzstd.so.1")
_zstd.ZSTD_compressBound.restype = ctypes.c_size_t
_zstd.ZSTD_compress.restype = ctypes.c_size_t
_zstd.ZSTD_isError.restype = ctypes.c_uint


def zstd_compress(data, level=1):
    bound = _zstd.ZSTD_compressBound(len(data))
    buf = ctypes.create_string_buffer(bound)
    n = _zstd.ZSTD_compress(buf, bound, data, len(data), level)
    assert _zstd.ZSTD_isError(n) == 0
    return buf.raw[:n]


# ---- blocks of the encoding.go family ----


def bytes_block(data):
    # encoding.go:343-360
    if len(data) < 128:
        return bytes([0, len(data)]) + data
    comp = zstd_compress(data, 1)
    return bytes([1]) + varuint(len(comp)) + comp


def uint64_items(a):
    # encoding.go:190-242
    nmax = max(a) if a else 0
    consts = len(a) >= 2 and all(x == a[0] for x in a)
    if nmax < (1 << 8):
        if consts:
            return bytes([4, a[0]])
        return bytes([0]) + bytes(a)
    if nmax < (1 << 16):
        if consts:
            return bytes([5]) + be16(a[0])
        return bytes([1]) + b"".join(be16(x) for x in a)
    if nmax < (1 << 32):
        if consts:
            return bytes([6]) + be32(a[0])
        return bytes([2]) + b"".join(be32(x) for x in a)
    if consts:
        return bytes([7]) + be64(a[0])
    return bytes([3]) + b"".join(be64(x) for x in a)


def uint64_block(a):
    # encoding.go:149-155: items wrapped in a bytes block
    return bytes_block(uint64_items(a))


def strings_block(values):
    # encoding.go:16-50
    lens = [len(v) for v in values]
    out = uint64_block(lens)
    if len(values) >= 1 and all(v == values[0] for v in values) and len(values) > 1:
        out += bytes_block(values[0])
    else:
        out += bytes_block(b"".join(values))
    return out


# ---- the fixture part ----


def build_part(out_dir):
    os.makedirs(out_dir, exist_ok=True)
    ts_base = 1700000000000000000

    # rows per block; values are ASCII bytes
    blocks = [
        {
            "rows": 9,
            "ts0": ts_base,
            "ts_step": 1000,
            "msg": [b"alpha beta one", b"alpha beta two", b"gamma delta",
                    b"beta gamma", b"epsilon zeta eta", b"theta iota",
                    b"kappa lambda mu", b"nu xi omicron", b"pi rho sigma"],
            # 3 uniques -> dict (insertion order)
            "level": [b"info", b"error", b"info", b"warn", b"error",
                      b"info", b"warn", b"error", b"info"],
            # 9 uniques, numeric, max 9 -> uint8
            "code": [b"1", b"2", b"3", b"4", b"5", b"6", b"7", b"8", b"9"],
            "host": b"h1",
            "typed": True,  # adds one column per remaining valueType
        },
        {
            "rows": 3,
            "ts0": ts_base + 10**9,
            "ts_step": 500,
            # 3 uniques <= 8 -> dict-typed _msg (no bloom: block.go:159-168)
            "msg": [b"alpha omega", b"omega three", b"beta beta alpha"],
            "level": [b"debug", b"info", b"debug"],
            "code": [b"7", b"7", b"9"],  # 2 uniques -> dict (dict precedes uint)
            "host": b"h1",
        },
    ]

    # ---- timestamps.bin: DeltaConst per block ----
    ts_bin = b""
    ts_headers = []
    for b in blocks:
        payload = varint(b["ts_step"])  # encoding.go:232-246 DeltaConst
        ts_headers.append({
            "offset": len(ts_bin), "size": len(payload), "marshal_type": 2,
            "min": b["ts0"], "max": b["ts0"] + b["ts_step"] * (b["rows"] - 1),
        })
        ts_bin += payload

    # ---- column name table (ID = array index; column_names.go:101-134) ----
    names = ["", "level", "code", "host", "u16c", "u32c", "u64c", "i64c",
             "f64c", "ipc", "isoc"]
    name_id = {n: i for i, n in enumerate(names)}
    cn_plain = varuint(len(names)) + b"".join(marshal_bytes(n.encode())
                                              for n in names)
    column_names_bin = zstd_compress(cn_plain, 1)

    # column -> (values/bloom) shard: _msg goes to message_*; others shard 0
    # (marshalColumnIdxs: varuint count + {varuint nameID, varuint shard})
    idxs = [(name_id[n], 0) for n in ("level", "code", "u16c", "u32c",
                                      "u64c", "i64c", "f64c", "ipc", "isoc")]
    column_idxs_bin = varuint(len(idxs)) + b"".join(
        varuint(a) + varuint(b) for a, b in idxs)

    msg_values = bytearray()
    msg_bloom = bytearray()
    shard_values = bytearray()
    shard_bloom = bytearray()

    def encode_column(values, uniques_dict_ok=True):
        """Returns (valueType, encoded_rows, dict_values, minv, maxv).
        Mirrors the reference's encode priority for the shapes used here:
        dict for <=8 uniques, else uint8 for small numerics, else string
        (values_encoder.go:109-154)."""
        uniq = []
        for v in values:
            if v not in uniq:
                uniq.append(v)
        if len(uniq) > 1 and len(uniq) <= 8 and sum(len(u) for u in uniq) < 256:
            codes = [bytes([uniq.index(v)]) for v in values]
            return 2, codes, uniq, None, None
        if all(v.isdigit() for v in values):
            nums = [int(v) for v in values]
            if max(nums) < 256:
                return 3, [bytes([n]) for n in nums], None, min(nums), max(nums)
        return 1, list(values), None, None, None


    def zigzag(v):
        return ((v << 1) ^ (v >> 63)) & MASK64

    def iso_ns(i):
        # 2024-01-01T00:00:0i.000Z
        import calendar
        return (calendar.timegm((2024, 1, 1, 0, 0, i)) * 10**9)

    def typed_columns():
        """One column per remaining valueType (block_header.go:634-712
        per-type min/max layouts; values_encoder per-row BE encodings).
        9 distinct values defeat the dict encoding (<= 8 entries)."""
        import struct as st
        u16v = list(range(300, 309))
        u32v = list(range(70000, 70009))
        u64v = [5000000000 + i for i in range(9)]
        i64v = list(range(-4, 5))
        f64v = [0.5 + i for i in range(9)]
        ipv = [(10 << 24) | i for i in range(9)]
        isov = [iso_ns(i) for i in range(9)]
        fbits = lambda f: st.unpack(">Q", st.pack(">d", f))[0]
        cols = []
        cols.append(("u16c", 4, [str(v).encode() for v in u16v],
                     [be16(v) for v in u16v], be16(u16v[0]) + be16(u16v[-1])))
        cols.append(("u32c", 5, [str(v).encode() for v in u32v],
                     [be32(v) for v in u32v], be32(u32v[0]) + be32(u32v[-1])))
        cols.append(("u64c", 6, [str(v).encode() for v in u64v],
                     [be64(v) for v in u64v], be64(u64v[0]) + be64(u64v[-1])))
        # int64: encoding.MarshalInt64 is itself zig-zag + BE (int.go:69-74)
        # — used for BOTH the rows and the columnHeader min/max
        cols.append(("i64c", 10, [str(v).encode() for v in i64v],
                     [be64(zigzag(v)) for v in i64v],
                     be64(zigzag(i64v[0])) + be64(zigzag(i64v[-1]))))
        cols.append(("f64c", 7, [repr(v).encode() for v in f64v],
                     [be64(fbits(v)) for v in f64v],
                     be64(fbits(f64v[0])) + be64(fbits(f64v[-1]))))
        cols.append(("ipc", 8,
                     [("10.0.0.%d" % i).encode() for i in range(9)],
                     [be32(v) for v in ipv], be32(ipv[0]) + be32(ipv[-1])))
        cols.append(("isoc", 9,
                     [("2024-01-01T00:00:0%d.000Z" % i).encode()
                      for i in range(9)],
                     [be64(v) for v in isov], be64(isov[0]) + be64(isov[-1])))
        return cols

    # ---- per-block columns_header{,_index} + values/bloom files ----
    ch_bin = bytearray()      # columns_header.bin
    chi_bin = bytearray()     # columns_header_index.bin
    block_headers = []
    for bi, b in enumerate(blocks):
        cols = []  # (name, valueType, header-bytes builder info)
        for cname, vals in (("", b["msg"]), ("level", b["level"]),
                            ("code", b["code"])):
            vt, rows_enc, dictv, mn, mx = encode_column(vals)
            vblob = strings_block(rows_enc)
            if cname == "":
                vfile, bfile = msg_values, msg_bloom
            else:
                vfile, bfile = shard_values, shard_bloom
            voff = len(vfile)
            vfile += vblob
            h = bytes([vt])
            if vt == 2:  # dict: 1B count + varuint-len strings, NO bloom
                h += bytes([len(dictv)]) + b"".join(marshal_bytes(d)
                                                    for d in dictv)
                h += varuint(voff) + varuint(len(vblob))
            else:
                if vt == 3:  # uint8: 1B min + 1B max
                    h += bytes([mn, mx])
                boff = len(bfile)
                bloom = bloom_marshal([t for t in tokenize_unique(
                    [v.decode() for v in vals])])
                bfile += bloom
                h += varuint(voff) + varuint(len(vblob))
                h += varuint(boff) + varuint(len(bloom))
            cols.append((cname, h))

        if b.get("typed"):
            for cname, vt, raws, rows_enc, minmax in typed_columns():
                vblob = strings_block(rows_enc)
                voff = len(shard_values)
                shard_values.extend(vblob)
                bloom = bloom_marshal(
                    [t for t in tokenize_unique([r.decode() for r in raws])])
                boff = len(shard_bloom)
                shard_bloom.extend(bloom)
                h = bytes([vt]) + minmax
                h += varuint(voff) + varuint(len(vblob))
                h += varuint(boff) + varuint(len(bloom))
                cols.append((cname, h))

        # columnsHeader region (block_header.go:454-484): varuint count,
        # headers (offsets relative to region start), varuint const count,
        # const Field values (value-only for v1+)
        region = bytearray()
        refs = []
        region += varuint(len(cols))
        for cname, h in cols:
            refs.append((name_id[cname], len(region)))
            region += h
        const_cols = [("host", b["host"])]
        region += varuint(len(const_cols))
        crefs = []
        for cname, val in const_cols:
            crefs.append((name_id[cname], len(region)))
            region += marshal_bytes(val)

        # columnsHeaderIndex (block_header.go:275-347)
        idx = bytearray()
        idx += varuint(len(refs))
        for nid, off in refs:
            idx += varuint(nid) + varuint(off)
        idx += varuint(len(crefs))
        for nid, off in crefs:
            idx += varuint(nid) + varuint(off)

        chi_off, ch_off = len(chi_bin), len(ch_bin)
        chi_bin += idx
        ch_bin += region

        th = ts_headers[bi]
        bh = bytearray()
        # streamID: tenant (BE u32 acct, BE u32 proj) + u128 (BE hi, BE lo)
        bh += be32(0) + be32(0) + be64(1) + be64(2)
        bh += varuint(sum(len(v) for v in b["msg"]) + 8 * b["rows"])
        bh += varuint(b["rows"])
        bh += be64(th["offset"]) + be64(th["size"])
        bh += be64(th["min"]) + be64(th["max"])
        bh += bytes([th["marshal_type"]])
        bh += varuint(chi_off) + varuint(len(idx))
        bh += varuint(ch_off) + varuint(len(region))
        block_headers.append(bytes(bh))

    index_plain = b"".join(block_headers)
    index_bin = zstd_compress(index_plain, 1)

    # metaindex.bin: one indexBlockHeader (index_block_header.go:81-88)
    mih = (be32(0) + be32(0) + be64(1) + be64(2) +
           be64(ts_headers[0]["min"]) + be64(ts_headers[-1]["max"]) +
           be64(0) + be64(len(index_bin)))
    metaindex_bin = zstd_compress(mih, 1)

    files = {
        "metaindex.bin": metaindex_bin,
        "index.bin": index_bin,
        "columns_header_index.bin": bytes(chi_bin),
        "columns_header.bin": bytes(ch_bin),
        "column_names.bin": column_names_bin,
        "column_idxs.bin": column_idxs_bin,
        "message_values.bin": bytes(msg_values),
        "message_bloom.bin": bytes(msg_bloom),
        "values.bin0": bytes(shard_values),
        "bloom.bin0": bytes(shard_bloom),
        "timestamps.bin": ts_bin,
    }
    total_rows = sum(b["rows"] for b in blocks)
    meta = {
        "FormatVersion": 3,
        "CompressedSizeBytes": sum(len(v) for v in files.values()),
        "UncompressedSizeBytes": sum(
            sum(len(v) for v in b["msg"]) + 8 * b["rows"] for b in blocks),
        "RowsCount": total_rows,
        "BlocksCount": len(blocks),
        "MinTimestamp": ts_headers[0]["min"],
        "MaxTimestamp": ts_headers[-1]["max"],
        "BloomValuesShardsCount": 1,
    }
    for fname, data in files.items():
        with open(os.path.join(out_dir, fname), "wb") as f:
            f.write(data)
    with open(os.path.join(out_dir, "metadata.json"), "w") as f:
        json.dump(meta, f)
    print(f"wrote handmade part: {out_dir} ({total_rows} rows, "
          f"{len(blocks)} blocks)")


if __name__ == "__main__":
    out = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "tests", "golden", "handmade_part")
    build_part(out)
