"""ctypes bindings for the product (HIP) and oracle (CPU) C ABIs.

The product library is the hot path (include/vlogsql.h); the oracle library
is TEST INFRASTRUCTURE only — used by tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg (see oracle/oracle_filter.h).
"""

import ctypes
import os

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def lib_path() -> str:
    # VQL_LIB overrides for A/B kernel experiments (same box, two builds)
    return os.environ.get("VQL_LIB") or os.path.join(
        _ROOT, "victorialogs_amd", "libvlogsql.so")


def oracle_lib_path() -> str:
    return os.path.join(_ROOT, "oracle", "liboracle.so")


_product = None
_oracle = None


def load_product() -> ctypes.CDLL:
    """Loads the HIP scan library.  Raises if missing — no CPU fallback."""
    global _product
    if _product is not None:
        return _product
    path = lib_path()
    if not os.path.exists(path):
        raise RuntimeError(
            f"HIP extension missing: {path}. Run `make hip` (hipcc, gfx950); "
            "the product path has no CPU fallback."
        )
    lib = ctypes.CDLL(path)
    lib.vql_errstr.restype = ctypes.c_char_p
    lib.vql_open_part.restype = ctypes.c_void_p
    lib.vql_open_part.argtypes = [ctypes.c_char_p]
    lib.vql_close_part.argtypes = [ctypes.c_void_p]
    lib.vql_part_blocks.restype = ctypes.c_long
    lib.vql_part_blocks.argtypes = [ctypes.c_void_p]
    lib.vql_part_rows.restype = ctypes.c_longlong
    lib.vql_part_rows.argtypes = [ctypes.c_void_p]
    lib.vql_block_rows.restype = ctypes.c_long
    lib.vql_block_rows.argtypes = [ctypes.c_void_p, ctypes.c_long]
    lib.vql_compile_filter.restype = ctypes.c_void_p
    lib.vql_compile_filter.argtypes = [ctypes.c_char_p]
    lib.vql_free_filter.argtypes = [ctypes.c_void_p]
    lib.vql_stage.restype = ctypes.c_void_p
    lib.vql_stage.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                              ctypes.c_long, ctypes.c_long]
    lib.vql_stage_free.argtypes = [ctypes.c_void_p]
    lib.vql_stage_bytes.restype = ctypes.c_longlong
    lib.vql_stage_bytes.argtypes = [ctypes.c_void_p]
    lib.vql_stage_algo_bytes.restype = ctypes.c_longlong
    lib.vql_stage_algo_bytes.argtypes = [ctypes.c_void_p]
    lib.vql_stage_rows.restype = ctypes.c_longlong
    lib.vql_stage_rows.argtypes = [ctypes.c_void_p]
    lib.vql_stage_live_rows.restype = ctypes.c_longlong
    lib.vql_stage_live_rows.argtypes = [ctypes.c_void_p]
    lib.vql_scan_staged.restype = ctypes.c_longlong
    lib.vql_scan_staged.argtypes = [ctypes.c_void_p]
    lib.vql_last_kernel_ms.restype = ctypes.c_double
    lib.vql_last_kernel_ms.argtypes = [ctypes.c_void_p]
    lib.vql_fetch_bitmaps.restype = ctypes.c_int
    lib.vql_fetch_bitmaps.argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(ctypes.c_ulonglong),
                                      ctypes.c_longlong]
    lib.vql_scan_batch.restype = ctypes.c_longlong
    lib.vql_scan_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long,
                                   ctypes.c_long,
                                   ctypes.POINTER(ctypes.c_ulonglong),
                                   ctypes.c_longlong,
                                   ctypes.POINTER(ctypes.c_ulonglong)]
    lib.vql_fetch_block_hits.restype = ctypes.c_int
    lib.vql_fetch_block_hits.argtypes = [ctypes.c_void_p,
                                         ctypes.POINTER(ctypes.c_ulonglong),
                                         ctypes.c_longlong]
    lib.vql_gather_sizes.restype = ctypes.c_int
    lib.vql_gather_sizes.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                     ctypes.POINTER(ctypes.c_ulonglong),
                                     ctypes.POINTER(ctypes.c_ulonglong)]
    lib.vql_gather.restype = ctypes.c_longlong
    lib.vql_gather.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                               ctypes.POINTER(ctypes.c_ubyte), ctypes.c_longlong,
                               ctypes.POINTER(ctypes.c_ulonglong), ctypes.c_longlong,
                               ctypes.POINTER(ctypes.c_ulonglong)]
    _product = lib
    return lib


def _load_oracle() -> ctypes.CDLL:
    global _oracle
    if _oracle is not None:
        return _oracle
    path = oracle_lib_path()
    if not os.path.exists(path):
        raise RuntimeError(f"oracle library missing: {path}. Run `make oracle`.")
    lib = ctypes.CDLL(path)
    lib.orc_errstr.restype = ctypes.c_char_p
    lib.orc_open_part.restype = ctypes.c_void_p
    lib.orc_open_part.argtypes = [ctypes.c_char_p]
    lib.orc_close_part.argtypes = [ctypes.c_void_p]
    lib.orc_block_count.restype = ctypes.c_long
    lib.orc_block_count.argtypes = [ctypes.c_void_p]
    lib.orc_block_rows.restype = ctypes.c_long
    lib.orc_block_rows.argtypes = [ctypes.c_void_p, ctypes.c_long]
    lib.orc_part_rows.restype = ctypes.c_long
    lib.orc_part_rows.argtypes = [ctypes.c_void_p]
    lib.orc_compile_filter.restype = ctypes.c_void_p
    lib.orc_compile_filter.argtypes = [ctypes.c_char_p]
    lib.orc_free_filter.argtypes = [ctypes.c_void_p]
    lib.orc_scan_blocks.restype = ctypes.c_longlong
    lib.orc_scan_blocks.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long,
                                    ctypes.c_long,
                                    ctypes.POINTER(ctypes.c_ulonglong),
                                    ctypes.c_longlong, ctypes.c_int]
    lib.orc_generate_part.restype = ctypes.c_longlong
    lib.orc_generate_part.argtypes = [ctypes.c_char_p, ctypes.c_ulonglong,
                                      ctypes.c_ulonglong, ctypes.c_ulonglong,
                                      ctypes.c_ulonglong, ctypes.c_ulonglong]
    lib.orc_write_custom_part.restype = ctypes.c_int
    lib.orc_write_custom_part.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
    lib.orc_xxhash64.restype = ctypes.c_ulonglong
    lib.orc_xxhash64.argtypes = [ctypes.c_char_p, ctypes.c_long]
    lib.orc_bloom_marshal_tokens.restype = ctypes.c_long
    lib.orc_bloom_marshal_tokens.argtypes = [ctypes.c_char_p, ctypes.c_char_p,
                                             ctypes.c_long]
    lib.orc_match_phrase.restype = ctypes.c_int
    lib.orc_match_phrase.argtypes = [ctypes.c_char_p, ctypes.c_long,
                                     ctypes.c_char_p, ctypes.c_long]
    lib.orc_tokenize.restype = ctypes.c_long
    lib.orc_tokenize.argtypes = [ctypes.c_char_p, ctypes.c_long, ctypes.c_char_p,
                                 ctypes.c_long]
    lib.orc_format_float64.restype = ctypes.c_long
    lib.orc_format_float64.argtypes = [ctypes.c_double, ctypes.c_char_p,
                                       ctypes.c_long]
    lib.orc_format_iso8601.restype = ctypes.c_long
    lib.orc_format_iso8601.argtypes = [ctypes.c_longlong, ctypes.c_char_p,
                                       ctypes.c_long]
    lib.orc_parse_iso8601.restype = ctypes.c_int
    lib.orc_parse_iso8601.argtypes = [ctypes.c_char_p, ctypes.c_long,
                                      ctypes.POINTER(ctypes.c_longlong)]
    _oracle = lib
    return lib


def generate_part(dir_path, rows, streams=1, rows_per_block=8192, msg_len=256,
                  seed=1):
    """Generates a reference-format part with vlogsgenerator-shaped rows
    (app/vlogsgenerator/main.go:234-297 shapes).  Returns total _msg bytes."""
    lib = _load_oracle()
    r = lib.orc_generate_part(dir_path.encode(), rows, streams, rows_per_block,
                              msg_len, seed)
    if r < 0:
        raise RuntimeError(lib.orc_errstr().decode())
    return r


def write_custom_part(dir_path, spec):
    """Writes a part from a JSON fixture spec (test infrastructure; mirrors
    the filter_test.go:158-277 fixture pattern)."""
    import json as _json
    lib = _load_oracle()
    if lib.orc_write_custom_part(dir_path.encode(), _json.dumps(spec).encode()) != 0:
        raise RuntimeError(lib.orc_errstr().decode())


def oracle_helpers():
    """Raw oracle lib for golden-vector tests."""
    return _load_oracle()


class OracleScanner:
    """TEST INFRASTRUCTURE — CPU oracle scans for parity checks and the
    cpu_baseline bench leg."""

    def __init__(self, part_dir):
        self.lib = _load_oracle()
        self.part = self.lib.orc_open_part(part_dir.encode())
        if not self.part:
            raise RuntimeError(self.lib.orc_errstr().decode())

    def close(self):
        if self.part:
            self.lib.orc_close_part(self.part)
            self.part = None

    @property
    def blocks(self):
        return self.lib.orc_block_count(self.part)

    def block_rows(self, i):
        return self.lib.orc_block_rows(self.part, i)

    def words_for(self, lo=0, hi=-1):
        if hi < 0:
            hi = self.blocks
        return sum((self.block_rows(i) + 63) // 64 for i in range(lo, hi))

    def scan(self, filter_json, lo=0, hi=-1, with_bitmaps=False, threads=1):
        f = self.lib.orc_compile_filter(filter_json.encode())
        if not f:
            raise RuntimeError(self.lib.orc_errstr().decode())
        try:
            if with_bitmaps:
                nwords = self.words_for(lo, hi if hi >= 0 else self.blocks)
                buf = (ctypes.c_ulonglong * max(nwords, 1))()
                hits = self.lib.orc_scan_blocks(self.part, f, lo, hi, buf, nwords,
                                                threads)
                if hits < 0:
                    raise RuntimeError(self.lib.orc_errstr().decode())
                return hits, bytes(buf)[: nwords * 8]
            hits = self.lib.orc_scan_blocks(self.part, f, lo, hi, None, 0, threads)
            if hits < 0:
                raise RuntimeError(self.lib.orc_errstr().decode())
            return hits, None
        finally:
            self.lib.orc_free_filter(f)


class Part:
    def __init__(self, part_dir):
        self.lib = load_product()
        self.h = self.lib.vql_open_part(part_dir.encode())
        if not self.h:
            raise RuntimeError(self.lib.vql_errstr().decode())

    def close(self):
        if self.h:
            self.lib.vql_close_part(self.h)
            self.h = None

    @property
    def blocks(self):
        return self.lib.vql_part_blocks(self.h)

    @property
    def rows(self):
        return self.lib.vql_part_rows(self.h)

    def block_rows(self, i):
        return self.lib.vql_block_rows(self.h, i)


class Filter:
    def __init__(self, filter_json):
        self.lib = load_product()
        self.h = self.lib.vql_compile_filter(filter_json.encode())
        if not self.h:
            raise RuntimeError(self.lib.vql_errstr().decode())

    def close(self):
        if self.h:
            self.lib.vql_free_filter(self.h)
            self.h = None


class Stage:
    """Staged (HBM-resident) scan context: decode-once, scan-many.

    `part` may be a single Part or a list of Parts — a multi-part stage scans
    every block of every part in ONE kernel launch (vql_stage_parts)."""

    def __init__(self, part, filt: Filter, device=0, lo=0, hi=-1):
        self.lib = load_product()
        if isinstance(part, (list, tuple)):
            arr = (ctypes.c_void_p * len(part))(*[p.h for p in part])
            self.lib.vql_stage_parts.restype = ctypes.c_void_p
            self.lib.vql_stage_parts.argtypes = [
                ctypes.POINTER(ctypes.c_void_p), ctypes.c_int,
                ctypes.c_void_p, ctypes.c_int]
            self.h = self.lib.vql_stage_parts(arr, len(part), filt.h, device)
        else:
            self.h = self.lib.vql_stage(part.h, filt.h, device, lo, hi)
        if not self.h:
            raise RuntimeError(self.lib.vql_errstr().decode())

    def close(self):
        if self.h:
            self.lib.vql_stage_free(self.h)
            self.h = None

    @property
    def staged_bytes(self):
        return self.lib.vql_stage_bytes(self.h)

    @property
    def algo_bytes(self):
        return self.lib.vql_stage_algo_bytes(self.h)

    @property
    def rows(self):
        return self.lib.vql_stage_rows(self.h)

    @property
    def live_rows(self):
        """Rows of blocks that reach the kernel (statically-pruned blocks
        are compacted out of the dispatch)."""
        return self.lib.vql_stage_live_rows(self.h)

    def scan(self):
        hits = self.lib.vql_scan_staged(self.h)
        if hits < 0:
            raise RuntimeError(self.lib.vql_errstr().decode())
        return hits

    @property
    def last_kernel_ms(self):
        return self.lib.vql_last_kernel_ms(self.h)

    def fetch_bitmaps(self, nwords):
        buf = (ctypes.c_ulonglong * max(nwords, 1))()
        if self.lib.vql_fetch_bitmaps(self.h, buf, nwords) != 0:
            raise RuntimeError(self.lib.vql_errstr().decode())
        return bytes(buf)[: nwords * 8]

    def gather(self, field, with_rowids=True):
        """Gathers matched rows' decoded values of `field` (blockResult
        materialization).  Returns (values: list[bytes], rowids)."""
        nrows = ctypes.c_ulonglong()
        nbytes = ctypes.c_ulonglong()
        if self.lib.vql_gather_sizes(self.h, field.encode(), ctypes.byref(nrows),
                                     ctypes.byref(nbytes)) != 0:
            raise RuntimeError(self.lib.vql_errstr().decode())
        n, b = nrows.value, nbytes.value
        bytes_buf = (ctypes.c_ubyte * max(b, 1))()
        offs_buf = (ctypes.c_ulonglong * (n + 1))()
        rowids_buf = (ctypes.c_ulonglong * max(n, 1))() if with_rowids else None
        r = self.lib.vql_gather(self.h, field.encode(), bytes_buf, b, offs_buf,
                                n + 1, rowids_buf)
        if r < 0:
            raise RuntimeError(self.lib.vql_errstr().decode())
        raw = bytes(bytes_buf)[:b]
        values = [raw[offs_buf[i]:offs_buf[i + 1]] for i in range(n)]
        rowids = list(rowids_buf)[:n] if with_rowids else None
        return values, rowids

    def fetch_block_hits(self, nblocks):
        """Per-block matched-row counts (`| stats count()` fast path)."""
        buf = (ctypes.c_ulonglong * max(nblocks, 1))()
        if self.lib.vql_fetch_block_hits(self.h, buf, nblocks) != 0:
            raise RuntimeError(self.lib.vql_errstr().decode())
        return list(buf)[:nblocks]


def gpu_bloom_build(values, device=0):
    """GPU ingest-side bloom build (vql_bloom_build): returns the marshaled
    bloom bytes for a list of value byte-strings, bit-identical to the CPU
    writer's column bloom."""
    lib = load_product()
    lib.vql_bloom_build.restype = ctypes.c_longlong
    data = b"".join(values)
    offs = [0]
    for v in values:
        offs.append(offs[-1] + len(v))
    offs_arr = (ctypes.c_uint32 * len(offs))(*offs)
    buf = ctypes.create_string_buffer(max(len(data) * 2 + 1024, 1 << 16))
    n = lib.vql_bloom_build(data, len(data), offs_arr, len(values), device,
                            ctypes.cast(buf, ctypes.POINTER(ctypes.c_ubyte)),
                            len(buf))
    if n < 0:
        raise RuntimeError(lib.vql_errstr().decode())
    if n > len(buf):
        raise RuntimeError("bloom output larger than buffer")
    return buf.raw[:n]
