"""C-ABI checks that run without a GPU: the product library must load and
export every symbol include/vlogsql.h declares, and filter compilation
(host-side) must work/fail as specified."""

import ctypes
import os
import re

import pytest

from tests.conftest import ROOT, FILTERS, TYPED_FILTERS
from victorialogs_amd.api import lib_path, load_product


def test_product_library_exports_header_symbols():
    lib = load_product()
    header = open(os.path.join(ROOT, "include", "vlogsql.h")).read()
    symbols = re.findall(r"\b(vql_\w+)\s*\(", header)
    assert symbols, "no symbols found in header?"
    for sym in set(symbols):
        assert hasattr(lib, sym), f"libvlogsql.so does not export {sym}"


def test_product_library_is_in_tree():
    # the .so must live in-tree so it travels with the gpurun snapshot
    assert os.path.dirname(lib_path()).startswith(ROOT)
    assert os.path.exists(lib_path())


def test_filter_compile_on_cpu():
    lib = load_product()
    for f in FILTERS + TYPED_FILTERS:
        h = lib.vql_compile_filter(f.encode())
        assert h, f"filter failed to compile: {f}: {lib.vql_errstr().decode()}"
        lib.vql_free_filter(h)


def test_filter_compile_errors():
    lib = load_product()
    bad = [
        '{"type":"phrase"}',                       # missing fields
        '{"type":"wat","field":"x","phrase":"y"}', # unknown type
        'not json at all',
        '{"type":"regexp","field":"x","re":"a\\\\p{L}"}',  # unicode class
        '{"type":"regexp","field":"x","re":"a{2000}"}', # repeat count too big
        '{"type":"regexp","field":"x","re":"(a|^b)c"}', # mid-pattern anchor
    ]
    for f in bad:
        h = lib.vql_compile_filter(f.encode())
        assert not h, f"expected compile failure for {f}"
        assert lib.vql_errstr().decode()


def test_scan_requires_gpu_fails_loudly(gen_part):
    """On a box without a GPU the product scan path must error, not silently
    fall back to CPU."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; the loud-failure path is for CPU boxes")
    lib = load_product()
    part = lib.vql_open_part(gen_part.encode())
    assert part
    filt = lib.vql_compile_filter(b'{"type":"phrase","field":"_msg","phrase":"x"}')
    assert filt
    stage = lib.vql_stage(part, filt, 0, 0, -1)
    assert not stage, "staging must fail without a GPU"
    assert "HIP" in lib.vql_errstr().decode() or "hip" in lib.vql_errstr().decode()
    lib.vql_free_filter(filt)
    lib.vql_close_part(part)


def test_part_metadata_via_product(gen_part):
    lib = load_product()
    part = lib.vql_open_part(gen_part.encode())
    assert part
    assert lib.vql_part_rows(part) == 30000
    nb = lib.vql_part_blocks(part)
    assert nb == 8
    assert sum(lib.vql_block_rows(part, i) for i in range(nb)) == 30000
    lib.vql_close_part(part)


def test_error_unsupported_classification():
    """vql_error_unsupported distinguishes valid-LogsQL-outside-the-class
    rejections (shim falls back to the Go path) from
    corruption/IO errors."""
    lib = load_product()
    lib.vql_error_unsupported.restype = ctypes.c_int
    # residual regex class: loud reject, classified unsupported
    f = lib.vql_compile_filter(
        b'{"type":"regexp","field":"x","re":"\\\\p{L}x"}')
    assert not f
    assert b"regex" in lib.vql_errstr()
    assert lib.vql_error_unsupported() == 1
    # malformed input: error, NOT classified unsupported
    f = lib.vql_compile_filter(b'{"type":"nope"}')
    assert not f
    assert lib.vql_error_unsupported() == 0
    # supported since round 2: these must compile
    for ok in [b'{"type":"regexp","field":"x","re":"^01|04$"}',
               b'{"type":"regexp","field":"x","re":"(?i)foo"}',
               b'{"type":"regexp","field":"x","re":"ab+?"}',
               b'{"type":"regexp","field":"x","re":"\\\\bfoo\\\\b"}']:
        h = lib.vql_compile_filter(ok)
        assert h, lib.vql_errstr()
        lib.vql_free_filter(h)
