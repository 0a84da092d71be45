"""RFC3339 inputs without a timezone suffix use the HOST LOCAL timezone,
matching the reference's TryParseTimestampRFC3339Nano →
timeutil.GetLocalTimezoneOffsetNsecs (vendor/.../lib/timeutil/timezone.go:9-19,
sampled from the current time).  ADVICE r01: TZ-varied coverage.

Each case runs in a subprocess with a different TZ because the offset is
cached per process (as the reference caches it)."""

import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)

SNIPPET = r"""
import ctypes, sys
lib = ctypes.CDLL(sys.argv[1])
fn = lib.orc_parse_math_number
fn.restype = ctypes.c_double
fn.argtypes = [ctypes.c_char_p, ctypes.c_long]
s = b"2023-06-15T12:00:00"
print(int(fn(s, len(s))))
"""

ROWOPS_SNIPPET = r"""
import ctypes, sys
lib = ctypes.CDLL(sys.argv[1])
fn = lib.h_dev_parse_math
fn.restype = ctypes.c_double
fn.argtypes = [ctypes.c_char_p, ctypes.c_long]
s = b"2023-06-15T12:00:00"
print(int(fn(s, len(s))))
"""


def run_with_tz(tz, snippet, libname):
    env = dict(os.environ, TZ=tz)
    out = subprocess.run(
        [sys.executable, "-c", snippet, os.path.join(ROOT, libname)],
        capture_output=True, text=True, env=env, check=True)
    return int(out.stdout.strip())


UTC_NSECS = 1686830400 * 10**9  # 2023-06-15T12:00:00Z


def test_oracle_local_tz_offset():
    # POSIX TZ "XXX-3" = UTC+3 (no DST): local noon is 09:00 UTC
    assert run_with_tz("UTC0", SNIPPET, "oracle/liboracle.so") == UTC_NSECS
    assert (run_with_tz("XXX-3", SNIPPET, "oracle/liboracle.so")
            == UTC_NSECS - 3 * 3600 * 10**9)
    # west-of-UTC zone: TZ "XXX5" = UTC-5
    assert (run_with_tz("XXX5", SNIPPET, "oracle/liboracle.so")
            == UTC_NSECS + 5 * 3600 * 10**9)


def test_device_mirror_local_tz_offset():
    # the host build of the exact per-row device code follows the same rule
    lib = "tools/host_rowops/librowops.so"
    assert run_with_tz("UTC0", ROWOPS_SNIPPET, lib) == UTC_NSECS
    assert (run_with_tz("XXX-3", ROWOPS_SNIPPET, lib)
            == UTC_NSECS - 3 * 3600 * 10**9)


def test_explicit_suffix_ignores_local_tz():
    # a 'Z' or explicit offset must not consult the local zone
    snippet = SNIPPET.replace('b"2023-06-15T12:00:00"',
                              'b"2023-06-15T12:00:00Z"')
    assert run_with_tz("XXX-3", snippet, "oracle/liboracle.so") == UTC_NSECS
    snippet = SNIPPET.replace('b"2023-06-15T12:00:00"',
                              'b"2023-06-15T14:00:00+02:00"')
    assert run_with_tz("XXX5", snippet, "oracle/liboracle.so") == UTC_NSECS
