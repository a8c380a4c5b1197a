"""Property-based fuzz (hypothesis) of the oracle's primitive surface
against the independent Python model and the data-layer codec — wider input
coverage than the fixed KAT vectors (tests/test_crc64.py,
tests/test_codec_kats.py) for the same reference semantics:

- crc64 (src/utils/crc.cpp:289-481): chunk-splitting invariant + model parity
- pegasus_key_hash (pegasus_key_schema.h:148-165)
- value schema v0/v1/v2 expire-ts round trips (pegasus_value_schema.h:58-125)
- compaction-rule pattern matching (string_pattern_match,
  src/server/compaction_filter_rule.cpp:31-53)
- generate_key/restore_key round trip (pegasus_key_schema.h:35-74)
"""
import ctypes

import pytest
from hypothesis import given, settings, strategies as st

from incubator_pegasus_amd import data as D
from tests import pymodel

BYTES = st.binary(min_size=0, max_size=300)
SMALL = st.binary(min_size=0, max_size=64)


@pytest.fixture(scope="module")
def hooks():
    import os
    lib = ctypes.CDLL(os.path.join(os.path.dirname(__file__), "..", "oracle",
                                   "liboracle.so"))
    lib.orc_crc64.restype = ctypes.c_uint64
    lib.orc_crc64.argtypes = [ctypes.c_char_p, ctypes.c_uint64, ctypes.c_uint64]
    lib.orc_key_hash.restype = ctypes.c_uint64
    lib.orc_key_hash.argtypes = [ctypes.c_char_p, ctypes.c_uint64]
    lib.orc_extract_expire_ts.restype = ctypes.c_uint32
    lib.orc_extract_expire_ts.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint64]
    lib.orc_ts_expired.restype = ctypes.c_int
    lib.orc_ts_expired.argtypes = [ctypes.c_uint32, ctypes.c_uint32]
    lib.orc_pattern_match.restype = ctypes.c_int
    lib.orc_pattern_match.argtypes = [ctypes.c_char_p, ctypes.c_uint64, ctypes.c_int,
                                      ctypes.c_char_p, ctypes.c_uint64]
    return lib


@settings(max_examples=300, deadline=None)
@given(BYTES)
def test_crc64_model_parity(hooks, data):
    assert hooks.orc_crc64(data, len(data), 0) == pymodel.crc64(data)


@settings(max_examples=200, deadline=None)
@given(BYTES, st.integers(min_value=0, max_value=299))
def test_crc64_chunk_splitting(hooks, data, cut):
    """crc64(a+b) == crc64(b, init=crc64(a)) — the reference's streaming use."""
    cut = min(cut, len(data))
    a, b = data[:cut], data[cut:]
    whole = hooks.orc_crc64(data, len(data), 0)
    part = hooks.orc_crc64(a, len(a), 0)
    assert hooks.orc_crc64(b, len(b), part) == whole


@settings(max_examples=200, deadline=None)
@given(SMALL, SMALL)
def test_key_roundtrip_and_hash(hooks, hk, sk):
    key = D.generate_key(hk, sk)
    rhk, rsk = D.restore_key(key)
    assert (rhk, rsk) == (hk, sk)
    assert hooks.orc_key_hash(key, len(key)) == pymodel.crc64(hk if hk else key[2:])


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=0, max_value=2), st.integers(min_value=0, max_value=2**32 - 1),
       SMALL, st.integers(min_value=0, max_value=2**64 - 1))
def test_value_expire_roundtrip(hooks, ver, ts, body, timetag):
    val = D.encode_value(body, ts, timetag, ver)
    assert hooks.orc_extract_expire_ts(ver, val, len(val)) == ts
    rts, rtag, user = D.decode_value(val, ver)
    assert (rts, user) == (ts, body)
    if ver >= 1:
        assert rtag == timetag


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=0, max_value=2**32 - 1), st.integers(min_value=0, max_value=2**32 - 1))
def test_expired_rule(hooks, now, ts):
    """check_if_ts_expired: expired iff ts > 0 and ts <= now
    (pegasus_value_schema.h:113-125)."""
    assert hooks.orc_ts_expired(now, ts) == (1 if (ts > 0 and ts <= now) else 0)


@settings(max_examples=300, deadline=None)
@given(SMALL, SMALL, st.integers(min_value=0, max_value=2))
def test_pattern_match(hooks, v, pat, ft):
    """string_pattern_match (compaction_filter_rule.cpp:31-53), the
    compaction-rule matcher: SMT anywhere/prefix/postfix; empty pattern or
    value shorter than pattern never matches."""
    got = hooks.orc_pattern_match(v, len(v), ft, pat, len(pat))
    if len(pat) == 0 or len(v) < len(pat):
        exp = False
    elif ft == 0:
        exp = pat in v
    elif ft == 1:
        exp = v.startswith(pat)
    else:
        exp = v.endswith(pat)
    assert got == (1 if exp else 0)
