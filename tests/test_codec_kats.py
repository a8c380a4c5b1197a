"""Value/key codec KATs lifted from the reference's own tests:
  - src/base/test/value_schema_test.cpp:73-136 (generate_and_extract,
    update_expire_ts over versions 0/1/2)
  - key layout: src/base/pegasus_key_schema.h:35-122
Checked against both the Python codec (product host helpers) and the oracle's
C codec hooks — two independent restatements must agree on every case."""
import ctypes

from conftest import ORACLE_SO
from incubator_pegasus_amd import data as D

U32_MAX = 0xFFFFFFFF
U64_MAX = 0xFFFFFFFFFFFFFFFF

# test table from value_schema_test.cpp:76-100
GENERATE_EXTRACT_CASES = [
    (0, 1000, 0, b""),
    (0, U32_MAX, 0, b"pegasus"),
    (0, U32_MAX, 0, b""),
    (0, 0, 0, b"a"),
    (1, 1000, 10001, b""),
    (1, U32_MAX, U64_MAX, b"pegasus"),
    (1, U32_MAX, U64_MAX, b""),
    (1, 0, 0, b"a"),
    (2, 1000, 10001, b""),
    (2, U32_MAX, U64_MAX, b"pegasus"),
    (2, U32_MAX, U64_MAX, b""),
    (2, 0, 0, b"a"),
]


def _orc():
    lib = ctypes.CDLL(ORACLE_SO)
    lib.orc_extract_expire_ts.restype = ctypes.c_uint32
    lib.orc_extract_expire_ts.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint64]
    lib.orc_update_expire_ts.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint32]
    lib.orc_ts_expired.restype = ctypes.c_int
    lib.orc_ts_expired.argtypes = [ctypes.c_uint32, ctypes.c_uint32]
    return lib


def test_generate_and_extract():
    orc = _orc()
    for ver, expire_ts, timetag, user_data in GENERATE_EXTRACT_CASES:
        raw = D.encode_value(user_data, expire_ts, timetag, ver)
        e, t, u = D.decode_value(raw, ver)
        assert e == expire_ts
        if ver >= 1:
            assert t == timetag
        assert u == user_data
        assert orc.orc_extract_expire_ts(ver, raw, len(raw)) == expire_ts
        assert len(raw) == D.value_header_len(ver) + len(user_data)


def test_update_expire_ts():
    """value_schema_test.cpp:109-136: {ver, 1000 -> 10086}"""
    orc = _orc()
    for ver in (0, 1, 2):
        raw = bytearray(D.encode_value(b"", 1000, 0, ver))
        buf = (ctypes.c_char * len(raw)).from_buffer(raw)
        orc.orc_update_expire_ts(ver, buf, 10086)
        assert D.decode_value(bytes(raw), ver)[0] == 10086


def test_v2_magic_byte():
    """value_schema_v2.cpp:88: leading byte 0x80|2"""
    raw = D.encode_value(b"x", 5, 7, 2)
    assert raw[0] == 0x82


def test_check_if_ts_expired():
    """pegasus_value_schema.h:113-116: expired iff ts>0 and ts<=now"""
    orc = _orc()
    assert orc.orc_ts_expired(100, 0) == 0
    assert orc.orc_ts_expired(100, 100) == 1
    assert orc.orc_ts_expired(100, 101) == 0
    assert orc.orc_ts_expired(100, 1) == 1


def test_key_roundtrip_and_next_blob():
    """pegasus_key_schema.h:41-122"""
    for hk, sk in [(b"h", b""), (b"hash", b"sort"), (b"", b"s"), (b"a" * 300, b"b")]:
        raw = D.generate_key(hk, sk)
        assert raw[:2] == len(hk).to_bytes(2, "big")
        h2, s2 = D.restore_key(raw)
        assert (h2, s2) == (hk, sk)
    # next_blob: increment last non-FF byte, truncate FF tail
    assert D.generate_next_blob(b"ab") == D.generate_key(b"ab")[:-1] + b"c"
    assert D.generate_next_blob(b"a\xff") == b"\x00\x02b"
    assert D.generate_next_blob(b"a", b"b\xff\xff") == b"\x00\x01ac"


def test_sort_order_matches_bytewise():
    """fixed-width decimal hashkeys sort bytewise == numerically"""
    import numpy as np

    ids = np.array([0, 1, 9, 10, 99, 12345, 10**13], dtype=np.uint64)
    raw = D.make_raw_keys(ids)
    keys = [bytes(raw[i]) for i in range(len(ids))]
    assert keys == sorted(keys)
