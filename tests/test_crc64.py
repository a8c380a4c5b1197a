"""crc64 restatement pinned against the reference's own crc.cpp compiled
standalone (oracle/_ref/libcrc_ref.so; reference src/utils/crc.cpp:289-481)."""
import ctypes
import os
import random

import pytest

from conftest import ORACLE_SO, REF_CRC_SO
from pymodel import crc64 as py_crc64


def _orc():
    lib = ctypes.CDLL(ORACLE_SO)
    lib.orc_crc64.restype = ctypes.c_uint64
    lib.orc_crc64.argtypes = [ctypes.c_char_p, ctypes.c_uint64, ctypes.c_uint64]
    return lib


def test_crc64_vs_reference_build():
    if not os.path.exists(REF_CRC_SO):
        pytest.skip("oracle/_ref not built (reference tree absent and no prebuilt _ref)")
    orc = _orc()
    ref = ctypes.CDLL(REF_CRC_SO)
    f = getattr(ref, "_ZN3dsn5utils10crc64_calcEPKvmm")  # dsn::utils::crc64_calc
    f.restype = ctypes.c_uint64
    f.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_uint64]
    rnd = random.Random(20260915)
    for _ in range(500):
        n = rnd.randrange(0, 100)
        b = bytes(rnd.randrange(256) for _ in range(n))
        init = rnd.choice([0, 0xFFFFFFFFFFFFFFFF, rnd.getrandbits(64)])
        assert orc.orc_crc64(b, n, init) == f(b, n, init)


def test_crc64_vs_pymodel():
    orc = _orc()
    rnd = random.Random(7)
    for _ in range(100):
        n = rnd.randrange(0, 64)
        b = bytes(rnd.randrange(256) for _ in range(n))
        assert orc.orc_crc64(b, n, 0) == py_crc64(b)


def test_crc64_known_answers():
    """Golden vectors frozen from the reference build (stability anchor that
    travels to the GPU box where /root/reference is absent)."""
    orc = _orc()
    assert orc.orc_crc64(b"", 0, 0) == 0
    assert orc.orc_crc64(b"123456789", 9, 0) == 0xAE8B14860A799888
    assert orc.orc_crc64(b"hashkey", 7, 0) == py_crc64(b"hashkey")


def test_key_hash_partition_routing():
    """pegasus_key_hash + check_pegasus_key_hash (pegasus_key_schema.h:148-183):
    hklen>0 -> crc64(hashkey); hklen==0 -> crc64(rest)."""
    orc = _orc()
    orc.orc_key_hash.restype = ctypes.c_uint64
    orc.orc_key_hash.argtypes = [ctypes.c_char_p, ctypes.c_uint64]
    import struct

    hk = b"u:00000000000042"
    raw = struct.pack(">H", len(hk)) + hk + b"sort"
    assert orc.orc_key_hash(raw, len(raw)) == py_crc64(hk)
    raw0 = struct.pack(">H", 0) + b"justsort"
    assert orc.orc_key_hash(raw0, len(raw0)) == py_crc64(b"justsort")


def test_crc64_vs_committed_golden_vectors():
    """tests/golden/crc64_golden.json: vectors generated from the reference's
    own crc.cpp (oracle/_ref) in the dev container; this check runs anywhere,
    including GPU boxes where /root/reference does not exist."""
    import json

    path = os.path.join(os.path.dirname(__file__), "golden", "crc64_golden.json")
    cases = json.load(open(path))["cases"]
    assert len(cases) > 100
    orc = _orc()
    for c in cases:
        b = bytes.fromhex(c["data"])
        assert orc.orc_crc64(b, len(b), c["init"]) == c["crc"]
