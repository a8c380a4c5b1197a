"""Write path (§8(f)1): put/remove -> memtable -> flush -> sorted run,
mirroring pegasus_write_service put/remove + rocksdb_wrapper write_batch
semantics (reference pegasus_write_service.h:119-207,
rocksdb_wrapper.cpp:121-247).  Reads see committed writes immediately
(lazy flush = the rocksdb memtable read path)."""
import ctypes

import pytest

from incubator_pegasus_amd import data as D
from incubator_pegasus_amd.capi import NOT_FOUND, OK, SCAN_COMPLETED


def _bind_write(lib):
    L = lib._lib
    L.rrdb_put.restype = ctypes.c_int32
    L.rrdb_put.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64, ctypes.c_char_p,
                           ctypes.c_uint64, ctypes.c_char_p, ctypes.c_uint64, ctypes.c_uint32]
    L.rrdb_remove.restype = ctypes.c_int32
    L.rrdb_remove.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64,
                              ctypes.c_char_p, ctypes.c_uint64]
    L.rrdb_flush.restype = ctypes.c_int32
    L.rrdb_flush.argtypes = [ctypes.c_void_p]
    L.rrdb_memtable_entries.restype = ctypes.c_uint64
    L.rrdb_memtable_entries.argtypes = [ctypes.c_void_p]
    return L


def put(lib, part, hk, sk, val, expire=0):
    assert _bind_write(lib).rrdb_put(part._h, hk, len(hk), sk, len(sk), val, len(val),
                                     expire) == OK


def remove(lib, part, hk, sk):
    assert _bind_write(lib).rrdb_remove(part._h, hk, len(hk), sk, len(sk)) == OK


def flush(lib, part):
    assert _bind_write(lib).rrdb_flush(part._h) == OK


def entries(lib, part):
    return _bind_write(lib).rrdb_memtable_entries(part._h)


def _exercise(lib, part, now=1000):
    # writes visible without explicit flush
    put(lib, part, b"whk", b"a", b"v1")
    put(lib, part, b"whk", b"b", b"v2", expire=now + 50)
    put(lib, part, b"whk", b"c", b"v3", expire=now - 1)  # already expired
    assert part.get(D.generate_key(b"whk", b"a"), now) == (OK, b"v1")
    assert part.ttl(D.generate_key(b"whk", b"b"), now) == (OK, 50)
    assert part.get(D.generate_key(b"whk", b"c"), now)[0] == NOT_FOUND
    # overwrite in a later memtable generation wins
    put(lib, part, b"whk", b"a", b"v1-new")
    assert part.get(D.generate_key(b"whk", b"a"), now) == (OK, b"v1-new")
    # remove hides, compaction drops
    remove(lib, part, b"whk", b"b")
    assert part.get(D.generate_key(b"whk", b"b"), now)[0] == NOT_FOUND
    st, cnt = part.sortkey_count(b"whk", now)
    assert (st, cnt) == (OK, 1)  # only "a" live ("b" removed, "c" expired)
    err, stats = part.manual_compact(now)
    assert err == OK
    assert stats.output_records == 1 and stats.tombstones >= 1 and stats.expired == 1
    res = part.scan_open(b"\x00\x00", b"\xff\xff", now, validate_partition_hash=False)
    assert [(k, v) for k, v in res.kvs] == [(D.generate_key(b"whk", b"a"), b"v1-new")]
    assert res.context_id == SCAN_COMPLETED
    # memtable upsert keeps one entry per key
    put(lib, part, b"u", b"s", b"x")
    put(lib, part, b"u", b"s", b"y")
    assert entries(lib, part) in (1, 2)  # engine upserts (1); oracle logs (2)
    flush(lib, part)
    assert entries(lib, part) == 0
    assert part.get(D.generate_key(b"u", b"s"), now) == (OK, b"y")
    # writes interleave with pre-built ingested runs by seqno
    part.ingest_run([(D.generate_key(b"zing", b""), D.encode_value(b"iv", 0, 0, 1),
                      10_000_000, 0)])
    put(lib, part, b"zing", b"", b"overwrites-ingested")
    assert part.get(D.generate_key(b"zing", b""), now) == (OK, b"overwrites-ingested")


def test_write_path_oracle(oracle_lib):
    p = oracle_lib.open(1, 0, -1)
    try:
        _exercise(oracle_lib, p)
    finally:
        p.close()


@pytest.mark.gpu
def test_write_path_hip(hip_lib):
    p = hip_lib.open(1, 0, 0)
    try:
        _exercise(hip_lib, p)
    finally:
        p.close()


@pytest.mark.gpu
def test_write_path_parity(oracle_lib, hip_lib):
    import random

    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        rnd = random.Random(42)
        now = 1000
        keys = []
        for i in range(300):
            hk = f"wh{rnd.randrange(20):02d}".encode()
            sk = f"s{rnd.randrange(10):02d}".encode()
            keys.append(D.generate_key(hk, sk))
            if rnd.random() < 0.15:
                for lib, part in ((oracle_lib, o), (hip_lib, g)):
                    remove(lib, part, hk, sk)
            else:
                val = f"v{i}".encode()
                exp = 0 if rnd.random() < 0.8 else now + rnd.randrange(1, 100)
                for lib, part in ((oracle_lib, o), (hip_lib, g)):
                    put(lib, part, hk, sk, val, exp)
            if rnd.random() < 0.05:
                flush(oracle_lib, o)
                flush(hip_lib, g)
        for k in sorted(set(keys)):
            assert o.get(k, now) == g.get(k, now), k
        eo, so = o.manual_compact(now)
        eg, sg = g.manual_compact(now)
        assert eo == eg == OK and so == sg
        ro = o.scan_open(b"\x00\x00", b"\xff\xff", now, validate_partition_hash=False)
        rg = g.scan_open(b"\x00\x00", b"\xff\xff", now, validate_partition_hash=False)
        assert ro.kvs == rg.kvs
    finally:
        o.close()
        g.close()
