"""Checkpoint/restore (§8(f)2): serialize runs to checkpoint.{decree}
directories (reference naming: pegasus_server_impl.cpp:1951-2137) and restore
into a fresh handle — including cross-backend (oracle checkpoint -> HIP
engine restore and back)."""
import ctypes
import random

import pytest

from incubator_pegasus_amd import data as D
from incubator_pegasus_amd.capi import OK


def _bind(lib):
    L = lib._lib
    L.rrdb_checkpoint.restype = ctypes.c_int32
    L.rrdb_checkpoint.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64]
    L.rrdb_restore.restype = ctypes.c_int32
    L.rrdb_restore.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint64]
    return L


def _fill(part, rnd, now=1000):
    keys = []
    seq = 1
    for run in range(3):
        recs = {}
        for i in range(40):
            k = D.generate_key(f"ck{rnd.randrange(30):02d}".encode(),
                               f"s{rnd.randrange(4)}".encode())
            kind = 1 if rnd.random() < 0.1 else 0
            v = b"\x00" * 12 if kind else D.encode_value(f"r{run}i{i}".encode(), 0, 0, 1)
            recs[k] = (v, kind)
        records = []
        for k in sorted(recs):
            v, kind = recs[k]
            records.append((k, v, seq, kind))
            keys.append(k)
            seq += 1
        part.ingest_run(records)
    return sorted(set(keys))


def _same_reads(a, b, keys, now=1000):
    for k in keys:
        assert a.get(k, now) == b.get(k, now), k
    ra = a.scan_open(b"\x00\x00", b"\xff\xff", now, validate_partition_hash=False)
    rb = b.scan_open(b"\x00\x00", b"\xff\xff", now, validate_partition_hash=False)
    assert ra.kvs == rb.kvs


def test_checkpoint_restore_oracle(oracle_lib, tmp_path):
    rnd = random.Random(1)
    src = oracle_lib.open(1, 0, -1)
    dst = oracle_lib.open(1, 0, -1)
    try:
        keys = _fill(src, rnd)
        L = _bind(oracle_lib)
        assert L.rrdb_checkpoint(src._h, str(tmp_path).encode(), 42) == OK
        assert L.rrdb_restore(dst._h, str(tmp_path).encode(), 42) == OK
        assert dst.num_runs() == src.num_runs()
        assert dst.num_records() == src.num_records()
        _same_reads(src, dst, keys)
        # restore into a non-empty handle is refused
        assert L.rrdb_restore(dst._h, str(tmp_path).encode(), 42) != OK
        # both sides can continue: compact and keep reading identically
        assert src.manual_compact(1000)[0] == OK
        assert dst.manual_compact(1000)[0] == OK
        _same_reads(src, dst, keys)
    finally:
        src.close()
        dst.close()


def test_checkpoint_missing_decree(oracle_lib, tmp_path):
    p = oracle_lib.open(1, 0, -1)
    try:
        L = _bind(oracle_lib)
        assert L.rrdb_restore(p._h, str(tmp_path).encode(), 7) != OK
    finally:
        p.close()


@pytest.mark.gpu
def test_checkpoint_cross_backend(oracle_lib, hip_lib, tmp_path):
    rnd = random.Random(2)
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    g2 = hip_lib.open(1, 0, 0)
    o2 = oracle_lib.open(1, 0, -1)
    try:
        keys = _fill(o, rnd)
        Lo, Lg = _bind(oracle_lib), _bind(hip_lib)
        # oracle checkpoint -> HIP restore
        assert Lo.rrdb_checkpoint(o._h, str(tmp_path).encode(), 1) == OK
        assert Lg.rrdb_restore(g._h, str(tmp_path).encode(), 1) == OK
        _same_reads(o, g, keys)
        # HIP checkpoint -> both restore
        assert Lg.rrdb_checkpoint(g._h, str(tmp_path).encode(), 2) == OK
        assert Lg.rrdb_restore(g2._h, str(tmp_path).encode(), 2) == OK
        assert Lo.rrdb_restore(o2._h, str(tmp_path).encode(), 2) == OK
        _same_reads(g, g2, keys)
        _same_reads(g2, o2, keys)
    finally:
        for p in (o, g, g2, o2):
            p.close()
