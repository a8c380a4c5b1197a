"""ctypes binding for the rrdb engine C-ABI (include/rrdb_engine.h).

The same binding drives either backend .so:
  - incubator_pegasus_amd/csrc/librrdb_hip.so  (the product, MI355X/gfx950)
  - oracle/liboracle.so                        (CPU restatement; tests only)

Mirrors the reference's read-service request/response shapes
(reference src/server/pegasus_read_service.h:54-68, idl/rrdb.thrift:185-345).
"""
from __future__ import annotations

import ctypes as C
import os
from dataclasses import dataclass, field
from typing import Optional

# status codes (rocksdb::Status::Code, v8.5.3)
OK = 0
NOT_FOUND = 1
CORRUPTION = 2
INVALID_ARGUMENT = 4
INCOMPLETE = 7

# filter types (idl/rrdb.thrift:27-33)
FT_NO_FILTER = 0
FT_MATCH_ANYWHERE = 1
FT_MATCH_PREFIX = 2
FT_MATCH_POSTFIX = 3

KIND_PUT = 0
KIND_DELETE = 1

SCAN_COMPLETED = -1


class _CSlice(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)), ("len", C.c_uint64)]


class _Slice(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)), ("len", C.c_uint64)]


class _Result(C.Structure):
    _fields_ = [
        ("error", C.c_int32),
        ("count", C.c_uint64),
        ("context_id", C.c_int64),
        ("i64", C.c_int64),
        ("keys", C.POINTER(_Slice)),
        ("values", C.POINTER(_Slice)),
        ("expire_ts", C.POINTER(C.c_int32)),
        ("group_counts", C.POINTER(C.c_uint64)),
        ("group_errors", C.POINTER(C.c_int32)),
        ("dev_keys", C.c_void_p),
        ("dev_key_offs", C.c_void_p),
        ("dev_vals", C.c_void_p),
        ("dev_val_offs", C.c_void_p),
        ("_arena", C.c_void_p),
    ]


class _MultiGetRequest(C.Structure):
    _fields_ = [
        ("hash_key", _CSlice),
        ("start_sortkey", _CSlice),
        ("stop_sortkey", _CSlice),
        ("start_inclusive", C.c_uint8),
        ("stop_inclusive", C.c_uint8),
        ("max_kv_count", C.c_int32),
        ("max_kv_size", C.c_int32),
        ("no_value", C.c_uint8),
        ("reverse", C.c_uint8),
        ("sort_key_filter_type", C.c_int32),
        ("sort_key_filter_pattern", _CSlice),
        ("n_sort_keys", C.c_uint64),
        ("sort_keys", C.POINTER(C.c_uint8)),
        ("sort_key_offs", C.POINTER(C.c_uint64)),
        ("on_device_out", C.c_uint8),
    ]


class _ScanRequest(C.Structure):
    _fields_ = [
        ("start_key", _CSlice),
        ("stop_key", _CSlice),
        ("start_inclusive", C.c_uint8),
        ("stop_inclusive", C.c_uint8),
        ("batch_size", C.c_int32),
        ("no_value", C.c_uint8),
        ("hash_key_filter_type", C.c_int32),
        ("hash_key_filter_pattern", _CSlice),
        ("sort_key_filter_type", C.c_int32),
        ("sort_key_filter_pattern", _CSlice),
        ("full_scan", C.c_uint8),
        ("validate_partition_hash", C.c_uint8),
        ("return_expire_ts", C.c_uint8),
        ("only_return_count", C.c_uint8),
        ("on_device_out", C.c_uint8),
    ]


class _CompactOptions(C.Structure):
    _fields_ = [("target_level", C.c_int32), ("bottommost_force", C.c_uint8),
                ("keep_inputs", C.c_uint8)]


class _CompactStats(C.Structure):
    _fields_ = [
        ("input_records", C.c_uint64),
        ("output_records", C.c_uint64),
        ("expired", C.c_uint64),
        ("filtered", C.c_uint64),
        ("tombstones", C.c_uint64),
        ("shadowed", C.c_uint64),
        ("output_bytes", C.c_uint64),
    ]


def _cslice(b: bytes, keep: list) -> _CSlice:
    """Build a _CSlice over a copy of b; the backing buffer is appended to
    `keep`, which the caller must hold alive across the C call (struct field
    assignment copies the slice by value, so the slice itself cannot own the
    buffer)."""
    if not b:
        return _CSlice(None, 0)
    buf = (C.c_uint8 * len(b)).from_buffer_copy(b)
    keep.append(buf)
    return _CSlice(C.cast(buf, C.POINTER(C.c_uint8)), len(b))


def _read_slice(s: _Slice) -> bytes:
    if s.len == 0:
        return b""
    return C.string_at(s.data, s.len)


@dataclass
class ScanResult:
    error: int
    context_id: int
    kvs: list  # [(key, value)]
    kv_count: Optional[int] = None  # only_return_count
    expire_ts: Optional[list] = None
    dev: Optional[dict] = None  # device-resident output descriptors


@dataclass
class CompactStats:
    input_records: int = 0
    output_records: int = 0
    expired: int = 0
    filtered: int = 0
    tombstones: int = 0
    shadowed: int = 0
    output_bytes: int = 0


class RrdbLib:
    """One loaded engine .so."""

    def __init__(self, so_path: str):
        self.path = os.path.abspath(so_path)
        self._lib = C.CDLL(self.path)
        L = self._lib
        L.rrdb_open.restype = C.c_void_p
        L.rrdb_open.argtypes = [C.c_int32, C.c_int32, C.c_int32]
        L.rrdb_close.argtypes = [C.c_void_p]
        L.rrdb_backend.restype = C.c_char_p
        L.rrdb_set_envs.restype = C.c_int32
        L.rrdb_set_envs.argtypes = [C.c_void_p, C.POINTER(C.c_char_p), C.POINTER(C.c_char_p), C.c_int32]
        L.rrdb_set_partition_version.restype = C.c_int32
        L.rrdb_set_partition_version.argtypes = [C.c_void_p, C.c_int32]
        L.rrdb_ingest_run.restype = C.c_int32
        L.rrdb_ingest_run.argtypes = [
            C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_uint64]
        L.rrdb_get.restype = C.c_int32
        L.rrdb_get.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64, C.c_uint32, C.POINTER(_Result)]
        L.rrdb_ttl.restype = C.c_int32
        L.rrdb_ttl.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64, C.c_uint32, C.POINTER(_Result)]
        L.rrdb_sortkey_count.restype = C.c_int32
        L.rrdb_sortkey_count.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64, C.c_uint32, C.POINTER(_Result)]
        L.rrdb_batch_get.restype = C.c_int32
        L.rrdb_batch_get.argtypes = [
            C.c_void_p, C.c_uint64, C.c_void_p, C.c_void_p, C.c_uint32, C.POINTER(_Result)]
        L.rrdb_multi_get.restype = C.c_int32
        L.rrdb_multi_get.argtypes = [C.c_void_p, C.POINTER(_MultiGetRequest), C.c_uint32, C.POINTER(_Result)]
        L.rrdb_scan_open.restype = C.c_int32
        L.rrdb_scan_open.argtypes = [C.c_void_p, C.POINTER(_ScanRequest), C.c_uint32, C.POINTER(_Result)]
        L.rrdb_scan_count_begin.restype = C.c_int32
        L.rrdb_scan_count_begin.argtypes = [C.c_void_p, C.POINTER(_ScanRequest), C.c_uint32]
        L.rrdb_scan_count_finish.restype = C.c_int32
        L.rrdb_scan_count_finish.argtypes = [C.c_void_p, C.POINTER(_Result)]
        L.rrdb_scan_next.restype = C.c_int32
        L.rrdb_scan_next.argtypes = [C.c_void_p, C.c_int64, C.c_uint32, C.POINTER(_Result)]
        L.rrdb_clear_scanner.argtypes = [C.c_void_p, C.c_int64]
        L.rrdb_manual_compact.restype = C.c_int32
        L.rrdb_manual_compact.argtypes = [
            C.c_void_p, C.POINTER(_CompactOptions), C.c_uint32, C.POINTER(_CompactStats)]
        L.rrdb_manual_compact_begin.restype = C.c_int32
        L.rrdb_manual_compact_begin.argtypes = [
            C.c_void_p, C.POINTER(_CompactOptions), C.c_uint32]
        L.rrdb_manual_compact_finish.restype = C.c_int32
        L.rrdb_manual_compact_finish.argtypes = [C.c_void_p, C.POINTER(_CompactStats)]
        L.rrdb_phase_ms.restype = C.c_double
        L.rrdb_phase_ms.argtypes = [C.c_void_p, C.c_char_p]
        L.rrdb_put.restype = C.c_int32
        L.rrdb_put.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64, C.c_char_p, C.c_uint64,
                               C.c_char_p, C.c_uint64, C.c_uint32, C.c_uint32]
        L.rrdb_remove.restype = C.c_int32
        L.rrdb_remove.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64, C.c_char_p, C.c_uint64]
        L.rrdb_flush.restype = C.c_int32
        L.rrdb_flush.argtypes = [C.c_void_p]
        L.rrdb_memtable_entries.restype = C.c_uint64
        L.rrdb_memtable_entries.argtypes = [C.c_void_p]
        L.rrdb_checkpoint.restype = C.c_int32
        L.rrdb_checkpoint.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64]
        L.rrdb_restore.restype = C.c_int32
        L.rrdb_restore.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64]
        L.rrdb_multi_get_batch.restype = C.c_int32
        L.rrdb_multi_get_batch.argtypes = [C.c_void_p, C.c_uint64, C.c_void_p, C.c_void_p,
                                           C.POINTER(_MultiGetRequest), C.c_uint32,
                                           C.POINTER(_Result)]
        L.rrdb_num_runs.restype = C.c_uint64
        L.rrdb_num_runs.argtypes = [C.c_void_p]
        L.rrdb_num_records.restype = C.c_uint64
        L.rrdb_num_records.argtypes = [C.c_void_p]
        L.rrdb_free_result.argtypes = [C.POINTER(_Result)]

    @property
    def backend(self) -> str:
        return self._lib.rrdb_backend().decode()

    def open(self, app_id: int, pidx: int, gpu_id: int) -> "RrdbPartition":
        h = self._lib.rrdb_open(app_id, pidx, gpu_id)
        if not h:
            raise RuntimeError(f"rrdb_open failed (backend={self.backend}, gpu_id={gpu_id})")
        return RrdbPartition(self, h)


def _pack(items):
    """list[bytes] -> (packed buffer, offsets u64 array)"""
    import numpy as np

    offs = np.zeros(len(items) + 1, dtype=np.uint64)
    total = 0
    for i, b in enumerate(items):
        total += len(b)
        offs[i + 1] = total
    buf = b"".join(items)
    return np.frombuffer(bytearray(buf), dtype=np.uint8), offs


class RrdbPartition:
    def __init__(self, lib: RrdbLib, handle):
        self.lib = lib
        self._L = lib._lib
        self._h = handle

    def close(self):
        if self._h:
            self._L.rrdb_close(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def set_envs(self, envs: dict):
        n = len(envs)
        keys = (C.c_char_p * n)(*[k.encode() for k in envs])
        vals = (C.c_char_p * n)(*[str(v).encode() for v in envs.values()])
        return self._L.rrdb_set_envs(self._h, keys, vals, n)

    def set_partition_version(self, pv: int):
        return self._L.rrdb_set_partition_version(self._h, pv)

    def ingest_run_arrays(self, keys_u8, key_offs_u64, vals_u8, val_offs_u64, seq_kind_u64):
        """Ingest from packed numpy arrays (zero python-loop path for bench)."""
        import numpy as np

        for a, dt in ((keys_u8, np.uint8), (key_offs_u64, np.uint64), (vals_u8, np.uint8),
                      (val_offs_u64, np.uint64), (seq_kind_u64, np.uint64)):
            assert a.dtype == dt and a.flags["C_CONTIGUOUS"], (a.dtype, dt)
        n = len(seq_kind_u64)
        st = self._L.rrdb_ingest_run(
            self._h,
            keys_u8.ctypes.data_as(C.c_void_p),
            key_offs_u64.ctypes.data_as(C.c_void_p),
            vals_u8.ctypes.data_as(C.c_void_p),
            val_offs_u64.ctypes.data_as(C.c_void_p),
            seq_kind_u64.ctypes.data_as(C.c_void_p),
            n,
        )
        if st != OK:
            raise RuntimeError(f"rrdb_ingest_run failed: status {st}")
        return st

    def ingest_run(self, records):
        """records: list of (raw_key: bytes, raw_value: bytes, seqno: int, kind: int),
        already sorted by raw_key ascending."""
        keys, koffs = _pack([r[0] for r in records])
        vals, voffs = _pack([r[1] for r in records])
        import numpy as np

        sk = np.array([(r[2] << 1) | r[3] for r in records], dtype=np.uint64)
        return self.ingest_run_arrays(keys, koffs, vals, voffs, sk)

    def num_runs(self):
        return self._L.rrdb_num_runs(self._h)

    def num_records(self):
        return self._L.rrdb_num_records(self._h)

    def _call_result(self, fn, *args):
        res = _Result()
        fn(self._h, *args, C.byref(res))
        return res

    def get(self, raw_key: bytes, epoch_now: int):
        res = self._call_result(self._L.rrdb_get, raw_key, len(raw_key), epoch_now)
        try:
            if res.error != OK:
                return res.error, None
            return OK, _read_slice(res.values[0])
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def ttl(self, raw_key: bytes, epoch_now: int):
        res = self._call_result(self._L.rrdb_ttl, raw_key, len(raw_key), epoch_now)
        try:
            return res.error, (res.i64 if res.error == OK else None)
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def sortkey_count(self, hash_key: bytes, epoch_now: int):
        res = self._call_result(self._L.rrdb_sortkey_count, hash_key, len(hash_key), epoch_now)
        try:
            return res.error, res.i64
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def batch_get(self, raw_keys, epoch_now: int):
        keys, offs = _pack(raw_keys)
        res = _Result()
        self._L.rrdb_batch_get(
            self._h, len(raw_keys), keys.ctypes.data_as(C.c_void_p),
            offs.ctypes.data_as(C.c_void_p), epoch_now, C.byref(res))
        try:
            kvs = [(_read_slice(res.keys[i]), _read_slice(res.values[i]))
                   for i in range(res.count)]
            return res.error, kvs
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def multi_get(self, hash_key: bytes, epoch_now: int, *, start_sortkey=b"", stop_sortkey=b"",
                  start_inclusive=True, stop_inclusive=False, max_kv_count=-1, max_kv_size=-1,
                  no_value=False, reverse=False, sort_key_filter_type=FT_NO_FILTER,
                  sort_key_filter_pattern=b"", sort_keys=None):
        keep = []
        req = _MultiGetRequest()
        req.hash_key = _cslice(hash_key, keep)
        req.start_sortkey = _cslice(start_sortkey, keep)
        req.stop_sortkey = _cslice(stop_sortkey, keep)
        req.start_inclusive = 1 if start_inclusive else 0
        req.stop_inclusive = 1 if stop_inclusive else 0
        req.max_kv_count = max_kv_count
        req.max_kv_size = max_kv_size
        req.no_value = 1 if no_value else 0
        req.reverse = 1 if reverse else 0
        req.sort_key_filter_type = sort_key_filter_type
        req.sort_key_filter_pattern = _cslice(sort_key_filter_pattern, keep)
        if sort_keys:
            sk, offs = _pack(sort_keys)
            req.n_sort_keys = len(sort_keys)
            req.sort_keys = sk.ctypes.data_as(C.POINTER(C.c_uint8))
            req.sort_key_offs = offs.ctypes.data_as(C.POINTER(C.c_uint64))
            keep += [sk, offs]
        res = _Result()
        self._L.rrdb_multi_get(self._h, C.byref(req), epoch_now, C.byref(res))
        del keep
        try:
            kvs = [(_read_slice(res.keys[i]), _read_slice(res.values[i]))
                   for i in range(res.count)]
            return res.error, kvs
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def multi_get_batch(self, hash_keys, epoch_now: int, **shared_kwargs):
        """N full-range (or shared-shape) multi_gets in one call; returns
        (error, [per-request (error, [(sortkey, value), ...])])."""
        keep = []
        req = _MultiGetRequest()
        req.hash_key = _CSlice(None, 0)
        req.start_sortkey = _cslice(shared_kwargs.get("start_sortkey", b""), keep)
        req.stop_sortkey = _cslice(shared_kwargs.get("stop_sortkey", b""), keep)
        req.start_inclusive = 1 if shared_kwargs.get("start_inclusive", True) else 0
        req.stop_inclusive = 1 if shared_kwargs.get("stop_inclusive", False) else 0
        req.max_kv_count = shared_kwargs.get("max_kv_count", -1)
        req.max_kv_size = shared_kwargs.get("max_kv_size", -1)
        req.no_value = 1 if shared_kwargs.get("no_value", False) else 0
        req.reverse = 1 if shared_kwargs.get("reverse", False) else 0
        req.sort_key_filter_type = shared_kwargs.get("sort_key_filter_type", FT_NO_FILTER)
        req.sort_key_filter_pattern = _cslice(
            shared_kwargs.get("sort_key_filter_pattern", b""), keep)
        hks, offs = _pack(hash_keys)
        res = _Result()
        self._L.rrdb_multi_get_batch(self._h, len(hash_keys),
                                     hks.ctypes.data_as(C.c_void_p),
                                     offs.ctypes.data_as(C.c_void_p), C.byref(req), epoch_now,
                                     C.byref(res))
        del keep
        try:
            if res.error != OK:
                return res.error, []
            groups = []
            m = 0
            for i in range(len(hash_keys)):
                n = res.group_counts[i]
                kvs = [(_read_slice(res.keys[m + j]), _read_slice(res.values[m + j]))
                       for j in range(n)]
                m += n
                groups.append((res.group_errors[i], kvs))
            return OK, groups
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def _scan_result(self, res: _Result, only_return_count, return_expire_ts, on_device):
        kvs = []
        ets = None
        dev = None
        if on_device:
            dev = dict(count=res.count, dev_keys=res.dev_keys, dev_key_offs=res.dev_key_offs,
                       dev_vals=res.dev_vals, dev_val_offs=res.dev_val_offs)
        elif not only_return_count:
            kvs = [(_read_slice(res.keys[i]), _read_slice(res.values[i]))
                   for i in range(res.count)]
            if return_expire_ts and res.expire_ts:
                ets = [res.expire_ts[i] for i in range(res.count)]
        return ScanResult(error=res.error, context_id=res.context_id, kvs=kvs,
                          kv_count=(res.i64 if only_return_count else None), expire_ts=ets,
                          dev=dev)

    def scan_open(self, start_key: bytes, stop_key: bytes, epoch_now: int, *,
                  start_inclusive=True, stop_inclusive=False, batch_size=-1, no_value=False,
                  hash_key_filter_type=FT_NO_FILTER, hash_key_filter_pattern=b"",
                  sort_key_filter_type=FT_NO_FILTER, sort_key_filter_pattern=b"",
                  full_scan=False, validate_partition_hash=True, return_expire_ts=False,
                  only_return_count=False, on_device_out=False) -> ScanResult:
        keep = []
        req = _ScanRequest()
        req.start_key = _cslice(start_key, keep)
        req.stop_key = _cslice(stop_key, keep)
        req.start_inclusive = 1 if start_inclusive else 0
        req.stop_inclusive = 1 if stop_inclusive else 0
        req.batch_size = batch_size
        req.no_value = 1 if no_value else 0
        req.hash_key_filter_type = hash_key_filter_type
        req.hash_key_filter_pattern = _cslice(hash_key_filter_pattern, keep)
        req.sort_key_filter_type = sort_key_filter_type
        req.sort_key_filter_pattern = _cslice(sort_key_filter_pattern, keep)
        req.full_scan = 1 if full_scan else 0
        req.validate_partition_hash = 1 if validate_partition_hash else 0
        req.return_expire_ts = 1 if return_expire_ts else 0
        req.only_return_count = 1 if only_return_count else 0
        req.on_device_out = 1 if on_device_out else 0
        res = _Result()
        self._L.rrdb_scan_open(self._h, C.byref(req), epoch_now, C.byref(res))
        try:
            out = self._scan_result(res, only_return_count, return_expire_ts, on_device_out)
            if out.context_id != SCAN_COMPLETED:
                # flags follow the parked context (ids change on every batch)
                self._scan_flags = getattr(self, "_scan_flags", {})
                self._scan_flags[out.context_id] = (only_return_count, return_expire_ts,
                                                    on_device_out)
            return out
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def scan_count_begin(self, start_key: bytes, stop_key: bytes, epoch_now: int, *,
                         stop_inclusive=False, batch_size=2**31 - 1,
                         hash_key_filter_type=FT_NO_FILTER, hash_key_filter_pattern=b"",
                         sort_key_filter_type=FT_NO_FILTER, sort_key_filter_pattern=b"",
                         validate_partition_hash=True):
        """Submit a fused pipelined count scan; returns the C status.
        kInvalidArgument = shape unsupported -> fall back to scan_open."""
        keep = []
        req = _ScanRequest()
        req.start_key = _cslice(start_key, keep)
        req.stop_key = _cslice(stop_key, keep)
        req.start_inclusive = 1
        req.stop_inclusive = 1 if stop_inclusive else 0
        req.batch_size = batch_size
        req.no_value = 1
        req.hash_key_filter_type = hash_key_filter_type
        req.hash_key_filter_pattern = _cslice(hash_key_filter_pattern, keep)
        req.sort_key_filter_type = sort_key_filter_type
        req.sort_key_filter_pattern = _cslice(sort_key_filter_pattern, keep)
        req.full_scan = 1
        req.validate_partition_hash = 1 if validate_partition_hash else 0
        req.return_expire_ts = 0
        req.only_return_count = 1
        req.on_device_out = 0
        rc = self._L.rrdb_scan_count_begin(self._h, C.byref(req), epoch_now)
        del keep
        return rc

    def scan_count_finish(self):
        """(error, count) for a submitted scan_count_begin."""
        res = _Result()
        self._L.rrdb_scan_count_finish(self._h, C.byref(res))
        try:
            return res.error, res.i64
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def scan_next(self, context_id: int, epoch_now: int) -> ScanResult:
        res = _Result()
        self._L.rrdb_scan_next(self._h, context_id, epoch_now, C.byref(res))
        try:
            flags = getattr(self, "_scan_flags", {})
            orc, ret, dev = flags.pop(context_id, (False, False, False))
            out = self._scan_result(res, orc, ret, dev)
            if out.context_id != SCAN_COMPLETED and out.error == OK:
                flags[out.context_id] = (orc, ret, dev)
            return out
        finally:
            self._L.rrdb_free_result(C.byref(res))

    def clear_scanner(self, context_id: int):
        self._L.rrdb_clear_scanner(self._h, context_id)

    def phase_ms(self, phase: str) -> float:
        return self._L.rrdb_phase_ms(self._h, phase.encode())

    # ---- write path (§8(f)1) ----
    def put(self, hash_key: bytes, sort_key: bytes, value: bytes, expire_ts: int = 0,
            epoch_now: int = 0):
        """epoch_now = the write's clock; with a default_ttl table env and
        expire_ts 0, the stored expire becomes epoch_now + default_ttl at
        WRITE time (rocksdb_wrapper.cpp:280-286)."""
        return self._L.rrdb_put(self._h, hash_key, len(hash_key), sort_key, len(sort_key),
                                value, len(value), expire_ts, epoch_now)

    def remove(self, hash_key: bytes, sort_key: bytes):
        return self._L.rrdb_remove(self._h, hash_key, len(hash_key), sort_key, len(sort_key))

    def flush(self):
        return self._L.rrdb_flush(self._h)

    def memtable_entries(self):
        return self._L.rrdb_memtable_entries(self._h)

    # ---- checkpoint (§8(f)2) ----
    def checkpoint(self, directory: str, decree: int):
        return self._L.rrdb_checkpoint(self._h, directory.encode(), decree)

    def restore(self, directory: str, decree: int):
        return self._L.rrdb_restore(self._h, directory.encode(), decree)

    def manual_compact(self, epoch_now: int, *, target_level=-1, bottommost_force=True,
                       keep_inputs=False):
        opts = _CompactOptions(target_level, 1 if bottommost_force else 0,
                               1 if keep_inputs else 0)
        st = _CompactStats()
        err = self._L.rrdb_manual_compact(self._h, C.byref(opts), epoch_now, C.byref(st))
        stats = CompactStats(**{f[0]: getattr(st, f[0]) for f in _CompactStats._fields_})
        return err, stats

    def manual_compact_begin(self, epoch_now: int, *, target_level=-1, bottommost_force=True,
                             keep_inputs=False):
        """Submit the compaction's merge phase without blocking; pair with
        manual_compact_finish.  The pipelined-partitions seam
        (include/rrdb_engine.h)."""
        opts = _CompactOptions(target_level, 1 if bottommost_force else 0,
                               1 if keep_inputs else 0)
        return self._L.rrdb_manual_compact_begin(self._h, C.byref(opts), epoch_now)

    def manual_compact_finish(self, epoch_now: int = 0):
        st = _CompactStats()
        err = self._L.rrdb_manual_compact_finish(self._h, C.byref(st))
        stats = CompactStats(**{f[0]: getattr(st, f[0]) for f in _CompactStats._fields_})
        return err, stats
