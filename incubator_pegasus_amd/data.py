"""Host-side run building + synthetic YCSB-shaped data generation.

This is product host code: it builds the sorted runs that rrdb_ingest_run
consumes (the memtable-flush / bulk-load equivalent).
Codecs restated from the reference:
  key   = [u16 BE hash_key_len][hash_key][sort_key]
          (reference src/base/pegasus_key_schema.h:35-58)
  value v0 = [u32 BE expire_ts][user_data]
          (reference src/base/pegasus_value_schema.h:158-172)
  value v1 = [u32 BE expire_ts][u64 BE timetag][user_data]
          (reference src/base/pegasus_value_schema.h:205-226)
  value v2 = [u8 0x80|2][u32 BE expire_ts][u64 BE timetag][user_data]
          (reference src/base/value_schema_v2.cpp:86-98)

Synthetic workload shape per SURVEY.md §8(d) / BASELINE.md: 16B hashkey
("u:" + 14 decimal digits), sortkey "" (point tables) or 8B counter (scan
tables), 100B value, schema v1, splitmix64 seeding (seed 20260915).
"""
from __future__ import annotations

import struct

import numpy as np

DEFAULT_SEED = 20260915


# ---------------- scalar codecs (tests / small paths) ----------------

def generate_key(hash_key: bytes, sort_key: bytes = b"") -> bytes:
    assert len(hash_key) < 0xFFFF
    return struct.pack(">H", len(hash_key)) + hash_key + sort_key


def restore_key(raw: bytes):
    (hklen,) = struct.unpack(">H", raw[:2])
    return raw[2 : 2 + hklen], raw[2 + hklen :]


def generate_next_blob(hash_key: bytes, sort_key: bytes = None) -> bytes:
    """pegasus_generate_next_blob (pegasus_key_schema.h:64-98): +1 on the last
    non-0xFF byte, truncating the 0xFF tail."""
    raw = bytearray(generate_key(hash_key, sort_key or b""))
    p = len(raw) - 1
    while raw[p] == 0xFF:
        p -= 1
    raw[p] += 1
    return bytes(raw[: p + 1])


def encode_value(user_data: bytes, expire_ts: int = 0, timetag: int = 0, version: int = 1) -> bytes:
    if version == 0:
        return struct.pack(">I", expire_ts) + user_data
    if version == 1:
        return struct.pack(">IQ", expire_ts, timetag) + user_data
    if version == 2:
        return struct.pack(">BIQ", 0x80 | 2, expire_ts, timetag) + user_data
    raise ValueError(version)


def value_header_len(version: int) -> int:
    return {0: 4, 1: 12, 2: 13}[version]


def decode_value(raw: bytes, version: int = 1):
    hdr = value_header_len(version)
    off = 1 if version == 2 else 0
    (expire_ts,) = struct.unpack(">I", raw[off : off + 4])
    timetag = 0
    if version >= 1:
        (timetag,) = struct.unpack(">Q", raw[off + 4 : off + 12])
    return expire_ts, timetag, raw[hdr:]


# ---------------- splitmix64 (vectorized) ----------------

def splitmix64(x: np.ndarray) -> np.ndarray:
    z = (x.astype(np.uint64) + np.uint64(0x9E3779B97F4A7C15))
    z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return z ^ (z >> np.uint64(31))


# ---------------- vectorized dataset generation ----------------

def _digits(ids: np.ndarray, width: int) -> np.ndarray:
    """ids (u64) -> (N, width) array of ASCII digit bytes, zero padded."""
    out = np.empty((len(ids), width), dtype=np.uint8)
    x = ids.copy()
    ten = np.uint64(10)
    for i in range(width - 1, -1, -1):
        out[:, i] = (x % ten).astype(np.uint8) + ord("0")
        x //= ten
    return out


def make_hashkeys(ids: np.ndarray) -> np.ndarray:
    """(N,16) u8: b'u:' + 14 digits — fixed 16B hashkeys."""
    n = len(ids)
    hk = np.empty((n, 16), dtype=np.uint8)
    hk[:, 0] = ord("u")
    hk[:, 1] = ord(":")
    hk[:, 2:] = _digits(ids, 14)
    return hk


def make_raw_keys(ids: np.ndarray, sort_ids: np.ndarray = None) -> np.ndarray:
    """(N,18) or (N,26) u8 raw rocksdb keys (16B hashkey, optional 8B sortkey).
    ids ascending => keys ascending bytewise (fixed width decimal)."""
    hk = make_hashkeys(ids)
    n = len(ids)
    if sort_ids is None:
        raw = np.empty((n, 18), dtype=np.uint8)
        raw[:, 0] = 0
        raw[:, 1] = 16
        raw[:, 2:] = hk
        return raw
    raw = np.empty((n, 26), dtype=np.uint8)
    raw[:, 0] = 0
    raw[:, 1] = 16
    raw[:, 2:18] = hk
    raw[:, 18:] = _digits(sort_ids, 8)
    return raw


def make_values(ids: np.ndarray, value_len: int = 100, expire_ts: np.ndarray = None,
                timetag: np.ndarray = None, version: int = 1, salt: int = 0) -> np.ndarray:
    """(N, hdr+value_len) u8 encoded values; body = seeded bytes from splitmix64."""
    n = len(ids)
    hdr = value_header_len(version)
    out = np.zeros((n, hdr + value_len), dtype=np.uint8)
    off = 1 if version == 2 else 0
    if version == 2:
        out[:, 0] = 0x82
    if expire_ts is not None:
        e = expire_ts.astype(np.uint32)
        out[:, off + 0] = (e >> 24).astype(np.uint8)
        out[:, off + 1] = (e >> 16).astype(np.uint8)
        out[:, off + 2] = (e >> 8).astype(np.uint8)
        out[:, off + 3] = e.astype(np.uint8)
    if version >= 1 and timetag is not None:
        t = timetag.astype(np.uint64)
        for b in range(8):
            out[:, off + 4 + b] = (t >> np.uint64(8 * (7 - b))).astype(np.uint8)
    # value body: 8 bytes per splitmix word (little-endian u64 -> u8 view)
    words = (value_len + 7) // 8
    idx = (ids.astype(np.uint64)[:, None] * np.uint64(words) +
           np.arange(words, dtype=np.uint64)[None, :] + np.uint64(salt))
    body = np.ascontiguousarray(splitmix64(idx.reshape(-1))).view(np.uint8).reshape(n, words * 8)
    out[:, hdr:] = body[:, :value_len]
    return out


def fixed_offsets(n: int, reclen: int) -> np.ndarray:
    return (np.arange(n + 1, dtype=np.uint64) * np.uint64(reclen))


def build_point_table_runs(n_keys: int, n_runs: int, *, seed: int = DEFAULT_SEED,
                           value_len: int = 100, dup_fraction: float = 0.1,
                           delete_fraction: float = 0.0, ttl_fraction: float = 0.0,
                           ttl_expire_ts: int = 0, version: int = 1):
    """Generate n_runs sorted runs over n_keys unique 16B hashkeys (sortkey "").

    Each key's base version goes to run (splitmix(id) % n_runs); a
    dup_fraction of keys ALSO get a newer version in a later run (exercises
    newest-wins), a delete_fraction get a newer tombstone.  Returns list of
    dicts {keys,koff,vals,voff,sk} oldest-first (ingest order).
    """
    ids = np.arange(n_keys, dtype=np.uint64)
    h = splitmix64(ids + np.uint64(seed))
    base_run = (h % np.uint64(n_runs)).astype(np.int64)
    dup_sel = (splitmix64(h) % np.uint64(1000)).astype(np.float64) / 1000.0
    ttl_sel = (splitmix64(h + np.uint64(7)) % np.uint64(1000)).astype(np.float64) / 1000.0

    runs = []
    seq_base = 1
    for r in range(n_runs):
        mask = base_run == r
        rids = ids[mask]
        n = len(rids)
        raw = make_raw_keys(rids)
        expire = np.zeros(n, dtype=np.uint32)
        if ttl_fraction > 0:
            expire[ttl_sel[mask] < ttl_fraction] = ttl_expire_ts
        vals = make_values(rids, value_len, expire_ts=expire, version=version, salt=0)
        sk = ((np.arange(n, dtype=np.uint64) + np.uint64(seq_base)) << np.uint64(1))
        seq_base += n
        runs.append(dict(keys=raw.reshape(-1), koff=fixed_offsets(n, raw.shape[1]),
                         vals=vals.reshape(-1), voff=fixed_offsets(n, vals.shape[1]),
                         sk=sk))
    # newer overlay run: dups (new values) + deletes
    overlay_dup = dup_sel < dup_fraction
    overlay_del = (dup_sel >= dup_fraction) & (dup_sel < dup_fraction + delete_fraction)
    overlay = overlay_dup | overlay_del
    if overlay.any():
        oids = ids[overlay]
        n = len(oids)
        raw = make_raw_keys(oids)
        # fixed-size encoded values for every overlay record, tombstones
        # included — a tombstone's value bytes are never read (engine and
        # reference both branch on kind before touching the value)
        vals = make_values(oids, value_len, version=version, salt=1)
        kinds = np.where(overlay_del[overlay], np.uint64(1), np.uint64(0))
        sk = (((np.arange(n, dtype=np.uint64)) + np.uint64(seq_base)) << np.uint64(1)) | kinds
        seq_base += n
        runs.append(dict(keys=raw.reshape(-1), koff=fixed_offsets(n, raw.shape[1]),
                         vals=vals.reshape(-1), voff=fixed_offsets(n, vals.shape[1]), sk=sk))
    return runs


def build_scan_table_run(n_hashkeys: int, sortkeys_per_hash: int, *, seed: int = DEFAULT_SEED,
                         value_len: int = 100, version: int = 1):
    """One sorted run with sortkeys: hashkey i gets sortkeys_per_hash rows."""
    n = n_hashkeys * sortkeys_per_hash
    hk_ids = np.repeat(np.arange(n_hashkeys, dtype=np.uint64), sortkeys_per_hash)
    sk_ids = np.tile(np.arange(sortkeys_per_hash, dtype=np.uint64), n_hashkeys)
    raw = make_raw_keys(hk_ids, sk_ids)
    vals = make_values(hk_ids * np.uint64(sortkeys_per_hash) + sk_ids, value_len,
                       version=version)
    sk = ((np.arange(n, dtype=np.uint64) + np.uint64(1)) << np.uint64(1))
    return dict(keys=raw.reshape(-1), koff=fixed_offsets(n, raw.shape[1]),
                vals=vals.reshape(-1), voff=fixed_offsets(n, vals.shape[1]), sk=sk)


def zipfian_ids(n_samples: int, n_items: int, *, theta: float = 0.99,
                seed: int = DEFAULT_SEED) -> np.ndarray:
    """Bounded zipfian item ids (YCSB's Gray et al. generator, vectorized)."""
    # zeta(n, theta) approximated by integral for large n (exact enough for
    # workload shaping; this is data gen, not a parity surface)
    n_items = int(n_items)
    if n_items < 10000:
        zetan = np.sum(1.0 / np.power(np.arange(1, n_items + 1), theta))
    else:
        k = 10000
        zetan = np.sum(1.0 / np.power(np.arange(1, k + 1), theta))
        # integral tail approximation of sum_{k+1..n} x^-theta
        zetan += (np.power(float(n_items), 1 - theta) - np.power(float(k), 1 - theta)) / (1 - theta)
    alpha = 1.0 / (1.0 - theta)
    eta = (1 - np.power(2.0 / n_items, 1 - theta)) / (1 - np.sum(1.0 / np.power(np.arange(1, 3), theta)) / zetan)
    u = (splitmix64(np.arange(n_samples, dtype=np.uint64) + np.uint64(seed * 31 + 11)) >> np.uint64(11)).astype(np.float64) / float(1 << 53)
    uz = u * zetan
    ids = (n_items * np.power(eta * u - eta + 1.0, alpha)).astype(np.uint64)
    ids = np.where(uz < 1.0, np.uint64(0), ids)
    ids = np.where((uz >= 1.0) & (uz < 1.0 + np.power(0.5, theta)), np.uint64(1), ids)
    ids = np.minimum(ids, np.uint64(n_items - 1))
    # YCSB scrambles to spread hot items across the keyspace
    return splitmix64(ids) % np.uint64(n_items)
