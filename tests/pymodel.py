"""Independent pure-Python model of the hot-path semantics, used as a second
check on the CPU oracle at small sizes (the oracle itself is the parity anchor
for the HIP engine; this model keeps the oracle honest).

Written directly from the cited reference logic, independently from the C
oracle's code:
  - merging-iterator / newest-wins / tombstones: rocksdb v8.5.3 semantics at
    the engine boundary
  - handlers: reference src/server/pegasus_server_impl.cpp:418-1547
  - compaction filter: reference src/server/key_ttl_compaction_filter.h:55-121
"""
from __future__ import annotations

import struct


def crc64(data: bytes, init: int = 0) -> int:
    bits = [63, 61, 59, 58, 56, 55, 52, 49, 48, 47, 46, 44, 41, 37, 36, 34, 32, 31, 28, 26,
            23, 22, 19, 16, 13, 12, 10, 9, 6, 4, 3, 0]
    poly = 0
    for b in bits:
        poly |= 1 << (63 - b)
    tbl = []
    for i in range(256):
        k = i
        for _ in range(8):
            k = (k >> 1) ^ poly if k & 1 else k >> 1
        tbl.append(k)
    crc = ~init & 0xFFFFFFFFFFFFFFFF
    for byte in data:
        crc = tbl[(crc ^ byte) & 0xFF] ^ (crc >> 8)
    return ~crc & 0xFFFFFFFFFFFFFFFF


def hdr_len(ver):
    return {0: 4, 1: 12, 2: 13}[ver]


def expire_of(val: bytes, ver: int) -> int:
    off = 1 if ver == 2 else 0
    return struct.unpack(">I", val[off:off + 4])[0]


def expired(epoch_now: int, ts: int) -> bool:
    return ts > 0 and ts <= epoch_now


class Model:
    def __init__(self, pidx=0, data_version=1):
        self.runs = []  # list of dict key->(value, seq, kind); index = age order
        self.pidx = pidx
        self.partition_version = -1
        self.validate_hash = False
        self.data_version = data_version
        self.default_ttl = 0
        self.user_ops = []  # parsed op dicts
        self.next_seq_floor = 0  # write-path seqno source / ingest floor
        self.mem = []  # memtable log: [(key, value, seq, kind)]

    def ingest(self, records):
        """records: [(raw_key, raw_value, seq, kind)]"""
        self.runs.append({k: (v, s, kd) for k, v, s, kd in records})
        if records:
            self.next_seq_floor = max(self.next_seq_floor,
                                      max(s for _, _, s, _ in records) + 1)

    # -- write path (pegasus_write_service put/remove + memtable flush,
    #    reference pegasus_write_service.h:119-207, rocksdb_wrapper.cpp:
    #    121-247; mirrors the oracle/engine host memtable) --
    def put(self, hash_key: bytes, sort_key: bytes, body: bytes, expire_ts=0,
            epoch_now=0):
        key = struct.pack(">H", len(hash_key)) + hash_key + sort_key
        # write-time default_ttl (rocksdb_wrapper::db_expire_ts, :280-286)
        if expire_ts == 0 and self.default_ttl != 0:
            expire_ts = (epoch_now + self.default_ttl) & 0xFFFFFFFF
        ver = self.data_version
        if ver == 0:
            val = struct.pack(">I", expire_ts & 0xFFFFFFFF) + body
        elif ver == 1:
            val = struct.pack(">IQ", expire_ts & 0xFFFFFFFF, 0) + body
        else:
            val = b"\x82" + struct.pack(">IQ", expire_ts & 0xFFFFFFFF, 0) + body
        self.mem.append((key, val, self.next_seq_floor, 0))
        self.next_seq_floor += 1

    def remove(self, hash_key: bytes, sort_key: bytes):
        key = struct.pack(">H", len(hash_key)) + hash_key + sort_key
        self.mem.append((key, b"", self.next_seq_floor, 1))
        self.next_seq_floor += 1

    def flush(self):
        """memtable -> sorted run: newest version per key survives
        (memtable upsert semantics)."""
        if not self.mem:
            return
        newest = {}
        for k, v, s, kd in self.mem:
            if k not in newest or s > newest[k][1]:
                newest[k] = (v, s, kd)
        self.runs.append(newest)
        self.mem = []

    # -- visibility --
    def newest(self, key: bytes):
        best = None
        for run in self.runs:
            if key in run:
                v, s, kd = run[key]
                if best is None or s > best[1]:
                    best = (v, s, kd)
        return best

    def visible_items(self, lo=None, hi=None, lo_incl=True, hi_incl=False):
        """key-ordered [(key, value)] of visible (newest, non-tombstone)."""
        newest = {}
        for run in self.runs:
            for k, (v, s, kd) in run.items():
                if k not in newest or s > newest[k][1]:
                    newest[k] = (v, s, kd)
        out = []
        for k in sorted(newest):
            if lo is not None and (k < lo or (k == lo and not lo_incl)):
                continue
            if hi is not None and (k > hi or (k == hi and not hi_incl)):
                continue
            v, s, kd = newest[k]
            if kd == 0:
                out.append((k, v))
        return out

    # -- handlers --
    def get(self, key: bytes, now: int):
        r = self.newest(key)
        if r is None or r[2] == 1:
            return 1, None
        if expired(now, expire_of(r[0], self.data_version)):
            return 1, None
        return 0, r[0][hdr_len(self.data_version):]

    def ttl(self, key: bytes, now: int):
        r = self.newest(key)
        if r is None or r[2] == 1:
            return 1, None
        e = expire_of(r[0], self.data_version)
        if expired(now, e):
            return 1, None
        return 0, (e - now) if e > 0 else -1

    def batch_get(self, raw_keys, now: int):
        """on_batch_get (pegasus_server_impl.cpp:906-1016): multi-point
        lookup in request order; NotFound/expired keys are skipped, an
        empty request is kInvalidArgument."""
        if not raw_keys:
            return 4, []
        out = []
        for k in raw_keys:
            st, v = self.get(k, now)
            if st == 0:
                out.append((k, v))
        return 0, out

    def multi_get_sortkeys(self, hash_key: bytes, sort_keys, now: int, *,
                           max_kv_count=-1, max_kv_size=-1, no_value=False,
                           engine_max_iter=3000):
        """Point-list variant of on_multi_get (pegasus_server_impl.cpp:779-860):
        per requested sortkey in order, NotFound/expired skip; the cap check
        runs BEFORE the append (:841-845), so the NEXT surviving record after
        the caps fill returns kIncomplete; kv.key echoes the REQUESTED
        sortkey."""
        max_count = engine_max_iter if max_kv_count <= 0 \
            else min(max_kv_count, engine_max_iter)
        max_size = max_kv_size if max_kv_size > 0 else (1 << 31) - 1
        out = []
        count = size = 0
        exceed = False
        for sk in sort_keys:
            key = struct.pack(">H", len(hash_key)) + hash_key + sk
            st, body = self.get(key, now)
            if st != 0:
                continue
            if count >= max_count or size >= max_size:
                exceed = True
                break
            val = b"" if no_value else body
            out.append((sk, val))
            count += 1
            size += len(sk) + len(val)
        return (7 if exceed else 0), out

    def sortkey_count(self, hash_key: bytes, now: int):
        lo = struct.pack(">H", len(hash_key)) + hash_key
        hi = bytearray(lo)
        p = len(hi) - 1
        while hi[p] == 0xFF:
            p -= 1
        hi[p] += 1
        hi = bytes(hi[:p + 1])
        items = self.visible_items(lo, hi, True, False)
        return sum(1 for _, v in items if not expired(now, expire_of(v, self.data_version)))

    def _iter_visible(self, lo, hi, lo_incl, hi_incl):
        """iterator-level view: newest version per key is a PUT (expired
        records stay visible at this level — expiry is app-level)."""
        return self.visible_items(lo, hi, lo_incl, hi_incl)

    def _valid_beyond(self, bound: bytes, reverse: bool) -> bool:
        """it->Valid() after a limit exit: any iterator-visible record
        beyond the boundary anywhere in the DB (on_multi_get:777-788)."""
        if not reverse:
            return len(self.visible_items(bound, None, True, False)) > 0
        return len(self.visible_items(None, bound, True, False)) > 0

    def multi_get(self, hash_key: bytes, now: int, *, start_sortkey=b"",
                  stop_sortkey=b"", start_inclusive=True, stop_inclusive=False,
                  max_kv_count=-1, max_kv_size=-1, reverse=False,
                  sort_key_filter_type=0, sort_key_filter_pattern=b"",
                  no_value=False, engine_max_iter=3000):
        """Range variant of on_multi_get (pegasus_server_impl.cpp:540-799),
        restated independently: the forward/reverse iterator walk with the
        range_read_limiter count cap and the iterator-validity completion
        (kIncomplete iff a limit exit leaves the iterator Valid anywhere)."""
        pre = struct.pack(">H", len(hash_key)) + hash_key
        start = pre + start_sortkey
        if stop_sortkey == b"":
            hi = bytearray(pre)
            p = len(hi) - 1
            while hi[p] == 0xFF:
                p -= 1
            hi[p] += 1
            stop = bytes(hi[:p + 1])
            stop_incl = False
        else:
            stop = pre + stop_sortkey
            stop_incl = stop_inclusive
        # prefix filter clamps the range (on_multi_get:558-578)
        if sort_key_filter_type == 2 and len(sort_key_filter_pattern) > 0:
            ps = pre + sort_key_filter_pattern
            pe = bytearray(ps)
            q = len(pe) - 1
            while pe[q] == 0xFF:
                q -= 1
            pe[q] += 1
            pe = bytes(pe[:q + 1])
            if ps > start:
                start = ps
                start_inclusive = True
            if pe <= stop:
                stop = pe
                stop_incl = False
        c = (start > stop) - (start < stop)
        if c > 0 or (c == 0 and (not start_inclusive or not stop_incl)):
            return 0, []  # empty range
        max_count = engine_max_iter if max_kv_count <= 0 else min(max_kv_count,
                                                                  engine_max_iter)
        max_size = max_kv_size if max_kv_size > 0 else (1 << 31) - 1

        def sk_filter_ok(sk: bytes) -> bool:
            ft, pat = sort_key_filter_type, sort_key_filter_pattern
            if ft == 0 or len(pat) == 0:
                return True
            if len(sk) < len(pat):
                return False
            if ft == 1:
                return pat in sk
            if ft == 2:
                return sk.startswith(pat)
            return sk.endswith(pat)
        rows = self._iter_visible(None, None, True, True)  # whole DB in key order
        if reverse:
            rows = list(reversed(rows))
        out = []
        count = 0
        size = 0
        iteration = 0
        complete = False
        first_exclusive = (not start_inclusive) if not reverse else (not stop_incl)
        it_valid = False
        i = 0
        # Seek: first row >= start (fwd) / last row <= stop (rev)
        while i < len(rows):
            k = rows[i][0]
            if not reverse and k >= start:
                break
            if reverse and (k < stop or (k == stop and stop_incl)):
                break
            i += 1
        while True:
            if not (count < max_count and iteration < engine_max_iter and
                    size < max_size):
                it_valid = i < len(rows)
                break
            if i >= len(rows):
                it_valid = False
                break
            k, v = rows[i]
            if not reverse:
                cc = (k > stop) - (k < stop)
                if cc > 0 or (cc == 0 and not stop_incl):
                    complete = True
                    break
                if first_exclusive:
                    first_exclusive = False
                    if k == start:
                        i += 1
                        continue
            else:
                cc = (k > start) - (k < start)
                if cc < 0 or (cc == 0 and not start_inclusive):
                    complete = True
                    break
                if first_exclusive:
                    first_exclusive = False
                    if k == stop:
                        i += 1
                        continue
            iteration += 1
            sk = k[len(pre):]
            if not expired(now, expire_of(v, self.data_version)) and sk_filter_ok(sk):
                # no_value leaves kv.value empty, so the size budget counts
                # only the sortkey bytes (append_key_value_for_multi_get
                # :2498-2502 + the kv_size accounting :726-728)
                uv = b"" if no_value else v[hdr_len(self.data_version):]
                out.append((sk, uv))
                count += 1
                size += len(sk) + len(uv)
            if cc == 0:
                complete = True
                break
            i += 1
        if reverse:
            out.reverse()
        error = 0 if (complete or not it_valid) else 7  # kIncomplete
        return error, out

    def full_scan(self, now: int, validate_hash_req=True):
        """All visible non-expired rows (hash-valid) — count_data equivalent."""
        out = []
        for k, v in self.visible_items():
            if expired(now, expire_of(v, self.data_version)):
                continue
            if validate_hash_req and self.validate_hash:
                if self.partition_version < 0 or self.pidx > self.partition_version:
                    continue
                hklen = struct.unpack(">H", k[:2])[0]
                h = crc64(k[2:2 + hklen] if hklen else k[2:])
                if (h & self.partition_version) != self.pidx:
                    continue
            out.append((k, v[hdr_len(self.data_version):]))
        return out

    def scan(self, now: int, *, start_key=b"", stop_key=b"", start_inclusive=True,
             stop_inclusive=False, batch_size=-1, max_iteration_count=1000,
             no_value=False, hash_key_filter_type=0, hash_key_filter_pattern=b"",
             sort_key_filter_type=0, sort_key_filter_pattern=b"",
             validate_hash_req=True, return_expire_ts=False,
             only_return_count=False):
        """on_get_scanner + on_scan paging (pegasus_server_impl.cpp:1151-1397):
        returns (error, batches); each batch is (kvs, expire_ts_or_None,
        count_or_None).  Mirrors the handler's filter-type validation, the
        hashkey MATCH_PREFIX start clamp (:1206-1224), the empty-range check
        (:1227-1243), the first-exclusive skip (:1277-1283), and per-batch
        paging capped by min(batch_size, rocksdb.max_iteration_count) where
        expiry / hash-mismatch / filter rejects consume iterations but emit
        nothing (validate_for_scan equivalent)."""
        if not (0 <= hash_key_filter_type <= 3 and 0 <= sort_key_filter_type <= 3) \
                or (hash_key_filter_type == 2 and len(hash_key_filter_pattern) >= 0xFFFF):
            return 4, []  # kInvalidArgument
        start, s_incl = start_key, start_inclusive
        if hash_key_filter_type == 2 and len(hash_key_filter_pattern) > 0:
            pstart = struct.pack(">H", len(hash_key_filter_pattern)) + hash_key_filter_pattern
            if pstart > start:
                start, s_incl = pstart, True
        if start > stop_key or (start == stop_key and not (s_incl and stop_inclusive)):
            # early return (:1227-1243): no arrays allocated — expire_ts is
            # null even when requested, unlike a normal zero-row batch
            return 0, [([], None, 0 if only_return_count else None)]
        view = self.visible_items(start, stop_key, True, stop_inclusive)
        cursor = 0
        if not s_incl and view and view[0][0] == start:
            cursor = 1

        def match(ft, pat, v):
            if ft == 0 or len(pat) == 0:
                return True
            if len(v) < len(pat):
                return False
            if ft == 1:
                return pat in v
            if ft == 2:
                return v.startswith(pat)
            return v.endswith(pat)

        hdr = hdr_len(self.data_version)
        batch_count = max_iteration_count
        if batch_size > 0:
            batch_count = min(batch_size, batch_count)
        batches = []
        while True:
            kvs, ets, count, iteration = [], [], 0, 0
            while cursor < len(view):
                if count >= batch_count or iteration >= max_iteration_count:
                    break
                k, v = view[cursor]
                cursor += 1
                iteration += 1
                if expired(now, expire_of(v, self.data_version)):
                    continue
                if validate_hash_req and self.validate_hash:
                    if self.partition_version < 0 or self.pidx > self.partition_version:
                        continue
                    hklen = struct.unpack(">H", k[:2])[0]
                    h = crc64(k[2:2 + hklen] if hklen else k[2:])
                    if (h & self.partition_version) != self.pidx:
                        continue
                if hash_key_filter_type or sort_key_filter_type:
                    hklen = struct.unpack(">H", k[:2])[0]
                    if not match(hash_key_filter_type, hash_key_filter_pattern,
                                 k[2:2 + hklen]):
                        continue
                    if not match(sort_key_filter_type, sort_key_filter_pattern,
                                 k[2 + hklen:]):
                        continue
                count += 1
                if not only_return_count:
                    e = expire_of(v, self.data_version)
                    ets.append(e - (1 << 32) if e >= (1 << 31) else e)
                    kvs.append((k, b"" if no_value else v[hdr:]))
            batches.append((kvs,
                            ets if (return_expire_ts and not only_return_count)
                            else None,
                            count if only_return_count else None))
            if cursor >= len(view):
                return 0, batches

    def compact(self, now: int):
        """Manual compaction result: surviving {key: value} after newest-wins,
        tombstone drop, TTL filter, default-ttl rewrite and user ops."""
        surviving, _ = self.compact_full(now)
        return surviving

    def compact_full(self, now: int):
        """compact() plus the full CompactStats accounting the C-ABI reports:
        input_records (all physical records), shadowed (non-newest versions),
        tombstones (newest-version DELETEs dropped at the bottommost level),
        filtered (user-rule deletes + stale-split drops), expired (TTL drops
        on the post-default-ttl / pre-user-op expire), output_bytes (kept
        key+value bytes)."""
        surviving = {}
        stats = dict(input_records=0, output_records=0, expired=0, filtered=0,
                     tombstones=0, shadowed=0, output_bytes=0)
        newest = {}
        counts = {}
        for run in self.runs:
            stats["input_records"] += len(run)
            for k, (v, s, kd) in run.items():
                counts[k] = counts.get(k, 0) + 1
                if k not in newest or s > newest[k][1]:
                    newest[k] = (v, s, kd)
        stats["shadowed"] = sum(c - 1 for c in counts.values())
        for k in sorted(newest):
            v, s, kd = newest[k]
            if kd == 1:
                stats["tombstones"] += 1
                continue
            reason, newv = self._filter(k, v, now)
            if reason is None:
                surviving[k] = (newv if newv is not None else v, s)
                stats["output_records"] += 1
                stats["output_bytes"] += len(k) + len(v)
            else:
                stats[reason] += 1
        self.runs = [{k: (v, s, 0) for k, (v, s) in surviving.items()}]
        return surviving, stats

    def _filter(self, key: bytes, value: bytes, now: int):
        """KeyWithTTLCompactionFilter::Filter (:55-92).  Returns
        (drop_reason, new_value): reason None = keep, "filtered" = user-rule
        delete or stale-split drop, "expired" = TTL drop."""
        ver = self.data_version
        if len(key) < 2:
            return None, None
        expire_ts = expire_of(value, ver)
        new_value = None
        if self.default_ttl != 0 and expire_ts == 0:
            expire_ts = (now + self.default_ttl) & 0xFFFFFFFF
            off = 1 if ver == 2 else 0
            new_value = value[:off] + struct.pack(">I", expire_ts) + value[off + 4:]
        if self.user_ops:
            vv = new_value if new_value is not None else value
            hklen = struct.unpack(">H", key[:2])[0]
            hk, sk = key[2:2 + hklen], key[2 + hklen:]
            for op in self.user_ops:
                if not self._all_rules_match(op, hk, sk, vv, now):
                    continue
                if op["type"] == "delete":
                    return "filtered", None
                # update_ttl
                cur = expire_of(vv, ver)
                t = op["ut_type"]
                if t == "from_now":
                    new_ts = (now + op["value"]) & 0xFFFFFFFF
                elif t == "from_current":
                    if cur == 0:
                        continue
                    new_ts = (op["value"] + cur) & 0xFFFFFFFF
                elif t == "timestamp":
                    new_ts = (op["value"] - 1451606400) & 0xFFFFFFFF
                else:
                    continue
                off = 1 if ver == 2 else 0
                new_value = vv[:off] + struct.pack(">I", new_ts) + vv[off + 4:]
        if expired(now, expire_ts):
            return "expired", None
        if self.validate_hash and self.partition_version >= 0 and self.pidx <= self.partition_version:
            hklen = struct.unpack(">H", key[:2])[0]
            h = crc64(key[2:2 + hklen] if hklen else key[2:])
            if (h & self.partition_version) != self.pidx:
                return "filtered", None
        return None, new_value

    def _all_rules_match(self, op, hk, sk, value, now):
        if not op["rules"]:
            return False
        for r in op["rules"]:
            if not self._rule_match(r, hk, sk, value, now):
                return False
        return True

    def _rule_match(self, r, hk, sk, value, now):
        t = r["type"]
        if t in ("hashkey", "sortkey"):
            target = hk if t == "hashkey" else sk
            pat = r["pattern"]
            if not pat or len(target) < len(pat):
                return False
            mt = r["match_type"]
            if mt == "anywhere":
                return pat in target
            if mt == "prefix":
                return target.startswith(pat)
            if mt == "postfix":
                return target.endswith(pat)
            return False
        if t == "ttl_range":
            e = expire_of(value, self.data_version)
            if e == 0 and r["start_ttl"] == 0 and r["stop_ttl"] == 0:
                return True
            return ((r["start_ttl"] + now) & 0xFFFFFFFF) <= e <= ((r["stop_ttl"] + now) & 0xFFFFFFFF)
        return False
