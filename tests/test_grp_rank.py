"""GPU parity for the group-streaming rank (engine.rank_mode=grp, the
round-2 default): anchor-key groups partition the runs disjointly, each
workgroup streams its group's tail words through LDS once, ranks locally and
writes the contiguous global rank range.  Must be bit-exact against the CPU
oracle on every shape, including the branches the main suite cannot reach:

  - oversized groups (duplicate-clustered data) -> in-kernel global-probe
    fallback inside the same segment windows
  - equal-key sets split across a group boundary (anchor key present in
    several runs) -> the s_btail previous-segment shadow check
  - the scan view build (launch_rank_grp_view) via full drains

Reference semantics under test: the compaction-iterator newest-wins /
tombstone / filter behavior behind do_manual_compact
(/root/reference/src/server/pegasus_server_impl.cpp:3373-3420) and the
scan handlers (:1151-1547)."""
import numpy as np
import pytest

from incubator_pegasus_amd import data as D
from incubator_pegasus_amd.capi import OK, SCAN_COMPLETED

pytestmark = pytest.mark.gpu

NOW = 1000


def _drain(part, now, **kw):
    out = []
    res = part.scan_open(b"\x00\x00", b"\xff\xff", now, **kw)
    assert res.error == OK
    out.extend(res.kvs)
    while res.context_id != SCAN_COMPLETED:
        res = part.scan_next(res.context_id, now)
        assert res.error == OK
        out.extend(res.kvs)
    return out


def _ingest_arrays(parts, runs):
    for part in parts:
        for r in runs:
            part.ingest_run_arrays(np.ascontiguousarray(r["keys"]), r["koff"],
                                   np.ascontiguousarray(r["vals"]), r["voff"], r["sk"])


@pytest.mark.parametrize("n_keys", [2_000, 300_000])
def test_compact_parity_grp_rank(oracle_lib, hip_lib, n_keys):
    runs = D.build_point_table_runs(n_keys, 6, seed=D.DEFAULT_SEED + 177,
                                    dup_fraction=0.12, delete_fraction=0.03,
                                    ttl_fraction=0.05, ttl_expire_ts=500)
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        g.set_envs({"engine.rank_mode": "grp"})
        _ingest_arrays((o, g), runs)
        so = o.manual_compact(NOW)
        sg = g.manual_compact(NOW)
        assert so == sg
        assert o.num_records() == g.num_records()
        for k in [bytes(x) for x in D.make_raw_keys(D.zipfian_ids(300, n_keys, seed=7))]:
            assert o.get(k, NOW) == g.get(k, NOW), k
        assert _drain(o, NOW, validate_partition_hash=False) == \
               _drain(g, NOW, validate_partition_hash=False)
    finally:
        o.close()
        g.close()


def test_grp_rank_identical_runs_boundary_splits(oracle_lib, hip_lib):
    """Every run holds the SAME keys: every anchor key exists in all runs, so
    every group boundary splits an equal-key set — the worst case for the
    previous-segment shadow check."""
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        g.set_envs({"engine.rank_mode": "grp"})
        seq = 1
        for run in range(5):
            keys = D.make_raw_keys(np.arange(3000, dtype=np.uint64))
            recs = []
            for j in range(3000):
                recs.append((bytes(keys[j]),
                             D.encode_value(b"r%dv%d" % (run, j), 0, seq, 1), seq, 0))
                seq += 1
            o.ingest_run(recs)
            g.ingest_run(recs)
        so = o.manual_compact(NOW)
        sg = g.manual_compact(NOW)
        assert so == sg
        assert so[1].shadowed == 4 * 3000
        assert _drain(o, NOW, validate_partition_hash=False) == \
               _drain(g, NOW, validate_partition_hash=False)
    finally:
        o.close()
        g.close()


def test_grp_rank_oversized_group_fallback(oracle_lib, hip_lib):
    """Run 1's keys all cluster between two anchor keys of run 0 (the anchor
    run): one group far exceeds GRP_CAP and takes the in-kernel global-probe
    fallback; the rest stay staged."""
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        g.set_envs({"engine.rank_mode": "grp"})
        # run 0 (anchor: largest): 10000 keys spaced 1e6 apart
        ids0 = np.arange(10000, dtype=np.uint64) * np.uint64(1000000)
        keys0 = D.make_raw_keys(ids0)
        recs0 = [(bytes(keys0[j]), D.encode_value(b"a%d" % j, 0, j + 1, 1), j + 1, 0)
                 for j in range(len(ids0))]
        # run 1: 5000 keys ALL between two consecutive run-0 keys -> one
        # group of ~5000+stride records, far over GRP_CAP
        ids1 = np.uint64(500000) + np.arange(5000, dtype=np.uint64)
        keys1 = D.make_raw_keys(ids1)
        s0 = len(ids0) + 1
        recs1 = [(bytes(keys1[j]), D.encode_value(b"b%d" % j, 0, s0 + j, 1), s0 + j, 0)
                 for j in range(len(ids1))]
        for part in (o, g):
            part.ingest_run(recs0)
            part.ingest_run(recs1)
        so = o.manual_compact(NOW)
        sg = g.manual_compact(NOW)
        assert so == sg
        assert so[1].output_records == len(ids0) + len(ids1)
        assert _drain(o, NOW, validate_partition_hash=False) == \
               _drain(g, NOW, validate_partition_hash=False)
    finally:
        o.close()
        g.close()


def test_grp_rank_many_runs_uses_fallback_mode(oracle_lib, hip_lib):
    """R > LDST_MAXR (32): grp ineligible, host dispatches the standard
    kernel — results must still match.  (A 17-run compact regression also
    lives here: the overlay run once pushed R one past the old 16-run org
    limit and silently fell back 6x slower.)"""
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        g.set_envs({"engine.rank_mode": "grp"})
        seq = 1
        for run in range(34):
            ids = np.arange(run, 2000, 1, dtype=np.uint64)[::34][:50]
            keys = D.make_raw_keys(np.sort(ids))
            recs = []
            for j in range(len(keys)):
                recs.append((bytes(keys[j]), D.encode_value(b"m%d" % run, 0, seq, 1), seq, 0))
                seq += 1
            o.ingest_run(recs)
            g.ingest_run(recs)
        assert o.manual_compact(NOW) == g.manual_compact(NOW)
        assert _drain(o, NOW, validate_partition_hash=False) == \
               _drain(g, NOW, validate_partition_hash=False)
    finally:
        o.close()
        g.close()


def test_mg_graph_lane_parity(oracle_lib, hip_lib):
    """The hipGraph-captured serving lane must return byte-identical
    multi_get results to the plain path and the oracle, across run-set
    changes (graph recapture) and both toggle states."""
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        seq = 1
        for run in range(3):
            recs = []
            for i in range(run, 120, 3):
                for sk in (b"s0", b"s1", b"s2"):
                    recs.append((D.generate_key(b"gk%03d" % i, sk),
                                 D.encode_value(b"r%dv%d" % (run, i), 0, seq, 1), seq, 0))
                    seq += 1
            recs.sort(key=lambda r: r[0])
            o.ingest_run(recs)
            g.ingest_run(recs)
        cases = [dict(), dict(reverse=True), dict(max_kv_count=2),
                 dict(start_sortkey=b"s1"), dict(no_value=True),
                 dict(sort_key_filter_type=2, sort_key_filter_pattern=b"s2")]
        for mode in ("on", "off"):
            g.set_envs({"engine.mg_graph": mode})
            for i in (0, 1, 2, 55, 119, 999):
                hk = b"gk%03d" % i
                for kw in cases:
                    assert o.multi_get(hk, NOW, **kw) == g.multi_get(hk, NOW, **kw), \
                        (mode, hk, kw)
        # run-set change invalidates and recaptures the graph
        g.set_envs({"engine.mg_graph": "on"})
        o.put(b"gk000", b"s9", b"new")
        g.put(b"gk000", b"s9", b"new")
        assert o.multi_get(b"gk000", NOW) == g.multi_get(b"gk000", NOW)
        o.manual_compact(NOW)
        g.manual_compact(NOW)
        assert o.multi_get(b"gk055", NOW) == g.multi_get(b"gk055", NOW)
    finally:
        o.close()
        g.close()


def test_mg_persistent_server_parity(oracle_lib, hip_lib):
    """engine.mg_persist=on forces the resident serving kernel; results must
    stay byte-identical to the oracle across bursts, run-set changes
    (server retire + relaunch), idle expiries and the off toggle."""
    import time as _t

    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        seq = 1
        recs = []
        for i in range(200):
            for sk in (b"p", b"q"):
                recs.append((D.generate_key(b"sv%04d" % i, sk),
                             D.encode_value(b"v%d" % i, 0, seq, 1), seq, 0))
                seq += 1
        recs.sort(key=lambda r: r[0])
        o.ingest_run(recs)
        g.ingest_run(recs)
        g.set_envs({"engine.mg_persist": "on"})
        # burst: keeps the resident kernel warm across calls
        for rep in range(3):
            for i in (0, 7, 50, 199, 4444):
                hk = b"sv%04d" % i
                assert o.multi_get(hk, NOW) == g.multi_get(hk, NOW), (rep, hk)
        # idle past the 2ms window: the kernel self-exits; next call relaunches
        _t.sleep(0.05)
        assert o.multi_get(b"sv0007", NOW) == g.multi_get(b"sv0007", NOW)
        # run-set change retires the server before buffers are touched
        o.put(b"sv0007", b"r", b"new")
        g.put(b"sv0007", b"r", b"new")
        assert o.multi_get(b"sv0007", NOW) == g.multi_get(b"sv0007", NOW)
        o.manual_compact(NOW)
        g.manual_compact(NOW)
        for i in (0, 7, 199):
            hk = b"sv%04d" % i
            assert o.multi_get(hk, NOW) == g.multi_get(hk, NOW), hk
        # off toggle retires it; plain path still agrees
        g.set_envs({"engine.mg_persist": "off"})
        assert o.multi_get(b"sv0050", NOW) == g.multi_get(b"sv0050", NOW)
    finally:
        o.close()
        g.close()


def test_prefix_bloom_parity_and_toggle(oracle_lib, hip_lib):
    """§8(f)3 second half: the hashkey-prefix bloom is lossless — multi_get
    and sortkey_count agree with the oracle for present and ABSENT hashkeys,
    filter on and off (reference prefix-extractor bloom,
    pegasus_server_impl_init.cpp:816-841, hashkey_transform.h:40-54)."""
    for filt in ("common", "none"):
        o = oracle_lib.open(1, 0, -1)
        g = hip_lib.open(1, 0, 0)
        try:
            g.set_envs({"rocksdb.filter_type": filt})
            seq = 1
            for run in range(5):
                recs = []
                for i in range(run, 300, 5):
                    for sk in (b"a", b"b", b"c"):
                        recs.append((D.generate_key(b"phk%04d" % i, sk),
                                     D.encode_value(b"v%d" % run, 0, seq, 1), seq, 0))
                        seq += 1
                recs.sort(key=lambda r: r[0])
                o.ingest_run(recs)
                g.ingest_run(recs)
            # hashkeys present in some runs, absent from others, fully absent
            for i in list(range(0, 300, 7)) + [9999]:
                hk = b"phk%04d" % i
                assert o.multi_get(hk, NOW) == g.multi_get(hk, NOW), (filt, hk)
                assert o.sortkey_count(hk, NOW) == g.sortkey_count(hk, NOW), (filt, hk)
            # range-limited multi_get under one hashkey
            assert o.multi_get(b"phk0010", NOW, start_sortkey=b"a", stop_sortkey=b"b",
                               stop_inclusive=True) == \
                   g.multi_get(b"phk0010", NOW, start_sortkey=b"a", stop_sortkey=b"b",
                               stop_inclusive=True), filt
            # blooms rebuilt after compaction
            o.manual_compact(NOW)
            g.manual_compact(NOW)
            for i in (0, 5, 9999):
                hk = b"phk%04d" % i
                assert o.multi_get(hk, NOW) == g.multi_get(hk, NOW), (filt, "post", hk)
        finally:
            o.close()
            g.close()


def _count_fused_or_fallback(part, now, **kw):
    rc = part.scan_count_begin(b"\x00\x00", b"\xff\xff", now, **kw)
    if rc == 0:
        err, cnt = part.scan_count_finish()
        assert err == 0
        return cnt, True
    res = part.scan_open(b"\x00\x00", b"\xff\xff", now, only_return_count=True,
                         full_scan=True, batch_size=2**31 - 1, **kw)
    assert res.error == OK and res.context_id == SCAN_COMPLETED
    return res.kv_count, False


def test_fused_count_scan_parity(oracle_lib, hip_lib):
    """rrdb_scan_count_begin/finish (the pipelined count_data path) must
    count exactly what the paged count-only scan counts: TTL-expired,
    tombstoned and shadowed records excluded; filters and hash validation
    honored."""
    runs = D.build_point_table_runs(120_000, 6, seed=D.DEFAULT_SEED + 31,
                                    dup_fraction=0.15, delete_fraction=0.05,
                                    ttl_fraction=0.10, ttl_expire_ts=500)
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        for p in (o, g):
            p.set_envs({"rocksdb.max_iteration_count": str(2**31 - 1)})
        _ingest_arrays((o, g), runs)
        co, _ = _count_fused_or_fallback(o, NOW, validate_partition_hash=False)
        cg, fused = _count_fused_or_fallback(g, NOW, validate_partition_hash=False)
        assert fused, "expected the fused path on an eligible table"
        assert co == cg
        # cross-check against the paged path on the same engine
        res = g.scan_open(b"\x00\x00", b"\xff\xff", NOW, only_return_count=True,
                          full_scan=True, batch_size=2**31 - 1,
                          validate_partition_hash=False)
        assert res.kv_count == cg
        # with a hashkey prefix filter (count shape still supported)
        co2, _ = _count_fused_or_fallback(o, NOW, validate_partition_hash=False,
                                          hash_key_filter_type=2,
                                          hash_key_filter_pattern=b"u:0000000000")
        cg2, _ = _count_fused_or_fallback(g, NOW, validate_partition_hash=False,
                                          hash_key_filter_type=2,
                                          hash_key_filter_pattern=b"u:0000000000")
        assert co2 == cg2
    finally:
        o.close()
        g.close()


def test_fused_count_scan_shape_gate(hip_lib):
    """Unsupported shapes return kInvalidArgument from begin (caller falls
    back); a small batch cap must not silently change count semantics."""
    g = hip_lib.open(1, 0, 0)
    try:
        runs = D.build_point_table_runs(5_000, 3, seed=1)
        for r in runs:
            g.ingest_run_arrays(np.ascontiguousarray(r["keys"]), r["koff"],
                                np.ascontiguousarray(r["vals"]), r["voff"], r["sk"])
        # default engine max_iteration_count (1000) < total -> gate
        rc = g.scan_count_begin(b"\x00\x00", b"\xff\xff", NOW)
        assert rc != 0
        g.set_envs({"rocksdb.max_iteration_count": str(2**31 - 1)})
        rc = g.scan_count_begin(b"\x00\x00", b"\xff\xff", NOW, batch_size=10)
        assert rc != 0
        rc = g.scan_count_begin(b"\x00\x00", b"\xff\xff", NOW)
        assert rc == 0
        err, cnt = g.scan_count_finish()
        res = g.scan_open(b"\x00\x00", b"\xff\xff", NOW, only_return_count=True,
                          full_scan=True, batch_size=2**31 - 1)
        assert err == 0 and cnt == res.kv_count
    finally:
        g.close()


def test_grp_rank_rules_and_ttl(oracle_lib, hip_lib):
    """Fused filter outputs (default_ttl rewrite + user delete rule) under
    the group rank."""
    import json
    ops_json = json.dumps({"ops": [
        {"type": "COT_DELETE", "params": "",
         "rules": [{"type": "FRT_HASHKEY_PATTERN", "params": json.dumps(
             {"pattern": "u:000000000001", "match_type": "SMT_MATCH_PREFIX"})}]},
    ]})
    envs = {"user_specified_compaction": ops_json, "default_ttl": "1000"}
    runs = D.build_point_table_runs(30_000, 4, seed=D.DEFAULT_SEED + 9,
                                    dup_fraction=0.1, ttl_fraction=0.2, ttl_expire_ts=NOW + 50)
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        g.set_envs({"engine.rank_mode": "grp"})
        for p in (o, g):
            p.set_envs(envs)
        _ingest_arrays((o, g), runs)
        so = o.manual_compact(NOW)
        sg = g.manual_compact(NOW)
        assert so == sg
        assert so[1].filtered > 0
        assert _drain(o, NOW, validate_partition_hash=False) == \
               _drain(g, NOW, validate_partition_hash=False)
    finally:
        o.close()
        g.close()


def test_concurrent_threads_one_handle(hip_lib):
    """The reference serves reads concurrently across thread pools
    (replication.codes.h:44-50); the C-ABI takes a per-handle lock, so
    concurrent callers must serialize safely — including while the resident
    serving lane and compactions interleave."""
    import threading

    g = hip_lib.open(1, 0, 0)
    errs = []
    try:
        seq = 1
        recs = []
        for i in range(500):
            recs.append((D.generate_key(b"th%04d" % i, b"s"),
                         D.encode_value(b"v%d" % i, 0, seq, 1), seq, 0))
            seq += 1
        g.ingest_run(recs)
        g.set_envs({"engine.mg_persist": "on",
                    "rocksdb.max_iteration_count": str(2**31 - 1)})
        stop = threading.Event()

        def reader(tid):
            try:
                n = 0
                while not stop.is_set() and n < 300:
                    hk = b"th%04d" % ((tid * 37 + n) % 500)
                    st, kvs = g.multi_get(hk, NOW)
                    assert st == 0 and len(kvs) == 1, (st, hk)
                    n += 1
            except Exception as ex:  # noqa: BLE001
                errs.append(ex)

        def compactor():
            try:
                for _ in range(5):
                    err, _ = g.manual_compact(NOW, keep_inputs=True)
                    assert err == 0
            except Exception as ex:  # noqa: BLE001
                errs.append(ex)

        ts = [threading.Thread(target=reader, args=(i,)) for i in range(4)]
        ts.append(threading.Thread(target=compactor))
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=120)
        stop.set()
        assert not errs, errs
    finally:
        g.close()


def test_batched_device_out_counts(oracle_lib, hip_lib):
    """on_device_out batched multi_get: per-request counts/errors must match
    the host-marshalled run (the packed blob uses the same kernel whose
    bytes the host path validates)."""
    import ctypes as C

    from incubator_pegasus_amd.capi import _MultiGetRequest, _Result, _cslice, _pack

    g = hip_lib.open(1, 0, 0)
    try:
        seq = 1
        recs = []
        for i in range(300):
            for sk in (b"x", b"y"):
                recs.append((D.generate_key(b"dv%04d" % i, sk),
                             D.encode_value(b"v%d" % i, 0, seq, 1), seq, 0))
                seq += 1
        recs.sort(key=lambda r: r[0])
        g.ingest_run(recs)
        hks = [b"dv%04d" % i for i in range(0, 310, 7)]
        err, host_groups = g.multi_get_batch(hks, NOW)
        assert err == OK
        # raw device-out call
        import numpy as np
        blob = b"".join(hks)
        offs = np.zeros(len(hks) + 1, dtype=np.uint64)
        np.cumsum([len(h) for h in hks], out=offs[1:])
        keep = []
        req = _MultiGetRequest()
        req.hash_key = _cslice(b"", keep)
        req.start_inclusive = 1
        req.max_kv_count = -1
        req.max_kv_size = -1
        req.on_device_out = 1
        res = _Result()
        barr = np.frombuffer(blob, dtype=np.uint8)
        g._L.rrdb_multi_get_batch(
            g._h, len(hks), np.ascontiguousarray(barr).ctypes.data_as(C.c_void_p),
            offs.ctypes.data_as(C.c_void_p), C.byref(req), NOW, C.byref(res))
        try:
            assert res.error == OK
            assert res.dev_vals  # device path actually taken
            got = [(res.group_counts[i], res.group_errors[i]) for i in range(len(hks))]
            want = [(len(kvs), e) for e, kvs in host_groups]
            assert got == want
            assert res.count == sum(c for c, _ in want)
        finally:
            g._L.rrdb_free_result(C.byref(res))
    finally:
        g.close()


def test_batched_device_out_fallback_is_host_marshalled(hip_lib):
    """Best-effort contract: when any request of a device-out batch falls
    back to the general path, the call returns normal host slices and
    dev_vals stays null."""
    import ctypes as C
    import numpy as np

    from incubator_pegasus_amd.capi import _MultiGetRequest, _Result, _cslice

    g = hip_lib.open(1, 0, 0)
    try:
        recs = [(D.generate_key(b"fbk%02d" % i, b"s"),
                 D.encode_value(b"v%d" % i, 0, i + 1, 1), i + 1, 0) for i in range(20)]
        recs.sort(key=lambda r: r[0])
        g.ingest_run(recs)
        hks = [b"fbk%02d" % i for i in range(3)] + [b"L" * 5000]  # 5000B hk -> fallback
        blob = b"".join(hks)
        offs = np.zeros(len(hks) + 1, dtype=np.uint64)
        np.cumsum([len(h) for h in hks], out=offs[1:])
        keep = []
        req = _MultiGetRequest()
        req.hash_key = _cslice(b"", keep)
        req.start_inclusive = 1
        req.max_kv_count = -1
        req.max_kv_size = -1
        req.on_device_out = 1
        res = _Result()
        barr = np.frombuffer(blob, dtype=np.uint8)
        g._L.rrdb_multi_get_batch(
            g._h, len(hks), np.ascontiguousarray(barr).ctypes.data_as(C.c_void_p),
            offs.ctypes.data_as(C.c_void_p), C.byref(req), NOW, C.byref(res))
        try:
            assert res.error == OK
            assert not res.dev_vals  # host-marshalled best-effort path
            assert [res.group_counts[i] for i in range(4)] == [1, 1, 1, 0]
        finally:
            g._L.rrdb_free_result(C.byref(res))
    finally:
        g.close()


def test_fused_count_hash_validation(oracle_lib, hip_lib):
    """Fused count with partition-hash validation (validate_key_value_for_scan
    crc64 branch, pegasus_server_impl.cpp:2382-2432): counts only keys whose
    crc64(hashkey) & partition_version == pidx, matching the oracle and the
    paged path."""
    o = oracle_lib.open(1, 2, -1)  # pidx 2
    g = hip_lib.open(1, 2, 0)
    try:
        envs = {"replica.split.validate_partition_hash": "true",
                "rocksdb.max_iteration_count": str(2**31 - 1)}
        for p in (o, g):
            p.set_envs(envs)
            p.set_partition_version(3)  # 4-way mask
        recs = []
        for i in range(2000):
            recs.append((D.generate_key(b"hv%05d" % i, b""),
                         D.encode_value(b"v", 0, i + 1, 1), i + 1, 0))
        recs.sort(key=lambda r: r[0])
        o.ingest_run(recs)
        g.ingest_run(recs)
        co, fo = _count_fused_or_fallback(o, NOW, validate_partition_hash=True)
        cg, fg = _count_fused_or_fallback(g, NOW, validate_partition_hash=True)
        assert fg, "expected the fused path"
        assert co == cg
        assert 0 < cg < 2000  # the mask really filtered
        res = g.scan_open(b"\x00\x00", b"\xff\xff", NOW, only_return_count=True,
                          full_scan=True, batch_size=2**31 - 1,
                          validate_partition_hash=True)
        assert res.kv_count == cg
    finally:
        o.close()
        g.close()
