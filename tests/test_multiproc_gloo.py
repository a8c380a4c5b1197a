"""Multi-process coverage of the §8(e) cross-partition aggregation path:
world_size-2 gloo on CPU, each rank scans its own partitions (CPU oracle as
the stand-in engine) and the full-table count is combined with
torch.distributed all_reduce — the same collective bench.py issues over
RCCL/xGMI on the GPU box."""
import os
import sys

import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, port, q):
    import torch
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from incubator_pegasus_amd import data as D
        from incubator_pegasus_amd.capi import RrdbLib

        lib = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so"))
        now = 1000
        local_rows = 0
        # two partitions per rank, partition ids disjoint across ranks
        for pidx in (rank * 2, rank * 2 + 1):
            part = lib.open(1, pidx, -1)
            n = 50 + 10 * pidx
            recs = [(D.generate_key(f"r{rank}p{pidx}k{i:04d}".encode(), b""),
                     D.encode_value(b"v", 0, i + 1, 1), i + 1, 0) for i in range(n)]
            part.ingest_run(recs)
            res = part.scan_open(b"\x00\x00", b"\xff\xff", now, only_return_count=True,
                                 full_scan=True, validate_partition_hash=False,
                                 batch_size=2**31 - 1)
            assert res.error == 0
            local_rows += res.kv_count
            part.close()
        t = torch.tensor([local_rows], dtype=torch.int64)
        dist.all_reduce(t)
        q.put((rank, local_rows, int(t.item())))
    finally:
        dist.destroy_process_group()


def _worker_c45(rank, world, port, q):
    """configs[4]/[5] shape on CPU: 64 partitions over `world` ranks, the
    rules-profile compaction (configs[5]) run per partition (pure replication,
    no exchange), then the full-table count_data all_reduce (configs[4]) via
    the pipelined scan_count API."""
    import json

    import torch
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from incubator_pegasus_amd import data as D
        from incubator_pegasus_amd.capi import RrdbLib

        lib = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so"))
        now = 100000
        ops = {"ops": [
            {"type": "COT_DELETE", "params": "", "rules": [
                {"type": "FRT_HASHKEY_PATTERN",
                 "params": json.dumps({"pattern": "7", "match_type": "SMT_MATCH_ANYWHERE"})}]},
        ]}
        envs = {"default_ttl": "3600", "user_specified_compaction": json.dumps(ops),
                "rocksdb.max_iteration_count": str(2**31 - 1)}
        n_parts_total = 64
        per_rank = n_parts_total // world
        local_rows = 0
        local_filtered = 0
        parts = []
        for pidx in range(rank * per_rank, (rank + 1) * per_rank):
            part = lib.open(1, pidx, -1)
            part.set_envs(envs)
            n = 40 + pidx  # distinct per partition
            recs = [(D.generate_key(f"p{pidx:02d}k{i:04d}".encode(), b""),
                     D.encode_value(b"v", 0, i + 1, 1), i + 1, 0) for i in range(n)]
            part.ingest_run(recs)
            recs2 = [(D.generate_key(f"p{pidx:02d}k{i:04d}".encode(), b""),
                      D.encode_value(b"w", 0, 1000 + i, 1), 1000 + i, 0)
                     for i in range(0, n, 3)]
            part.ingest_run(recs2)
            # configs[5]: rules compaction per partition (replicated work)
            err, st = part.manual_compact(now)
            assert err == 0
            local_filtered += st.filtered
            parts.append(part)
        # configs[4]: full-table count via the pipelined API + all_reduce
        for part in parts:
            rc = part.scan_count_begin(b"\x00\x00", b"\xff\xff", now,
                                       validate_partition_hash=False)
            assert rc == 0
        for part in parts:
            err, cnt = part.scan_count_finish()
            assert err == 0
            local_rows += cnt
        for part in parts:
            part.close()
        t = torch.tensor([local_rows, local_filtered], dtype=torch.int64)
        dist.all_reduce(t)
        q.put((rank, local_rows, int(t[0].item()), int(t[1].item())))
    finally:
        dist.destroy_process_group()


def test_configs45_shape_world8():
    """§8(e) readiness for the driver's 8-GPU node: 64 partitions, 8 ranks,
    rules-profile compaction + count_data reduce (gloo stand-in for RCCL)."""
    port = 29741
    world = 8
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_c45, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(world)]
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    # expected totals: every partition p holds 40+p keys, minus keys whose
    # decimal id contains '7' (deleted by the rule) — totals agree on all
    # ranks and with a local recomputation
    expect_total = 0
    for pidx in range(64):
        for i in range(40 + pidx):
            if "7" not in f"p{pidx:02d}k{i:04d}":
                expect_total += 1
    totals = {t for _, _, t, _ in results}
    assert totals == {expect_total}
    filtered_totals = {f for _, _, _, f in results}
    assert len(filtered_totals) == 1
    total_keys = sum(40 + p for p in range(64))
    assert filtered_totals.pop() == total_keys - expect_total


def test_count_reduce_world2():
    port = 29731
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    # expected: rank0 parts 0,1 -> 50+60; rank1 parts 2,3 -> 70+80
    by_rank = {r: (local, total) for r, local, total in results}
    assert by_rank[0][0] == 110 and by_rank[1][0] == 150
    assert by_rank[0][1] == by_rank[1][1] == 260
