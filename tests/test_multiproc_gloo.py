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
