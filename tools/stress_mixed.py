#!/usr/bin/env python3
"""Mixed-operation GPU stress: randomized interleavings of every C-ABI
entry point on one handle — writes, flushes, point/range/batched reads with
the serving lanes forced on, pipelined count scans, paged scans, real
compactions, checkpoints and restores — cross-checked against the CPU
oracle driven with the same operation stream.  Complements tools/soak.py
(which focuses on compaction/read parity) with the round-2 surfaces.

Usage: python tools/stress_mixed.py [steps=600] [seed=11]
"""
import os
import random
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from incubator_pegasus_amd import data as D  # noqa: E402
from incubator_pegasus_amd.capi import OK, RrdbLib  # noqa: E402


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 600
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 11
    rnd = random.Random(seed)
    now = 1000
    hip = RrdbLib(os.path.join(REPO, "incubator_pegasus_amd", "csrc", "librrdb_hip.so"))
    orc = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so"))
    g = hip.open(1, 0, 0)
    o = orc.open(1, 0, -1)
    envs = {"engine.mg_persist": "on",
            "rocksdb.max_iteration_count": str(2**31 - 1)}
    g.set_envs(envs)
    o.set_envs({"rocksdb.max_iteration_count": str(2**31 - 1)})
    checks = 0
    tmp = tempfile.mkdtemp(prefix="stress_ckpt_")
    tmp_o = tempfile.mkdtemp(prefix="stress_ckpt_o_")
    decree = 0
    for step in range(steps):
        op = rnd.randrange(10)
        if op <= 1:  # writes
            for _ in range(rnd.randrange(1, 25)):
                hk = b"hk%03d" % rnd.randrange(60)
                sk = b"s%02d" % rnd.randrange(6)
                if rnd.random() < 0.15:
                    g.remove(hk, sk)
                    o.remove(hk, sk)
                else:
                    v = b"v%d" % step
                    ttl = rnd.choice([0, 0, 0, now + 50])
                    g.put(hk, sk, v, ttl, now)
                    o.put(hk, sk, v, ttl, now)
        elif op == 2:  # point gets
            for _ in range(8):
                k = D.generate_key(b"hk%03d" % rnd.randrange(70), b"s%02d" % rnd.randrange(7))
                assert g.get(k, now) == o.get(k, now), k
                checks += 1
        elif op == 3:  # multi_get burst (keeps the resident lane warm)
            hk = b"hk%03d" % rnd.randrange(70)
            kw = dict(reverse=rnd.random() < 0.3,
                      max_kv_count=rnd.choice([-1, 1, 3]))
            for _ in range(rnd.randrange(1, 6)):
                assert g.multi_get(hk, now, **kw) == o.multi_get(hk, now, **kw), (hk, kw)
                checks += 1
        elif op == 4:  # batched multi_get
            hks = [b"hk%03d" % rnd.randrange(70) for _ in range(16)]
            eg, gg = g.multi_get_batch(hks, now)
            eo, go = o.multi_get_batch(hks, now)
            assert (eg, gg) == (eo, go)
            checks += 1
        elif op == 5:  # pipelined count
            rc_g = g.scan_count_begin(b"\x00\x00", b"\xff\xff", now,
                                      validate_partition_hash=False)
            rc_o = o.scan_count_begin(b"\x00\x00", b"\xff\xff", now,
                                      validate_partition_hash=False)
            assert rc_g == rc_o == OK
            assert g.scan_count_finish() == o.scan_count_finish()
            checks += 1
        elif op == 6:  # paged scan
            rg = g.scan_open(b"\x00\x00", b"\xff\xff", now, batch_size=9,
                             validate_partition_hash=False)
            ro = o.scan_open(b"\x00\x00", b"\xff\xff", now, batch_size=9,
                             validate_partition_hash=False)
            assert rg.error == ro.error == OK
            assert rg.kvs == ro.kvs
            while rg.context_id != -1:
                assert ro.context_id != -1
                rg = g.scan_next(rg.context_id, now)
                ro = o.scan_next(ro.context_id, now)
                assert rg.error == ro.error == OK and rg.kvs == ro.kvs
            assert ro.context_id == -1
            checks += 1
        elif op == 7:  # real compaction
            assert g.manual_compact(now) == o.manual_compact(now)
            checks += 1
        elif op == 8 and g.num_records() > 0:  # checkpoint + cross restore
            # checkpoint BOTH handles: checkpoint flushes the memtable, and
            # flush-point-dependent compact stats (input_records/shadowed)
            # only compare when both sides saw the same flush boundaries
            decree += 1
            assert g.checkpoint(tmp, decree) == OK
            assert o.checkpoint(tmp_o, decree) == OK
            r = hip.open(1, 1, 0)
            ro_ = orc.open(1, 1, -1)
            try:
                assert r.restore(tmp, decree) == OK
                assert ro_.restore(tmp_o, decree) == OK
                k = D.generate_key(b"hk%03d" % rnd.randrange(60), b"s00")
                assert r.get(k, now) == g.get(k, now) == ro_.get(k, now)
            finally:
                r.close()
                ro_.close()
            checks += 1
        else:  # sortkey_count
            hk = b"hk%03d" % rnd.randrange(70)
            assert g.sortkey_count(hk, now) == o.sortkey_count(hk, now), hk
            checks += 1
    g.close()
    o.close()
    print(f"STRESS-MIXED OK: {steps} steps, {checks} parity checks")


if __name__ == "__main__":
    main()
