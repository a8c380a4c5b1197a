#!/usr/bin/env python3
"""Long-running randomized differential soak: HIP engine vs CPU oracle.

Same op mix as tests/test_gpu_parity.py::test_mixed_op_soak but with a
configurable step count, wider keyspace, and periodic progress/HBM
reporting — stability evidence (arena growth, leaks, drift) beyond the
fast in-suite soak.  Oracle use here is as the checker only (test
infrastructure; the product path is the HIP engine).

Usage (GPU box):  python tools/soak.py --steps 3000 --seed 1 [--scale 200]
"""
import argparse
import os
import random
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from incubator_pegasus_amd import data as D  # noqa: E402
from incubator_pegasus_amd import hip_lib  # noqa: E402
from incubator_pegasus_amd.capi import (FT_MATCH_PREFIX, OK, SCAN_COMPLETED,  # noqa: E402
                                        RrdbLib)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def drain(part, now, **kw):
    out, counts = [], 0
    res = part.scan_open(b"\x00\x00", b"\xff\xff", now, **kw)
    assert res.error == OK
    out.extend(res.kvs)
    ets = list(res.expire_ts or [])
    while res.context_id != SCAN_COMPLETED:
        res = part.scan_next(res.context_id, now)
        assert res.error == OK
        out.extend(res.kvs)
        ets.extend(res.expire_ts or [])
    return out, ets


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2000)
    ap.add_argument("--seed", type=int, default=20260915)
    ap.add_argument("--scale", type=int, default=200, help="distinct hashkeys")
    args = ap.parse_args()

    rnd = random.Random(args.seed)
    o = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so")).open(1, 0, -1)
    glib = hip_lib()
    g = glib.open(1, 0, 0)
    now = 1000
    seq = [1_000_000]
    ckdir = tempfile.mkdtemp(prefix="soak_ck_")

    def hk(i):
        return f"soak{i:05d}".encode()

    def sk(i):
        return f"s{i:02d}".encode()

    checks = 0
    try:
        for step in range(args.steps):
            op = rnd.randrange(100)
            if op < 35:
                h, s = hk(rnd.randrange(args.scale)), sk(rnd.randrange(8))
                v = f"v{step}".encode() * rnd.randrange(1, 6)
                exp = 0 if rnd.random() < 0.9 else now + rnd.randrange(1, 50)
                for p in (o, g):
                    p.put(h, s, v, exp, now)
            elif op < 45:
                h, s = hk(rnd.randrange(args.scale)), sk(rnd.randrange(8))
                for p in (o, g):
                    p.remove(h, s)
            elif op < 50:
                for p in (o, g):
                    p.flush()
            elif op < 60:
                h, s = hk(rnd.randrange(args.scale)), sk(rnd.randrange(8))
                k = D.generate_key(h, s)
                assert o.get(k, now) == g.get(k, now), step
                checks += 1
            elif op < 70:
                h = hk(rnd.randrange(args.scale))
                kw = {}
                if rnd.random() < 0.3:
                    kw = dict(reverse=True)
                elif rnd.random() < 0.3:
                    kw = dict(sort_key_filter_type=FT_MATCH_PREFIX,
                              sort_key_filter_pattern=b"s0")
                assert o.multi_get(h, now, **kw) == g.multi_get(h, now, **kw), step
                checks += 1
            elif op < 78:
                h = hk(rnd.randrange(args.scale))
                assert o.sortkey_count(h, now) == g.sortkey_count(h, now), step
                checks += 1
            elif op < 86:
                bs = rnd.choice([3, 17, 1000])
                assert drain(o, now, batch_size=bs, validate_partition_hash=False) == \
                       drain(g, now, batch_size=bs, validate_partition_hash=False), step
                checks += 1
            elif op < 92:
                envs = {"default_ttl": str(rnd.choice([0, 1234]))}
                for p in (o, g):
                    p.set_envs(envs)
                assert o.manual_compact(now) == g.manual_compact(now), step
                checks += 1
            elif op < 96:
                n = rnd.randrange(1, 50)
                ks = sorted({D.generate_key(hk(rnd.randrange(args.scale)),
                                            sk(rnd.randrange(8))) for _ in range(n)})
                seq[0] = max(seq[0], 1_000_000 + step * 10_000)
                recs = []
                for k in ks:
                    seq[0] += 1
                    recs.append((k, D.encode_value(f"ing{step}".encode(), 0, 0, 1),
                                 seq[0], 0))
                for p in (o, g):
                    p.ingest_run(recs)
            else:
                d = os.path.join(ckdir, f"ck{step}")
                assert o.checkpoint(d, step) == 0, step
                # checkpoint the engine too: checkpoint flushes the memtable,
                # and flush-point-dependent compact stats (input_records /
                # shadowed) only compare when both sides saw the same flush
                # boundaries (see tools/stress_mixed.py, DESIGN parity notes)
                assert g.checkpoint(os.path.join(ckdir, f"ckg{step}"), step) == 0, step
                o2 = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so")).open(1, 0, -1)
                g2 = glib.open(1, 0, 0)
                try:
                    assert o2.restore(d, step) == 0, step
                    assert g2.restore(d, step) == 0, step
                    assert drain(o2, now, validate_partition_hash=False) == \
                           drain(g2, now, validate_partition_hash=False), step
                    checks += 1
                finally:
                    o2.close()
                    g2.close()
            if (step + 1) % 500 == 0:
                print(f"step {step + 1}/{args.steps}: {checks} parity checks, "
                      f"runs={g.num_runs()} records={g.num_records()}", flush=True)
        rows_o = drain(o, now, validate_partition_hash=False, return_expire_ts=True)
        rows_g = drain(g, now, validate_partition_hash=False, return_expire_ts=True)
        assert rows_o == rows_g
        print(f"SOAK OK: {args.steps} steps, {checks} parity checks, "
              f"final rows={len(rows_g[0])}")
    finally:
        o.close()
        g.close()


if __name__ == "__main__":
    main()
