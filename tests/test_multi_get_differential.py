"""Randomized differential coverage of the on_multi_get range variant
(pegasus_server_impl.cpp:540-799): the CPU oracle against the INDEPENDENT
Python restatement (tests/pymodel.py, which mirrors the reference's
iterator loop directly), and the HIP engine against the oracle on the same
trial shapes.

This caught a real round-2 bug: when the count limit filled exactly at the
inclusive boundary key, the reference's post-append c==0 break completes
(kOk) BEFORE the limit recheck, but the restatements consulted iterator
validity and returned kIncomplete."""
import random

import pytest

from incubator_pegasus_amd import data as D
from pymodel import Model

NOW = 1000
HKS = [b"a", b"bb", b"h0", b"h1"]
SKS = [b"", b"s1", b"s2", b"s3", b"zz"]


def _build(rnd, parts):
    model = Model()
    seq = 1
    for _ in range(rnd.randrange(1, 5)):
        recs = {}
        for _ in range(rnd.randrange(1, 15)):
            k = D.generate_key(rnd.choice(HKS), rnd.choice(SKS))
            kind = 1 if rnd.random() < 0.15 else 0
            ttl = rnd.choice([0, 0, 500, NOW + 9])
            v = D.encode_value(b"x%d" % seq, ttl, seq, 1) if kind == 0 else b"\x00" * 12
            recs[k] = (v, kind)
        rl = []
        for k in sorted(recs):
            v, kind = recs[k]
            rl.append((k, v, seq, kind))
            seq += 1
        for p in parts:
            p.ingest_run(rl)
        model.ingest(rl)
    return model


def _cases(rnd, n):
    for _ in range(n):
        yield rnd.choice(HKS + [b"miss"]), dict(
            start_sortkey=rnd.choice(SKS), stop_sortkey=rnd.choice(SKS),
            start_inclusive=rnd.random() < 0.5, stop_inclusive=rnd.random() < 0.5,
            reverse=rnd.random() < 0.5, max_kv_count=rnd.choice([-1, 1, 2, 3, 100]),
            max_kv_size=rnd.choice([-1, -1, 1, 7, 40]),
            no_value=rnd.random() < 0.25,
            sort_key_filter_type=rnd.choice([0, 0, 1, 2, 3]),
            sort_key_filter_pattern=rnd.choice([b"", b"s", b"1", b"zz"]))


@pytest.mark.parametrize("seed", range(8))
def test_multi_get_oracle_vs_model(oracle_lib, seed):
    rnd = random.Random(1000 + seed)
    p = oracle_lib.open(1, 0, -1)
    try:
        model = _build(rnd, [p])
        for hk, kw in _cases(rnd, 80):
            assert p.multi_get(hk, NOW, **kw) == model.multi_get(hk, NOW, **kw), (hk, kw)
    finally:
        p.close()


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(4))
def test_multi_get_engine_vs_oracle(oracle_lib, hip_lib, seed):
    """Same shapes through the HIP engine (fused kernel + serving lanes) —
    both serving toggles exercised."""
    rnd = random.Random(2000 + seed)
    o = oracle_lib.open(1, 0, -1)
    g = hip_lib.open(1, 0, 0)
    try:
        _build(rnd, [o, g])
        if seed % 2:
            g.set_envs({"engine.mg_persist": "on"})
        for hk, kw in _cases(rnd, 80):
            assert o.multi_get(hk, NOW, **kw) == g.multi_get(hk, NOW, **kw), (hk, kw)
    finally:
        o.close()
        g.close()


@pytest.mark.parametrize("seed", range(4))
def test_multi_get_sortkeys_oracle_vs_model(oracle_lib, seed):
    """Point-list variant: requested-order echo, miss/expired skips, and the
    check-before-append cap semantics (:841-845)."""
    rnd = random.Random(3000 + seed)
    p = oracle_lib.open(1, 0, -1)
    try:
        model = _build(rnd, [p])
        for _ in range(60):
            hk = rnd.choice(HKS + [b"miss"])
            sks = [rnd.choice(SKS + [b"nope"]) for _ in range(rnd.randrange(1, 8))]
            kw = dict(max_kv_count=rnd.choice([-1, 1, 2, 100]),
                      max_kv_size=rnd.choice([-1, -1, 1, 5, 30]),
                      no_value=rnd.random() < 0.3)
            got = p.multi_get(hk, NOW, sort_keys=sks, **kw)
            want = model.multi_get_sortkeys(hk, sks, NOW, **kw)
            assert got == want, (hk, sks, kw)
    finally:
        p.close()
