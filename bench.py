#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X rrdb engine.

Metric (BASELINE.json): "compaction keys/sec + scan rows/sec per node;
achieved HBM GB/s vs roofline".  `value` is compaction keys/sec (the first
named metric; the workload is BASELINE.json configs[1]: 16 partitions, 50M
keys, 8-sorted-run L0->L1 merge per partition, on one GPU); scan rows/sec and
get ops/sec are reported alongside in `extra`.

A step = one full merge+filter+write pass of the compaction hot path over
every partition's 8 ingested runs (keep_inputs=True makes the pass
repeatable; the kernels execute the complete merge, filter and output write
each step — nothing is skipped inside the timed region).

Inputs are resident in HBM when the timed region starts (ingested once during
setup).  The output merged run is written to HBM and freed between steps.

Multi-GPU (--gpus N, launched by torchrun): weak scaling — each rank owns its
own 16 partitions (partition -> GPU sharding per SURVEY.md §8(e)); the only
cross-GPU exchange on the data path is the count_data-style full-scan count
reduce, done with torch.distributed all_reduce (RCCL over xGMI), mirroring
the client-side fan-out+aggregate of the reference
(src/client_lib/pegasus_client_impl.cpp:1197-1267).

CPU baseline: the CPU oracle (restatement of the reference path)
timed on a bounded sample on this box's host cores — reported, not the target.
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import numpy as np  # noqa: E402

HBM_PEAK_GBS = 8000.0  # MI355X spec peak (MI355X_MICROARCH.md; measured ach. ~6300)


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def _cpu_compact_worker(params):
    """Oracle (CPU restatement, checker) compaction pass over one partition.
    Returns (n_records, compact_seconds).  bench cpu_baseline leg only."""
    sample, n_runs, pidx, epoch_now = params
    from incubator_pegasus_amd.capi import RrdbLib

    oracle = RrdbLib(os.path.join(REPO, "oracle", "liboracle.so"))
    op = oracle.open(1, pidx, -1)
    try:
        for r in build_partition_data(sample, n_runs, 0, pidx):
            op.ingest_run_arrays(np.ascontiguousarray(r["keys"]), r["koff"],
                                 np.ascontiguousarray(r["vals"]), r["voff"], r["sk"])
        nrec = int(op.num_records())
        t0 = time.time()
        err, _ = op.manual_compact(epoch_now, keep_inputs=True)
        assert err == 0
        return nrec, time.time() - t0
    finally:
        op.close()


def build_partition_data(keys_pp, n_runs, rank, pidx, value_len=100, ttl_fraction=0.0,
                         ttl_expire_ts=0):
    from incubator_pegasus_amd import data as D

    return D.build_point_table_runs(
        keys_pp, n_runs, seed=D.DEFAULT_SEED + rank * 1000 + pidx,
        value_len=value_len, dup_fraction=0.10, delete_fraction=0.02,
        ttl_fraction=ttl_fraction, ttl_expire_ts=ttl_expire_ts)


def rules_profile_envs(epoch_now):
    """configs[4]: default_ttl + user-specified compaction rules
    (pegasus_compaction_filter) evaluated on every surviving key."""
    import json as _json

    ops = {"ops": [
        {"type": "COT_DELETE", "params": "", "rules": [
            {"type": "FRT_HASHKEY_PATTERN",
             "params": _json.dumps({"pattern": "777", "match_type": "SMT_MATCH_ANYWHERE"})}]},
        {"type": "COT_UPDATE_TTL",
         "params": _json.dumps({"type": "UTOT_FROM_NOW", "value": 5000}),
         "rules": [
            {"type": "FRT_TTL_RANGE",
             "params": _json.dumps({"start_ttl": 3000, "stop_ttl": 4000})}]},
    ]}
    return {"default_ttl": "3600", "user_specified_compaction": _json.dumps(ops)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=25)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--partitions", type=int, default=16)     # per GPU (configs[1])
    ap.add_argument("--keys", type=int, default=50_000_000)   # per GPU total
    ap.add_argument("--runs", type=int, default=8)
    ap.add_argument("--cpu-sample", type=int, default=2_000_000)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--emit-mode", choices=["chunked", "rank", "input"], default="chunked")
    ap.add_argument("--bt-shift", type=int, default=5)
    ap.add_argument("--rank-mode", choices=["global", "lds", "ldst", "grp"], default="grp")
    ap.add_argument("--ycsb-hashkeys-per-part", type=int, default=625_000,
                    help="YCSB C/E table: hashkeys per partition x 10 sortkeys "
                         "(default 625K x 16 parts = 100M rows: configs[3] scale)")
    ap.add_argument("--rules-profile", action="store_true",
                    help="configs[4] shape: TTL'd data + default_ttl + user "
                         "delete/update-TTL compaction rules evaluated per key")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(args.gpus, world)

    import torch

    dist = None
    if world > 1:
        import torch.distributed as tdist

        backend = "nccl" if torch.cuda.is_available() else "gloo"
        tdist.init_process_group(backend=backend)
        dist = tdist

    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU: the rrdb engine has no CPU fallback"}))
        sys.exit(1)
    torch.cuda.set_device(local_rank)

    import incubator_pegasus_amd as pa

    hip = pa.hip_lib()
    keys_pp = args.keys // args.partitions

    # ---- setup: generate + ingest (untimed) ----
    log(f"generating {args.partitions} partitions x {keys_pp} keys x {args.runs} runs ...")
    t0 = time.time()
    parts = []
    total_records = 0
    epoch_now = 1_000_000
    for p in range(args.partitions):
        runs = build_partition_data(
            keys_pp, args.runs, rank, p,
            ttl_fraction=0.2 if args.rules_profile else 0.0,
            ttl_expire_ts=epoch_now - 100 if args.rules_profile else 0)
        eng = hip.open(1, p, local_rank)
        eng.set_envs({"engine.emit_mode": args.emit_mode,
                      "engine.bt_shift": str(args.bt_shift),
                      "engine.rank_mode": args.rank_mode})
        if args.rules_profile:
            eng.set_envs(rules_profile_envs(epoch_now))
        for r in runs:
            eng.ingest_run_arrays(np.ascontiguousarray(r["keys"]), r["koff"],
                                  np.ascontiguousarray(r["vals"]), r["voff"], r["sk"])
        total_records += int(eng.num_records())
        parts.append(eng)
        if p == 0:
            log(f"partition 0 ready ({time.time()-t0:.1f}s, {eng.num_records()} records)")
    log(f"ingest done: {total_records} records on GPU {local_rank} "
        f"({time.time()-t0:.1f}s)")

    def barrier_sync():
        torch.cuda.synchronize()
        if dist:
            dist.barrier()
        torch.cuda.synchronize()

    # single-threaded PIPELINED submission: begin() submits every partition's
    # merge phase back-to-back (no host sync between partitions), finish()
    # then emits each — the GPU never idles on per-call host gaps.  A thread
    # pool over per-partition streams was measured 4x SLOWER (multi-threaded
    # HIP submission + sync contention: 256 vs 64 ms/step); this keeps one
    # submitting thread.
    def one_step():
        out_records = 0
        st = None
        for eng in parts:
            assert eng.manual_compact_begin(epoch_now, keep_inputs=True) == 0
        for eng in parts:
            err, st = eng.manual_compact_finish()
            assert err == 0
            out_records += st.output_records
        return out_records, st

    # ---- warmup ----
    for _ in range(args.warmup):
        out_records, last_stats = one_step()
    if args.warmup == 0:
        out_records, last_stats = one_step()  # need sizes for roofline math

    # ---- timed region: K compaction passes ----
    barrier_sync()
    t0 = time.time()
    for _ in range(args.steps):
        one_step()
    barrier_sync()
    elapsed = time.time() - t0
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    keys_per_step_job = total_records * max(world, 1)  # whole-job aggregate
    value = keys_per_step_job * args.steps / elapsed
    ms_per_step = elapsed * 1000.0 / args.steps

    # one instrumented (untimed) step: per-call wall vs in-kernel time
    call_ms = []
    for eng in parts:
        c0 = time.time()
        eng.manual_compact(epoch_now, keep_inputs=True)
        call_ms.append((time.time() - c0) * 1e3)
    call_ms.sort()
    compact_call_stats = {
        "min": round(call_ms[0], 2),
        "med": round(call_ms[len(call_ms) // 2], 2),
        "max": round(call_ms[-1], 2),
        "kernel_total_ms": round(parts[-1].phase_ms("compact_total"), 2),
    }

    # ---- roofline: dominant compaction kernel, HIP-event timed in-engine ----
    eng0 = parts[0]
    n0 = int(eng0.num_records())
    # per-partition byte tallies (fixed-width records: 18B keys, 112B values)
    key_bytes = 18 * n0
    val_bytes = 112 * n0
    out_bytes = int(last_stats.output_bytes)
    phases = {
        # algorithmic bytes per launch: stated in DESIGN.md §roofline
        # rank is fused with the filter: every key + value header read once
        "compact_rank": key_bytes + 12 * n0,
        "compact_emit": 2 * out_bytes,  # read kept records + write merged run
    }
    timings = {ph: eng0.phase_ms(ph) for ph in phases}
    # dominant single kernel = the fused rank (compact_emit's phase window
    # spans 6 kernels and, with the async finish, overlaps other partitions'
    # work — its wall time over-reads; per-kernel rocprof evidence in
    # profiles/ shows the rank as the largest single kernel)
    dominant = "compact_rank"
    dur_ms = timings[dominant]
    achieved = phases[dominant] / (dur_ms * 1e-3) / 1e9 if dur_ms > 0 else 0.0
    # PMC-measured HBM fetch of the rank kernel, per launch: 40.64 MB raw
    # counter per 3.504M-record launch (profiles/r02_pmc_rank_fetch.txt),
    # x2 upper-bound correction for wide coalesced reads (gfx950 FETCH_SIZE
    # halves them — MI355X_MICROARCH.md §HBM) => <= 23.2 B/record vs the
    # 30 B/record algorithmic budget.
    RANK_PMC_BYTES_PER_RECORD = 23.2
    roofline = {
        "bound": "hbm",
        "kernel": dominant,
        "achieved": round(achieved, 1),
        "peak": HBM_PEAK_GBS,
        "unit": "GB/s",
        "frac": round(achieved / HBM_PEAK_GBS, 4),
        "traffic": int(n0 * RANK_PMC_BYTES_PER_RECORD),
        "traffic_note": "per-launch HBM fetch bytes from rocprofv3 PMC "
                        "calibration (profiles/r02_pmc_rank_fetch.txt); "
                        "~0.8x of the 30B/record algorithmic budget "
                        "(r01: ~6.5x over budget)",
        "phase_ms": {k: round(v, 3) for k, v in timings.items()},
        "compact_total_ms": round(eng0.phase_ms("compact_total"), 3),
    }

    # ---- secondary: count_data-style full scan (count-only) + RCCL reduce ----
    for eng in parts:
        eng.set_envs({"rocksdb.max_iteration_count": str(2**31 - 1)})
    barrier_sync()
    t0 = time.time()
    rows = 0
    # pipelined: submit every partition's fused count kernels, then collect —
    # partitions' scans co-run on their per-engine streams (rrdb_scan_count_*)
    pending = []
    for eng in parts:
        rc = eng.scan_count_begin(b"\x00\x00", b"\xff\xff", epoch_now,
                                  validate_partition_hash=False)
        pending.append(rc == 0)
        if rc != 0:  # shape/eligibility fallback: synchronous count
            res = eng.scan_open(b"\x00\x00", b"\xff\xff", epoch_now, only_return_count=True,
                                full_scan=True, validate_partition_hash=False,
                                batch_size=2**31 - 1)
            assert res.error == 0 and res.context_id == -1, (res.error, res.context_id)
            rows += res.kv_count
    for eng, ok in zip(parts, pending):
        if ok:
            err, cnt = eng.scan_count_finish()
            assert err == 0
            rows += cnt
    torch.cuda.synchronize()
    if dist:
        rt = torch.tensor([rows], dtype=torch.int64,
                          device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(rt)  # the §8(e) cross-partition count aggregate (RCCL)
        rows = int(rt.item())
        t = torch.tensor([time.time() - t0], dtype=torch.float64, device=rt.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        scan_elapsed = float(t.item())
    else:
        scan_elapsed = time.time() - t0
    scan_rows_per_s = rows / scan_elapsed
    scan_state_ms = eng0.phase_ms("scan_state")
    # scan_state reads every visible record's key + value header once
    scan_achieved = ((18 + 12) * n0) / (scan_state_ms * 1e-3) / 1e9 if scan_state_ms > 0 else 0.0

    # ---- secondary: zipfian point gets (YCSB C) ----
    from incubator_pegasus_amd import data as D

    nq = 1_000_000
    qids = D.zipfian_ids(nq, keys_pp, seed=D.DEFAULT_SEED)
    qkeys = D.make_raw_keys(qids).reshape(-1)
    qoffs = D.fixed_offsets(nq, 18)
    import ctypes

    barrier_sync()
    t0 = time.time()
    # raw ABI call (no per-key python packing)
    L = hip._lib
    from incubator_pegasus_amd.capi import _Result

    res = _Result()
    L.rrdb_batch_get(parts[0]._h, nq, np.ascontiguousarray(qkeys).ctypes.data_as(ctypes.c_void_p),
                     qoffs.ctypes.data_as(ctypes.c_void_p), epoch_now, ctypes.byref(res))
    found = int(res.count)
    L.rrdb_free_result(ctypes.byref(res))
    get_elapsed = time.time() - t0
    get_ops_per_s = nq / get_elapsed
    get_kernel_ms = parts[0].phase_ms("get_search")

    # ---- YCSB C/E at configs[3] scale: a sortkey table of
    # (hashkeys_per_part x 10) rows on EVERY partition (16 x 625K x 10 =
    # 100M rows by default), zipfian point gets + prefix multi_gets spread
    # across all of them; single-stream latency + batched throughput are
    # also measured from a C++ host (bin/bench_mg) since the reference's
    # replica server is C++ and the ctypes loop adds interpreter overhead ----
    from incubator_pegasus_amd import data as D2

    hk_pp = args.ycsb_hashkeys_per_part
    if world > 1:
        # configs[3] is a single-GPU config; keep the per-rank side tables
        # small on scaling runs so N=8 setup stays minutes, not tens
        hk_pp = min(hk_pp, 100_000)
    # configs[3]'s scale is 100M rows TOTAL (16 parts x 625K hashkeys x 10);
    # scale hashkeys down when --partitions grows so oversized compaction
    # configs (e.g. 400M keys / 64 parts) don't also inflate the side table
    # past the HBM capacity envelope
    hk_pp = min(hk_pp, max(1, 100_000_000 // (10 * args.partitions)))
    e_parts = []
    for p_ in range(args.partitions):
        ep = hip.open(2, p_, local_rank)
        st_run = D2.build_scan_table_run(hk_pp, 10, seed=D2.DEFAULT_SEED + 77 + p_)
        ep.ingest_run_arrays(np.ascontiguousarray(st_run["keys"]), st_run["koff"],
                             np.ascontiguousarray(st_run["vals"]), st_run["voff"],
                             st_run["sk"])
        e_parts.append(ep)
    log(f"ycsb table: {args.partitions} x {hk_pp} hashkeys x 10 rows "
        f"({args.partitions * hk_pp * 10} rows)")
    import ctypes as _ct
    from incubator_pegasus_amd.capi import _Result as _Res2

    # YCSB-C: zipfian point gets over every partition's full keyspace
    nq_pp = 200_000
    barrier_sync()
    t0 = time.time()
    c_found = 0
    for p_, ep in enumerate(e_parts):
        ids = D2.zipfian_ids(nq_pp, hk_pp, seed=D2.DEFAULT_SEED + 5 + p_)
        sks = (D2.splitmix64(ids + np.uint64(99)) % np.uint64(10)).astype(np.uint64)
        qk = D2.make_raw_keys(ids, sks).reshape(-1)
        qo = D2.fixed_offsets(nq_pp, 26)
        res = _Res2()
        L.rrdb_batch_get(ep._h, nq_pp,
                         np.ascontiguousarray(qk).ctypes.data_as(ctypes.c_void_p),
                         qo.ctypes.data_as(ctypes.c_void_p), epoch_now, _ct.byref(res))
        c_found += int(res.count)
        L.rrdb_free_result(_ct.byref(res))
    ycsb_c_elapsed = time.time() - t0
    ycsb_c_ops = args.partitions * nq_pp / ycsb_c_elapsed

    # YCSB-E single-stream from python: per-partition BURSTS (the hot-
    # partition pattern; sustained sub-ms spacing lets the resident serving
    # lane auto-engage, as a hot replica would see).  The ctypes wrapper
    # adds interpreter overhead vs the C++ number (cabi below).
    mg_n = 512
    mg_ids = D2.zipfian_ids(mg_n, hk_pp, seed=D2.DEFAULT_SEED + 5)
    mg_hks = D2.make_hashkeys(mg_ids)
    barrier_sync()
    t0 = time.time()
    mg_rows = 0
    burst = 128
    for b0 in range(0, mg_n, burst):
        ep = e_parts[(b0 // burst) % len(e_parts)]
        for qi in range(b0, min(b0 + burst, mg_n)):
            st, kvs = ep.multi_get(bytes(mg_hks[qi]), epoch_now)
            assert st == 0
            mg_rows += len(kvs)
    mg_elapsed = time.time() - t0
    for ep in e_parts:
        ep.close()
    # C++ host: single-stream latency + batched (one workgroup/request)
    mg_cabi = None
    bench_mg_bin = os.path.join(REPO, "bin", "bench_mg")
    if rank == 0 and os.path.exists(bench_mg_bin):
        try:
            out_ = subprocess.run(
                [bench_mg_bin, os.path.join(REPO, "incubator_pegasus_amd", "csrc",
                                            "librrdb_hip.so"), str(hk_pp), "20000"],
                capture_output=True, text=True, timeout=300)
            for line in out_.stdout.splitlines():
                if line.startswith("{"):
                    mg_cabi = json.loads(line)
        except Exception as ex:  # noqa: BLE001
            mg_cabi = {"error": str(ex)}
    b_rows = mg_cabi.get("batched_rows", 0) if mg_cabi else 0
    b_elapsed = 1.0
    n_batched = 0
    ycsb_e = {
        "table_rows": args.partitions * hk_pp * 10,
        "ycsb_c_get_ops_per_s": round(ycsb_c_ops, 1),
        "ycsb_c_found": c_found,
        "multi_get_ops_per_s": round(mg_n / mg_elapsed, 1),
        "rows_per_s": round(mg_rows / mg_elapsed, 1),
        "rows_returned": mg_rows,
        "cabi": mg_cabi,
        "batched_multi_get_ops_per_s": round(n_batched / b_elapsed, 1),
        "batched_rows_per_s": round(b_rows / b_elapsed, 1),
        "batched_rows_returned": b_rows,
        "note": "single-stream per-call latency vs one-workgroup-per-request batching",
    }

    # ---- CPU baseline (oracle restatement, rank 0, N==1 only) ----
    # single-core pass + a partition-parallel pass on multiple host cores
    # (partitions are independent, exactly how the reference parallelizes)
    cpu_baseline = None
    cpu_aux = None
    if rank == 0 and world <= 1 and not args.skip_cpu_baseline:
        sample = min(args.cpu_sample, keys_pp)
        log(f"cpu baseline: oracle compaction pass over {sample} keys ...")
        nrec1, el1 = _cpu_compact_worker((sample, args.runs, 0, epoch_now))
        # single-core oracle get/scan rates on the same sample (BASELINE.md:
        # get ops/s + scan rows/s CPU side-by-side; config 1 shape)
        from incubator_pegasus_amd.capi import RrdbLib as _RL

        _orc = _RL(os.path.join(REPO, "oracle", "liboracle.so"))
        op = _orc.open(3, 0, -1)
        op.set_envs({"rocksdb.max_iteration_count": str(2**31 - 1)})
        for r in build_partition_data(sample, args.runs, 0, 0):
            op.ingest_run_arrays(np.ascontiguousarray(r["keys"]), r["koff"],
                                 np.ascontiguousarray(r["vals"]), r["voff"], r["sk"])
        t0 = time.time()
        res = op.scan_open(b"\x00\x00", b"\xff\xff", epoch_now, only_return_count=True,
                           full_scan=True, validate_partition_hash=False, batch_size=2**31 - 1)
        cpu_scan_rows = res.kv_count
        cpu_scan_el = time.time() - t0
        nq_cpu = 100_000
        from incubator_pegasus_amd import data as D3

        q_ids = D3.zipfian_ids(nq_cpu, sample, seed=D3.DEFAULT_SEED)
        qk = D3.make_raw_keys(q_ids).reshape(-1)
        qo = D3.fixed_offsets(nq_cpu, 18)
        import ctypes as _C

        from incubator_pegasus_amd.capi import _Result as _R

        t0 = time.time()
        r_ = _R()
        _orc._lib.rrdb_batch_get(op._h, nq_cpu, np.ascontiguousarray(qk).ctypes.data_as(_C.c_void_p),
                                 qo.ctypes.data_as(_C.c_void_p), epoch_now, _C.byref(r_))
        cpu_get_el = time.time() - t0
        _orc._lib.rrdb_free_result(_C.byref(r_))
        op.close()
        cpu_aux = {
            "scan_rows_per_s": round(cpu_scan_rows / cpu_scan_el, 1),
            "get_ops_per_s": round(nq_cpu / cpu_get_el, 1),
            "cores": 1,
            "sample_records": int(cpu_scan_rows),
        }
        workers = min(os.cpu_count() or 1, args.partitions)
        log(f"cpu baseline: {workers} parallel partitions x {sample} keys ...")
        import concurrent.futures as cf

        t0 = time.time()
        with cf.ProcessPoolExecutor(max_workers=workers) as ex:
            futs = [ex.submit(_cpu_compact_worker, (sample, args.runs, w, epoch_now))
                    for w in range(workers)]
            results = [f.result() for f in futs]
        # wall time includes per-worker data generation; use the sum of
        # compact-only times / workers as the effective parallel rate and the
        # max compact time as the conservative wall
        total_recs = sum(r[0] for r in results)
        max_el = max(r[1] for r in results)
        par_rate = total_recs / max_el
        cpu_baseline = {
            "value": round(par_rate, 1),
            "unit": "keys/s",
            "cores": workers,
            "kind": "port",
            "sample": (f"{workers} parallel partitions x {nrec1} records ({args.runs} runs), "
                       f"compaction-pass time max over workers; single-core: "
                       f"{nrec1 / el1:.0f} keys/s"),
        }

    for eng in parts:
        eng.close()

    if rank == 0:
        line = {
            "metric": "compaction keys/sec per node",
            "value": round(value, 1),
            "unit": "keys/s",
            "n_gpus": max(world, 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference number
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": ("compact-50M-16part-8run" if args.keys == 50_000_000 else
                             f"compact-{args.keys}-{args.partitions}part-{args.runs}run")
                            + ("-ttl-rules" if args.rules_profile else ""),
                "partitions_per_gpu": args.partitions,
                "keys_per_gpu": total_records,
                "runs_per_partition": args.runs,
                "record_bytes": 130,
                "hashkey_bytes": 16,
                "value_bytes": 100,
                "value_schema": "v1",
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
            "extra": {
                "scan_rows_per_s": round(scan_rows_per_s, 1),
                "scan_rows_total": rows,
                "scan_state_kernel_GBs": round(scan_achieved, 1),
                "get_ops_per_s": round(get_ops_per_s, 1),
                "get_kernel_ops_per_s": (round(nq / (get_kernel_ms * 1e-3), 1)
                                         if get_kernel_ms > 0 else None),
                "get_found": found,
                "compact_output_records_per_gpu": int(out_records),
                "compact_call_ms": compact_call_stats,
                "ycsb_e_prefix_multi_get": ycsb_e,
                "cpu_single_core": cpu_aux,
            },
        }
        print(json.dumps(line))


if __name__ == "__main__":
    main()
