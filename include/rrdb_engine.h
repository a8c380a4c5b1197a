/* rrdb_engine.h — C-ABI of the MI355X-native replacement for Pegasus's
 * per-replica storage engine (the read / scan / compaction hot path).
 *
 * This is the engine side of the reference's storage-plugin seam:
 *   - plugin registration seam: replication_app_base::register_storage_engine
 *     (reference src/server/pegasus_server_impl.h:115-120,
 *      src/replica/replication_app_base.h:129)
 *   - the 8 virtual read handlers it replaces:
 *     pegasus_read_service::on_get / on_multi_get / on_batch_get /
 *     on_sortkey_count / on_ttl / on_get_scanner / on_scan / on_clear_scanner
 *     (reference src/server/pegasus_read_service.h:54-68)
 *   - manual compaction executor: pegasus_server_impl::do_manual_compact
 *     (reference src/server/pegasus_server_impl.cpp:3373-3420)
 *   - env-driven knobs: update_app_envs (reference
 *     src/server/pegasus_server_impl.cpp:2728-3001, src/common/replica_envs.cpp)
 *
 * One handle per partition (gpid).  All input blobs are (ptr,len) and
 * caller-owned; all outputs live in a callee arena released with
 * rrdb_free_result().  Status codes are rocksdb::Status::Code ints
 * (rocksdb v8.5.3 include/rocksdb/status.h): kOk=0, kNotFound=1,
 * kCorruption=2, kInvalidArgument=4, kIncomplete=7.
 *
 * Every read entry point takes `epoch_now` (seconds since 2016-01-01 UTC,
 * reference src/base/pegasus_utils.h:40-41) explicitly instead of calling
 * time() internally, so results are deterministic under test; the host shim
 * that embeds this engine passes utils::epoch_now().
 */
#ifndef RRDB_ENGINE_H
#define RRDB_ENGINE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes (rocksdb::Status::Code, v8.5.3) ---- */
enum {
    RRDB_OK = 0,
    RRDB_NOT_FOUND = 1,
    RRDB_CORRUPTION = 2,
    RRDB_INVALID_ARGUMENT = 4,
    RRDB_IO_ERROR = 5,
    RRDB_INCOMPLETE = 7,
};

/* ---- filter types (idl/rrdb.thrift:27-33) ---- */
enum {
    RRDB_FT_NO_FILTER = 0,
    RRDB_FT_MATCH_ANYWHERE = 1,
    RRDB_FT_MATCH_PREFIX = 2,
    RRDB_FT_MATCH_POSTFIX = 3,
};

/* record kind in a sorted run (our run format; mirrors rocksdb internal-key
 * value types at the engine boundary: PUT passes the compaction filter,
 * DELETE is a tombstone that suppresses older versions) */
enum {
    RRDB_KIND_PUT = 0,
    RRDB_KIND_DELETE = 1,
};

/* scan context sentinel (reference src/server/pegasus_scan_context.h:33-41) */
#define RRDB_SCAN_CONTEXT_ID_COMPLETED (-1)

typedef struct {
    const uint8_t *data;
    uint64_t len;
} rrdb_cslice; /* caller-owned input blob */

typedef struct {
    uint8_t *data;
    uint64_t len;
} rrdb_slice; /* callee-arena output blob */

/* Result of any read call.  Which fields are meaningful depends on the call;
 * unused fields are zeroed.  Free with rrdb_free_result (idempotent). */
typedef struct {
    int32_t error;       /* rocksdb status code */
    uint64_t count;      /* number of kvs in keys/values */
    int64_t context_id;  /* scan_open/scan_next continuation handle */
    int64_t i64;         /* ttl_seconds / sortkey_count / kv_count (count-only scan) */
    rrdb_slice *keys;    /* [count]; scan: full rocksdb key; multi_get: sort key */
    rrdb_slice *values;  /* [count]; user data (value header stripped) */
    int32_t *expire_ts;  /* [count] when return_expire_ts was set, else NULL */
    /* batched multi_get only: per-request row counts and status codes */
    uint64_t *group_counts; /* [n_req] */
    int32_t *group_errors;  /* [n_req] */
    /* engine extension: when a scan asked for device-resident output the kv
     * bytes stay in HBM and these describe the packed device buffers
     * (keys/values above are then NULL). */
    uint8_t *dev_keys;       /* device ptr, packed key bytes */
    uint64_t *dev_key_offs;  /* device ptr, [count+1] offsets */
    uint8_t *dev_vals;       /* device ptr, packed value bytes */
    uint64_t *dev_val_offs;  /* device ptr, [count+1] offsets */
    void *_arena; /* internal */
} rrdb_result;

/* on_multi_get request — field-for-field the thrift multi_get_request
 * (idl/rrdb.thrift:185-200) */
typedef struct {
    rrdb_cslice hash_key;
    rrdb_cslice start_sortkey;
    rrdb_cslice stop_sortkey;
    uint8_t start_inclusive;
    uint8_t stop_inclusive;
    int32_t max_kv_count; /* <=0: no client cap */
    int32_t max_kv_size;  /* <=0: no client cap */
    uint8_t no_value;
    uint8_t reverse;
    int32_t sort_key_filter_type;
    rrdb_cslice sort_key_filter_pattern;
    /* when n_sort_keys > 0: the point-list variant (DB::MultiGet path,
     * reference pegasus_server_impl.cpp:779-860) */
    uint64_t n_sort_keys;
    const uint8_t *sort_keys;      /* packed */
    const uint64_t *sort_key_offs; /* [n_sort_keys+1] */
    /* engine extension (batched path): leave the packed result blobs in
     * HBM — rrdb_result.dev_vals = the packed blob (one region per request,
     * layout [koff (n+1)*8][voff (n+1)*8][keys][vals]), dev_val_offs = a
     * DEVICE array of per-request blob offsets [n_req+1], group_counts /
     * group_errors host-side as usual.  Best-effort: when any request fell
     * back to the general path the call returns the normal host-marshalled
     * slices and dev_vals stays null. */
    uint8_t on_device_out;
} rrdb_multi_get_request;

/* on_get_scanner request — thrift get_scanner_request (idl/rrdb.thrift:313-329) */
typedef struct {
    rrdb_cslice start_key; /* full rocksdb keys */
    rrdb_cslice stop_key;
    uint8_t start_inclusive;
    uint8_t stop_inclusive;
    int32_t batch_size; /* <=0: server default */
    uint8_t no_value;
    int32_t hash_key_filter_type;
    rrdb_cslice hash_key_filter_pattern;
    int32_t sort_key_filter_type;
    rrdb_cslice sort_key_filter_pattern;
    uint8_t full_scan;
    uint8_t validate_partition_hash; /* __isset default: 1 */
    uint8_t return_expire_ts;
    uint8_t only_return_count;
    uint8_t on_device_out; /* engine extension: leave kv bytes in HBM */
} rrdb_scan_request;

/* manual compaction options — the subset of rocksdb::CompactRangeOptions that
 * do_manual_compact builds (reference pegasus_server_impl.cpp:3373-3420,
 * pegasus_manual_compact_service.cpp:231-271).  Our engine always merges all
 * runs of the partition into one (CompactRange over the full range with
 * bottommost_level_compaction=force ≡ tombstones dropped). */
typedef struct {
    int32_t target_level;          /* -1 = bottommost (informational) */
    uint8_t bottommost_force;      /* 1 = force (default in reference) */
    uint8_t keep_inputs;           /* engine extension for benchmarking: run the
                                    * full merge+filter+write but leave the input
                                    * runs in place (a repeatable pass); 0 =
                                    * reference semantics (inputs replaced) */
} rrdb_compact_options;

typedef struct {
    uint64_t input_records;  /* records read across all input runs */
    uint64_t output_records; /* records in the merged run */
    uint64_t expired;        /* dropped by TTL */
    uint64_t filtered;       /* dropped by user rules / stale-split hash */
    uint64_t tombstones;     /* DELETE records dropped (bottommost) */
    uint64_t shadowed;       /* older versions superseded by newer seqno */
    uint64_t output_bytes;   /* key+value bytes written */
} rrdb_compact_stats;

/* ---- lifecycle ---- */

/* Open the engine for partition (app_id, pidx) on GPU gpu_id.
 * gpu_id < 0: host-only handle (oracle library); the GPU engine requires
 * gpu_id >= 0 and fails loudly if no HIP device is present. */
void *rrdb_open(int32_t app_id, int32_t pidx, int32_t gpu_id);
void rrdb_close(void *h);

/* Table-env updates — mirrors update_app_envs / the compaction-filter factory
 * setters (reference key_ttl_compaction_filter.h:160-190, replica_envs.cpp).
 * Recognised keys:
 *   "default_ttl"                             seconds, 0 disables
 *   "user_specified_compaction"               JSON ops (compaction_operation.cpp:160-190)
 *   "replica.split.validate_partition_hash"   "true"/"false"
 *   "manual_compact.disabled"                 "true"/"false"
 *   "pegasus.data_version"                    0|1|2 (default 1)
 *   "replica.rocksdb_iteration_threshold_time_ms"   (init default 30000)
 *   "rocksdb.max_iteration_count"                   (init default 1000)
 *   "rocksdb.multi_get_max_iteration_count"         (init default 3000)
 *   "rocksdb.multi_get_max_iteration_size"          (init default 30<<20)
 * Unknown keys are ignored (reference behavior). */
int32_t rrdb_set_envs(void *h, const char *const *keys, const char *const *values, int32_t n);

/* Mirrors KeyWithTTLCompactionFilterFactory::SetPartitionIndex/-Version and
 * pegasus_server_impl::set_partition_version. */
int32_t rrdb_set_partition_version(void *h, int32_t partition_version);

/* Ingest one sorted run (memtable flush / external SST equivalent).
 * keys/values packed, offs arrays have n+1 entries; seq_kind[i] =
 * (seqno << 1) | kind.  Constraints checked (kInvalidArgument on violation):
 * keys strictly increasing bytewise within the run; every seqno in this run
 * must be greater than every seqno already ingested (L0 flush order). */
int32_t rrdb_ingest_run(void *h, const uint8_t *keys, const uint64_t *key_offs,
                        const uint8_t *values, const uint64_t *val_offs,
                        const uint64_t *seq_kind, uint64_t n_records);

/* ---- write path ----
 * Host-side memtable feeding sorted runs, mirroring the committed-mutation
 * apply path: pegasus_server_impl::on_batched_write_requests ->
 * pegasus_write_service put/remove (reference pegasus_write_service.h:119-207)
 * -> rocksdb_wrapper::write_batch_put/_delete (rocksdb_wrapper.cpp:121-247):
 * key = pegasus_generate_key(hash_key, sort_key); value encoded with
 * expire_ts (0 = no TTL); remove writes a tombstone.  Within the memtable the
 * newest write to a key wins (memtable upsert).  rrdb_flush turns the
 * memtable into a sorted run (the memtable-flush -> L0 SST step); reads
 * flush lazily so committed writes are immediately visible, as they are
 * through rocksdb's memtable read path.
 * epoch_now is the write's wall clock: when expire_ts == 0 and the table
 * carries a default_ttl env, the stored expire becomes
 * epoch_now + default_ttl AT WRITE TIME (rocksdb_wrapper::db_expire_ts,
 * rocksdb_wrapper.cpp:280-286) — not deferred to the compaction filter. */
int32_t rrdb_put(void *h, const uint8_t *hash_key, uint64_t hklen, const uint8_t *sort_key,
                 uint64_t sklen, const uint8_t *value, uint64_t vlen, uint32_t expire_ts,
                 uint32_t epoch_now);
int32_t rrdb_remove(void *h, const uint8_t *hash_key, uint64_t hklen, const uint8_t *sort_key,
                    uint64_t sklen);
uint64_t rrdb_memtable_entries(void *h);
int32_t rrdb_flush(void *h); /* no-op when the memtable is empty */

/* ---- read service (pegasus_read_service.h:54-68 semantics) ---- */

/* on_get (pegasus_server_impl.cpp:418-494): newest version of `key`;
 * DELETE or TTL-expired -> kNotFound; value = user data (header stripped). */
int32_t rrdb_get(void *h, const uint8_t *key, uint64_t key_len, uint32_t epoch_now,
                 rrdb_result *out);

/* on_batch_get (pegasus_server_impl.cpp:906-1016): multi-point lookup,
 * NotFound/expired keys silently skipped; out->keys = full rocksdb keys. */
int32_t rrdb_batch_get(void *h, uint64_t n_keys, const uint8_t *keys,
                       const uint64_t *key_offs, uint32_t epoch_now, rrdb_result *out);

/* on_ttl (pegasus_server_impl.cpp:1092-1150): out->i64 = ttl_seconds
 * (expire-now, or -1 when no ttl). */
int32_t rrdb_ttl(void *h, const uint8_t *key, uint64_t key_len, uint32_t epoch_now,
                 rrdb_result *out);

/* on_sortkey_count (pegasus_server_impl.cpp:1018-1090): out->i64 = count of
 * live sortkeys under hash_key (-1 if the time budget was exceeded). */
int32_t rrdb_sortkey_count(void *h, const uint8_t *hash_key, uint64_t hash_key_len,
                           uint32_t epoch_now, rrdb_result *out);

/* on_multi_get (pegasus_server_impl.cpp:496-904). */
int32_t rrdb_multi_get(void *h, const rrdb_multi_get_request *req, uint32_t epoch_now,
                       rrdb_result *out);

/* Batched multi_get (engine extension): N independent on_multi_get range
 * requests sharing every field of `shared` except hash_key, answered in one
 * launch (one workgroup per request) — models the reference's concurrent
 * THREAD_POOL_SCAN handlers without N round trips.  Results
 * concatenate in request order; out->group_counts / group_errors give each
 * request's row count and status.  shared->n_sort_keys must be 0. */
int32_t rrdb_multi_get_batch(void *h, uint64_t n_req, const uint8_t *hash_keys,
                             const uint64_t *hk_offs, const rrdb_multi_get_request *shared,
                             uint32_t epoch_now, rrdb_result *out);

/* on_get_scanner (pegasus_server_impl.cpp:1151-1397): first batch + parked
 * continuation in out->context_id (RRDB_SCAN_CONTEXT_ID_COMPLETED when done). */
int32_t rrdb_scan_open(void *h, const rrdb_scan_request *req, uint32_t epoch_now,
                       rrdb_result *out);

/* on_scan (pegasus_server_impl.cpp:1399-1547): next batch for a parked
 * context; kNotFound if the context id is unknown/expired. */
int32_t rrdb_scan_next(void *h, int64_t context_id, uint32_t epoch_now, rrdb_result *out);

/* on_clear_scanner (pegasus_server_impl.cpp:1549). */
void rrdb_clear_scanner(void *h, int64_t context_id);

/* ---- compaction (do_manual_compact, pegasus_server_impl.cpp:3373-3420) ----
 * k-way merge of all runs -> one run; newest seqno wins per key; tombstones
 * dropped (bottommost); KeyWithTTLCompactionFilter::Filter applied per
 * surviving PUT (key_ttl_compaction_filter.h:55-121): TTL expiry, default-TTL
 * rewrite, stale-split-hash drop, user delete/update-TTL rules. */
int32_t rrdb_manual_compact(void *h, const rrdb_compact_options *opts, uint32_t epoch_now,
                            rrdb_compact_stats *stats);

/* Split compaction: the pipelined-partitions seam.  The reference runs
 * per-replica compactions concurrently on THREAD_POOL_COMPACT
 * (pegasus_manual_compact_service, pegasus_server_impl.cpp:3373-3420);
 * here a single host thread gets the same overlap by calling _begin on
 * every partition handle (submits the merge + sizing work, returns
 * without blocking) and then _finish on each (waits, emits the output
 * run, returns stats).  _begin fails with kInvalidArgument if a split
 * compaction is already pending on the handle; _finish without a pending
 * _begin likewise.  rrdb_manual_compact == begin+finish. */
/* pipelined count scan (the shell count_data fan-out,
 * src/shell/commands/data_operations.cpp:2305): begin submits the fused
 * count kernels without blocking so every partition's scan co-runs; finish
 * returns the count in out->i64.  Only the full-count shape is supported
 * (only_return_count, forward, start-inclusive, count/iteration caps >= the
 * table's record count) — other shapes return kInvalidArgument and the
 * caller uses rrdb_scan_open. */
int32_t rrdb_scan_count_begin(void *h, const rrdb_scan_request *q, uint32_t epoch_now);
int32_t rrdb_scan_count_finish(void *h, rrdb_result *out);

int32_t rrdb_manual_compact_begin(void *h, const rrdb_compact_options *opts,
                                  uint32_t epoch_now);
int32_t rrdb_manual_compact_finish(void *h, rrdb_compact_stats *stats);

/* ---- checkpoint ----
 * Serializes the partition's runs to <dir>/checkpoint.<decree>/ in the
 * engine's own run format (decree-tagged like the reference's
 * `checkpoint.{decree}` snapshots, pegasus_server_impl.cpp:1951-2137;
 * the on-disk format is ours — SST-byte parity is not owed).
 * The memtable is flushed first.  rrdb_restore loads a checkpoint into an
 * EMPTY handle (learner/backup ingest path); checkpoints written by either
 * backend restore into either backend. */
int32_t rrdb_checkpoint(void *h, const char *dir, uint64_t decree);
int32_t rrdb_restore(void *h, const char *dir, uint64_t decree);

/* ---- introspection ---- */
uint64_t rrdb_num_runs(void *h);
uint64_t rrdb_num_records(void *h); /* total across runs, versions included */

/* Release a result's arena.  Safe to call twice. */
void rrdb_free_result(rrdb_result *r);

/* Library identity: "oracle-cpu" or "hip-gfx950". */
const char *rrdb_backend(void);

/* Last recorded GPU time (ms) of a named internal phase, measured with HIP
 * events on the engine's own stream (bench/roofline introspection; the CPU
 * oracle returns -1).  Phases: "compact_rank", "compact_flags",
 * "compact_emit", "compact_total", "scan_state", "scan_emit", "get_search",
 * "view_rank".  Returns -1 if never recorded. */
double rrdb_phase_ms(void *h, const char *phase);

#ifdef __cplusplus
}
#endif
#endif /* RRDB_ENGINE_H */
