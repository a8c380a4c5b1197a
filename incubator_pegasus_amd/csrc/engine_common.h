/* engine_common.h — shared host/device types for the MI355X rrdb engine.
 *
 * The engine holds each partition's sorted runs resident in HBM:
 *   keys blob + u64 offsets, values blob + u64 offsets, and
 *   seq_kind[i] = (seqno<<1)|kind per record (kind: 0=PUT, 1=DELETE).
 * Runs are ordered oldest -> newest (ingest order); every seqno in run r+1
 * is greater than every seqno in run r (enforced at ingest), which makes
 * "newest version wins" a run-index comparison for equal keys.
 *
 * Merge strategy (MI355X-native, no sequential merging iterator): fully
 * data-parallel rank merging with (key asc, run desc) ordering; a record is
 * shadowed iff its predecessor in rank order carries the same key.  The
 * round-2 production form is the group-streaming merge-path rank
 * (kernels.hip k_rank_grp): anchor keys cut the merged order into groups
 * whose per-run segments are disjoint, so each group's workgroup streams
 * its tail words through LDS exactly once and position-in-group IS local
 * rank.  Replaces rocksdb's merging-iterator heap.
 */
#pragma once
#include <stdint.h>

#define RRDB_MAX_RUNS 64
/* tail-word rank modes: max run count (the group rank's org word packs the
 * run in 16-GRP_ORG_SHIFT bits); group-streaming rank group budget */
#define LDST_MAXR 32
#define GRP_ORG_SHIFT 11
/* group-size sweep (r02, same box): target 512 -> 0.362ms rank, 1024 ->
 * 0.316ms (fewer barriers per element), 2048 -> 0.441ms (80KB LDS drops to
 * 2 workgroups/CU).  1024 is the sweet spot. */
#define GRP_CAP 2048
#define GRP_TARGET 1024

/* device-visible descriptor of one sorted run */
struct DevRun {
    const uint8_t *keys;
    const uint64_t *koff; /* n+1 */
    const uint8_t *vals;
    const uint64_t *voff; /* n+1 */
    const uint64_t *sk;   /* (seq<<1)|kind */
    uint64_t n;
    /* blocked bloom filter over full keys (§8(f)3; the reference's 10-bit
     * FullFilter equivalent, pegasus_server_impl_init.cpp:816-841): 64B
     * blocks, 6 probe bits; null = no filter (search every run). */
    const uint64_t *bloom;
    uint64_t bloom_blocks;
    /* all keys in the run share this length (0 = variable): searches then
     * compute key addresses directly instead of loading offset pairs */
    uint32_t fixed_klen;
    /* every key in a sorted run shares the lcp of its first and last key;
     * searches skip these leading bytes (multiple of 8, <=16) after one
     * per-call check that the query shares them too */
    uint32_t pfx_skip;
    /* exact (unfloored) shared-prefix length, capped at 32.  When the run is
     * fixed-stride and lcp_exact >= fixed_klen-8, the varying suffix fits one
     * u64 and probes compare a single big-endian word at key[klen-8..klen) */
    uint32_t lcp_exact;
    /* packed big-endian tail words (key[klen-8..klen) bswapped), built at run
     * creation when the single-word probe mode is eligible: 8B-strided probe
     * loads instead of klen-strided ones (null when ineligible) */
    const uint64_t *tails;
    /* hashkey-PREFIX blocked bloom (the reference's prefix-extractor bloom,
     * pegasus_server_impl_init.cpp:816-841 + hashkey_transform.h:40-54):
     * hashed over key[0 .. 2+hklen); prefix-scoped reads (multi_get, prefix
     * scans, sortkey_count) skip runs that cannot contain the hashkey. */
    const uint64_t *pfx_bloom;
    uint64_t pfx_bloom_blocks;
    /* per-record disposition column, built at run creation:
     * (expire_ts << 32) | kind.  Compaction-filter / scan-state evaluation
     * reads ONE coalescable 8B word instead of the sk word + voff pair + the
     * dependent 12B value-header parse (that gather chain dominated the
     * fused rank kernel).  Null only for transient descriptors that never
     * reach disposition (e.g. bloom builds). */
    const uint64_t *meta;
    /* all values in the run share this encoded length (0 = variable) */
    uint32_t fixed_vlen;
};

/* flattened user compaction rules/ops (device-resident)
 * mirrors compaction_filter_rule.{h,cpp} / compaction_operation.{h,cpp} */
enum { DFR_HASHKEY = 0, DFR_SORTKEY = 1, DFR_TTL_RANGE = 2 };
enum { DSM_ANYWHERE = 0, DSM_PREFIX = 1, DSM_POSTFIX = 2, DSM_INVALID = 3 };
enum { DOP_UPDATE_TTL = 0, DOP_DELETE = 1 };
enum { DUT_FROM_NOW = 0, DUT_FROM_CURRENT = 1, DUT_TIMESTAMP = 2, DUT_INVALID = 3 };

struct DevRule {
    int32_t type;       /* DFR_* */
    int32_t match_type; /* DSM_* (pattern rules) */
    uint32_t start_ttl, stop_ttl;
    uint32_t pat_off, pat_len; /* into pattern blob */
};

struct DevOp {
    int32_t type;    /* DOP_* */
    int32_t ut_type; /* DUT_* */
    uint32_t ut_value;
    int32_t rule_off, n_rules;
};

/* per-record scan/merge states (match the reference's range_iteration_state,
 * pegasus_server_impl.h) */
enum {
    ST_NORMAL = 0,
    ST_EXPIRED = 1,
    ST_FILTERED = 2,
    ST_HASH_INVALID = 3,
};

/* scan/filter parameter block passed to kernels by value */
struct ScanParams {
    uint32_t epoch_now;
    uint32_t data_version; /* value header: v0=4B, v1=12B, v2=13B */
    int32_t pidx;
    int32_t partition_version;
    uint8_t validate_hash;      /* engine-level env && request flag */
    int32_t hk_ft, sk_ft;       /* filter types (0..3) */
    const uint8_t *hk_pat;      /* device */
    uint64_t hk_pat_len;
    const uint8_t *sk_pat;
    uint64_t sk_pat_len;
    uint8_t no_value;
    uint64_t hash_key_skip;     /* multi_get: bytes of [len][hashkey] prefix to
                                   strip from emitted keys (0 for scan) */
};

struct CompactParams {
    uint32_t epoch_now;
    uint32_t data_version;
    uint32_t default_ttl;
    int32_t pidx;
    int32_t partition_version;
    uint8_t validate_hash;
    const DevOp *ops;     /* device */
    int32_t n_ops;
    const DevRule *rules; /* device */
    const uint8_t *pats;  /* device pattern blob */
};


/* fused small-multi_get argument block (one-launch YCSB-E path) */
#define MG_MAX_ROWS 4096
#define MG_SCRATCH_BYTES (1 << 20)
struct MgFusedArgs {
    const uint8_t *start; /* full rocksdb keys (device) */
    uint64_t start_len;
    const uint8_t *stop;
    uint64_t stop_len;
    uint8_t start_inclusive, stop_inclusive, reverse, no_value;
    uint32_t max_kv_count;
    uint32_t max_iteration_count;
    int64_t max_iteration_size;
    int32_t sk_ft;
    const uint8_t *sk_pat;
    uint64_t sk_pat_len;
    uint32_t epoch_now;
    uint32_t data_version;
    uint64_t hash_key_skip;
    /* outputs: [4] hdr = n_rows(-1 fallback), complete, kbytes, vbytes.
     * out_blob layout (one D2H): [koff (n+1)*8][voff (n+1)*8][keys][vals] */
    int64_t *out_hdr;
    uint8_t *out_blob; /* MG_BLOB_BYTES */
};
#define MG_BLOB_BYTES (2 * MG_SCRATCH_BYTES + 2 * (MG_MAX_ROWS + 1) * 8)

/* graph-captured serving lane: the per-call request is ONE pinned H2D of
 * [MgGraphHdr][start][stop][pattern]; the captured graph replays
 * H2D -> kernel -> D2H as a single launch (hipGraph), cutting the 3-4
 * submission round-trips of the small-op path */
#define MG_GRAPH_IN 4096
struct MgGraphHdr {
    uint32_t start_len, stop_len, sk_pat_len;
    uint8_t start_inclusive, stop_inclusive, reverse, no_value;
    uint32_t max_kv_count, max_iteration_count;
    int64_t max_iteration_size;
    int32_t sk_ft;
    uint32_t epoch_now, data_version;
    uint64_t hash_key_skip;
};

/* persistent serving kernel mailbox (pinned host memory).  Host writes the
 * request slice then release-stores req_seq; the resident 1-workgroup
 * kernel acquire-polls it, runs the fused multi_get and release-stores
 * done_seq after writing the response prefix.  The kernel ALWAYS exits
 * within MG_SRV_IDLE_MS of the last request (or on quit), so device-wide
 * synchronization can stall at most that long and can never hang. */
/* small: a resident kernel occupies one of the few HW queues, and
 * null-stream / device-wide syncs and co-scheduled streams can stall until
 * it exits — 2ms bounds that while call rates above ~1kHz keep it warm */
#define MG_SRV_IDLE_MS 2
struct MgMailbox {
    alignas(64) uint64_t req_seq;
    alignas(64) uint64_t done_seq;
    alignas(64) uint32_t quit;
    alignas(64) uint32_t alive; /* 1 while the kernel loop runs, 0 on exit */
    alignas(64) uint8_t req[MG_GRAPH_IN];
    alignas(64) uint8_t resp[32 + (16 << 10)];
};

/* compact per-record disposition written by the filter kernel */
struct CompactStatsDev {
    unsigned long long expired, filtered, tombstones, shadowed, output_records;
};
