/* kernels.hip — gfx950 (CDNA4) device kernels for the rrdb engine.
 *
 * Replaces, from scratch (no ported code): rocksdb's point-Get / MultiGet
 * block path, the merging iterator (k-way heap) and the compaction
 * iterator + KeyWithTTLCompactionFilter evaluation that the reference's
 * pegasus_server_impl drives (call sites: pegasus_server_impl.cpp:441,618,
 * 804,948,1243,3389).
 *
 * Design notes (MI355X, round-2 state):
 *  - wave64; 256-thread workgroups; capped grids with grid-stride loops
 *  - the production merge is the GROUP-STREAMING rank (k_rank_grp): anchor
 *    keys partition every run into disjoint per-group segments; one
 *    workgroup streams its group's packed tail words through LDS once and
 *    merges them with pairwise merge-path rounds — global traffic is one
 *    8B tail read per record and every output write lands in a contiguous
 *    rank range.  MODE 0 = compaction (fused filter via the per-run
 *    (expire<<32)|kind meta column), MODE 1 = scan view build, MODE 2 =
 *    fused count scan.  The r01 probe-based kernels (k_rank_compact[_ldst],
 *    bound tables) remain as fallbacks for ineligible run sets and A/B.
 *  - disposition tallies accumulate in per-thread registers and flush once
 *    per block into 8 stat banks (shared-counter atomics serialized at
 *    ~88 adds/us and floored every fused kernel before)
 *  - emits: direct-indexed 16B-chunk copies for 16B-aligned fixed strides,
 *    row-per-thread u64 copies for other fixed strides, chunk-anchor
 *    binary-search copies for variable layouts
 *  - serving: cooperative p-ary bounds + hashkey-prefix blooms in the fused
 *    multi_get; a hipGraph-captured lane and a persistent mailbox-polled
 *    1-workgroup server (bounded 2ms idle exit) remove the per-call
 *    dispatch cost
 *  - crc64 table lives in __device__ memory (L1/L2-resident after first use)
 *  - integer/byte work only: HBM-bound, no MFMA by design
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>

#include "engine_common.h"

#define WAVE 64
#define BLOCK 256
/* LDS tail-word staged rank: staged window capacity (u64 words) and the
 * max run count the staging handles (beyond either -> in-kernel fallback) */
#define LDST_CAP 3072
#define MAX_GRID 2048
#define PSUM_ITEMS_PER_THREAD 16
#define PSUM_BLOCK_ITEMS (BLOCK * PSUM_ITEMS_PER_THREAD) /* 4096 */

static inline int grid_for(uint64_t n, int per_block)
{
    uint64_t b = (n + per_block - 1) / per_block;
    if (b == 0)
        b = 1;
    if (b > MAX_GRID)
        b = MAX_GRID;
    return (int)b;
}

#define HIP_CHECK(x)                                                                               \
    do {                                                                                           \
        hipError_t err_ = (x);                                                                     \
        if (err_ != hipSuccess) {                                                                  \
            fprintf(stderr, "rrdb-hip fatal: %s at %s:%d\n", hipGetErrorString(err_), __FILE__,    \
                    __LINE__);                                                                     \
            abort();                                                                               \
        }                                                                                          \
    } while (0)

/* ================= crc64 (device) =================
 * Same polynomial/table construction as the reference crc.cpp:236-295 —
 * restated; table built on host (launchers.cpp) and copied here. */
__device__ uint64_t d_crc64_table[256];

__device__ static inline uint64_t dev_crc64(const uint8_t *p, uint64_t n)
{
    uint64_t crc = ~0ull;
    for (uint64_t i = 0; i < n; i++)
        crc = d_crc64_table[(uint8_t)(crc ^ p[i])] ^ (crc >> 8);
    return ~crc;
}

/* pegasus_key_hash (pegasus_key_schema.h:148-165) */
__device__ static inline uint64_t dev_key_hash(const uint8_t *key, uint64_t len)
{
    uint32_t hklen = ((uint32_t)key[0] << 8) | key[1];
    if (hklen > 0)
        return dev_crc64(key + 2, hklen);
    return dev_crc64(key + 2, len - 2);
}

/* ================= byte compare ================= */
__device__ static inline int dev_key_cmp(const uint8_t *a, uint64_t alen, const uint8_t *b,
                                         uint64_t blen)
{
    uint64_t m = alen < blen ? alen : blen;
    uint64_t i = 0;
    for (; i + 8 <= m; i += 8) {
        uint64_t wa, wb;
        __builtin_memcpy(&wa, a + i, 8);
        __builtin_memcpy(&wb, b + i, 8);
        if (wa != wb) {
            wa = __builtin_bswap64(wa);
            wb = __builtin_bswap64(wb);
            return wa < wb ? -1 : 1;
        }
    }
    for (; i < m; i++)
        if (a[i] != b[i])
            return a[i] < b[i] ? -1 : 1;
    return alen < blen ? -1 : (alen > blen ? 1 : 0);
}

/* per-wave contiguous work chunks.  Measured on gfx950 (r01 PMC): chunked
 * mappings RAISED fetch bytes for the search kernels (168 vs 125 MB/launch
 * rank_compact) and slowed the rank-major emit; strided grid mappings are
 * the default and this helper is used only by the input-major emit A/B. */
__device__ static inline void wave_chunk(uint64_t total, uint64_t *wstart, uint64_t *wend,
                                         int *lane_out)
{
    uint64_t nw = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    uint64_t wid = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t chunk = (total + nw - 1) / nw;
    uint64_t s = wid * chunk;
    uint64_t e = s + chunk;
    if (s > total)
        s = total;
    if (e > total)
        e = total;
    *wstart = s;
    *wend = e;
    *lane_out = threadIdx.x % WAVE;
}

/* lane-strided record copy: 4B unaligned chunks (consecutive lanes touch
 * consecutive addresses -> coalesced), byte tail */
__device__ static inline void wave_copy(uint8_t *dst, const uint8_t *src, uint64_t n, int lane)
{
    uint64_t n4 = n >> 2;
    for (uint64_t c = lane; c < n4; c += WAVE) {
        uint32_t w;
        __builtin_memcpy(&w, src + 4 * c, 4);
        __builtin_memcpy(dst + 4 * c, &w, 4);
    }
    for (uint64_t b = (n4 << 2) + lane; b < n; b += WAVE)
        dst[b] = src[b];
}

__device__ static inline const uint8_t *run_key(const DevRun &r, uint64_t i, uint64_t *len)
{
    uint64_t o = r.koff[i];
    *len = r.koff[i + 1] - o;
    return r.keys + o;
}
__device__ static inline const uint8_t *run_val(const DevRun &r, uint64_t i, uint64_t *len)
{
    uint64_t o = r.voff[i];
    *len = r.voff[i + 1] - o;
    return r.vals + o;
}

/* once per search: skip the run's shared key prefix iff the query shares it
 * too (sorted order guarantees every run key carries it and is >= that long) */
__device__ static inline uint64_t pfx_engage(const DevRun &r, const uint8_t *key, uint64_t klen)
{
    uint64_t ps = r.pfx_skip;
    if (!ps || klen < ps || dev_key_cmp(key, ps, r.keys, ps) != 0)
        return 0;
    return ps;
}
/* single-word probe mode: fixed stride, the varying suffix fits u64, and the
 * query is same-length sharing the run's constant first klen-8 bytes.  The
 * whole bytewise compare then reduces to one big-endian word compare. */
__device__ static inline bool word_probe_engage(const DevRun &r, const uint8_t *key,
                                                uint64_t klen, uint64_t *qw)
{
    uint64_t fk = r.fixed_klen;
    if (!r.tails || klen != fk)
        return false;
    if (fk > 8 && dev_key_cmp(key, fk - 8, r.keys, fk - 8) != 0)
        return false;
    uint64_t w;
    __builtin_memcpy(&w, key + fk - 8, 8);
    *qw = __builtin_bswap64(w);
    return true;
}
__device__ static inline uint64_t run_tail_word(const DevRun &r, uint64_t fk, uint64_t mid)
{
    (void)fk;
    return r.tails[mid];
}

__global__ void k_build_tails(const uint8_t *keys, uint64_t fk, uint64_t n, uint64_t *tails)
{
    for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t w;
        __builtin_memcpy(&w, keys + i * fk + fk - 8, 8);
        tails[i] = __builtin_bswap64(w);
    }
}
void launch_build_tails(const uint8_t *keys, uint64_t fk, uint64_t n, uint64_t *tails,
                        hipStream_t s)
{
    uint64_t blocks = (n + 255) / 256;
    if (blocks > 2048)
        blocks = 2048;
    hipLaunchKernelGGL(k_build_tails, dim3((uint32_t)blocks), dim3(256), 0, s, keys, fk, n,
                       tails);
}
/* first index in [lo,hi) with key >= target */
__device__ static uint64_t dev_lower_bound(const DevRun &r, const uint8_t *key, uint64_t klen,
                                           uint64_t lo, uint64_t hi)
{
    uint64_t fk = r.fixed_klen, qw;
    if (word_probe_engage(r, key, klen, &qw)) {
        while (lo < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (run_tail_word(r, fk, mid) < qw)
                lo = mid + 1;
            else
                hi = mid;
        }
        return lo;
    }
    uint64_t ps = pfx_engage(r, key, klen);
    const uint8_t *q = key + ps;
    uint64_t ql = klen - ps;
    if (fk) { /* fixed stride: no offset-pair loads per probe */
        uint64_t fl = fk - ps;
        while (lo < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (dev_key_cmp(r.keys + mid * fk + ps, fl, q, ql) < 0)
                lo = mid + 1;
            else
                hi = mid;
        }
        return lo;
    }
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1, ml;
        const uint8_t *mk = run_key(r, mid, &ml);
        if (dev_key_cmp(mk + ps, ml - ps, q, ql) < 0)
            lo = mid + 1;
        else
            hi = mid;
    }
    return lo;
}
/* first index in [lo,hi) with key > target */
__device__ static uint64_t dev_upper_bound(const DevRun &r, const uint8_t *key, uint64_t klen,
                                           uint64_t lo, uint64_t hi)
{
    uint64_t fk = r.fixed_klen, qw;
    if (word_probe_engage(r, key, klen, &qw)) {
        while (lo < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (run_tail_word(r, fk, mid) <= qw)
                lo = mid + 1;
            else
                hi = mid;
        }
        return lo;
    }
    uint64_t ps = pfx_engage(r, key, klen);
    const uint8_t *q = key + ps;
    uint64_t ql = klen - ps;
    if (fk) {
        uint64_t fl = fk - ps;
        while (lo < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (dev_key_cmp(r.keys + mid * fk + ps, fl, q, ql) <= 0)
                lo = mid + 1;
            else
                hi = mid;
        }
        return lo;
    }
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1, ml;
        const uint8_t *mk = run_key(r, mid, &ml);
        if (dev_key_cmp(mk + ps, ml - ps, q, ql) <= 0)
            lo = mid + 1;
        else
            hi = mid;
    }
    return lo;
}

/* ================= value codec (device) =================
 * pegasus_value_schema.h:58-125 / value_schema_v2.cpp:101-125 */
__device__ static inline uint32_t dev_hdr_len(uint32_t ver)
{
    return ver == 0 ? 4u : (ver == 1 ? 12u : 13u);
}
__device__ static inline uint32_t dev_expire_ts(uint32_t ver, const uint8_t *v)
{
    uint32_t off = (ver == 2) ? 1 : 0;
    return ((uint32_t)v[off] << 24) | ((uint32_t)v[off + 1] << 16) | ((uint32_t)v[off + 2] << 8) |
           (uint32_t)v[off + 3];
}
__device__ static inline int dev_ts_expired(uint32_t now, uint32_t ts)
{
    return ts > 0 && ts <= now;
}

/* per-record disposition column: (expire_ts << 32) | kind.  Built once at
 * run creation so filter/scan-state evaluation reads one 8B word instead of
 * the sk word + voff pair + dependent value-header parse. */
__global__ void k_build_meta(const uint8_t *vals, const uint64_t *voff, uint64_t fixed_vlen,
                             const uint64_t *sk, uint64_t n, uint32_t dv, uint64_t *meta)
{
    for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t kind = sk[i] & 1;
        uint64_t off = fixed_vlen ? i * fixed_vlen : voff[i];
        uint64_t vl = fixed_vlen ? fixed_vlen : (voff[i + 1] - voff[i]);
        uint32_t ts = (!kind && vl >= dev_hdr_len(dv)) ? dev_expire_ts(dv, vals + off) : 0;
        meta[i] = ((uint64_t)ts << 32) | kind;
    }
}
void launch_build_meta(const uint8_t *vals, const uint64_t *voff, uint64_t fixed_vlen,
                       const uint64_t *sk, uint64_t n, uint32_t dv, uint64_t *meta,
                       hipStream_t s)
{
    k_build_meta<<<grid_for(n, BLOCK), BLOCK, 0, s>>>(vals, voff, fixed_vlen, sk, n, dv, meta);
}


/* ================= pattern / rules (device) =================
 * compaction_filter_rule.cpp:31-90, compaction_operation.cpp:33-113 */
__device__ static int dev_mem_eq(const uint8_t *a, const uint8_t *b, uint64_t n)
{
    for (uint64_t i = 0; i < n; i++)
        if (a[i] != b[i])
            return 0;
    return 1;
}

__device__ static int dev_pattern_match(const uint8_t *v, uint64_t vlen, int type,
                                        const uint8_t *pat, uint64_t plen)
{
    if (plen == 0 || vlen < plen)
        return 0;
    switch (type) {
    case DSM_ANYWHERE:
        for (uint64_t i = 0; i + plen <= vlen; i++)
            if (dev_mem_eq(v + i, pat, plen))
                return 1;
        return 0;
    case DSM_PREFIX:
        return dev_mem_eq(v, pat, plen);
    case DSM_POSTFIX:
        return dev_mem_eq(v + vlen - plen, pat, plen);
    default:
        return 0;
    }
}

/* validate_filter (pegasus_server_impl.cpp:2350-2380): NO_FILTER / empty
 * pattern -> pass */
__device__ static int dev_validate_filter(int ft, const uint8_t *pat, uint64_t plen,
                                          const uint8_t *v, uint64_t vlen)
{
    if (ft == 0 || plen == 0)
        return 1;
    if (vlen < plen)
        return 0;
    if (ft == 1) { /* anywhere */
        for (uint64_t i = 0; i + plen <= vlen; i++)
            if (dev_mem_eq(v + i, pat, plen))
                return 1;
        return 0;
    }
    if (ft == 2)
        return dev_mem_eq(v, pat, plen);
    return dev_mem_eq(v + vlen - plen, pat, plen);
}

__device__ __host__ static inline uint64_t bloom_hash(const uint8_t *k, uint64_t n);
__device__ static inline int bloom_maybe_has(const DevRun &r, uint64_t h);
__device__ static inline int pfx_bloom_maybe_has(const DevRun &r, uint64_t h);

/* ================= bounds ================= */
__global__ void k_bounds(const DevRun *runs, int R, const uint8_t *key, uint64_t klen,
                         uint64_t *out /* [R] */, int upper)
{
    int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= R)
        return;
    if (key == nullptr) {
        out[r] = upper ? runs[r].n : 0;
        return;
    }
    out[r] = upper ? dev_upper_bound(runs[r], key, klen, 0, runs[r].n)
                   : dev_lower_bound(runs[r], key, klen, 0, runs[r].n);
}

/* ---- bound table: positions of every (1<<BT_SHIFT)-th record of each run
 * in every other run.  Narrows each record's cross-run binary searches from
 * the full run to a ~block-sized, L2-hot window (correctness never depends
 * on the narrowing: boundaries come from the same comparator). ---- */
#define BT_SHIFT_DEFAULT 8 /* 256 records per block (env engine.bt_shift) */

/* layout: for run r, rows j = 0..P_r (P_r = ceil(w_r / 256)); row j holds R
 * u64s = search bounds of run-r record (lo_r + j*256) in every run; the last
 * row is hi[].  bt_off[r] = row offset of run r's table. */
__global__ void k_bound_table(const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi,
                              const uint64_t *bt_off, uint64_t total_rows, int bt_shift,
                              uint64_t *bt, const uint64_t *cbt_off /* coarse table (optional) */,
                              const uint64_t *cbt, int cbt_shift)
{
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < total_rows;
         t += gridDim.x * (uint64_t)blockDim.x) {
        /* find the run this row belongs to */
        int r = 0;
        while (bt_off[r + 1] <= t)
            r++;
        uint64_t j = t - bt_off[r];
        uint64_t nrows = bt_off[r + 1] - bt_off[r]; /* P_r + 1 */
        uint64_t *row = bt + t * (uint64_t)R;
        uint64_t i = lo[r] + (j << bt_shift);
        if (j == nrows - 1 || i >= hi[r]) { /* sentinel row: hi */
            for (int q = 0; q < R; q++)
                row[q] = hi[q];
            continue;
        }
        uint64_t kl;
        const uint8_t *k = run_key(runs[r], i, &kl);
        const uint64_t *c0 = nullptr, *c1 = nullptr;
        if (cbt) {
            uint64_t j2 = (i - lo[r]) >> cbt_shift;
            c0 = cbt + (cbt_off[r] + j2) * (uint64_t)R;
            c1 = cbt + (cbt_off[r] + j2 + 1) * (uint64_t)R;
        }
        for (int q = 0; q < R; q++) {
            if (q == r) {
                row[q] = i;
                continue;
            }
            uint64_t qlo = c0 ? c0[q] : lo[q];
            uint64_t qhi = c1 ? c1[q] : hi[q];
            row[q] = (q > r) ? dev_upper_bound(runs[q], k, kl, qlo, qhi)
                             : dev_lower_bound(runs[q], k, kl, qlo, qhi);
        }
    }
}

/* ================= rank merge =================
 * For windowed record t -> (r,i):
 * rank = (i - lo[r]) + sum_{r'>r} upper_bound(r',key) + sum_{r'<r} lower_bound(r',key)
 * (run index order == age order: higher r == newer == first among equal keys)
 * Scatter order[rank] = (r<<40)|i.  wprefix[r] = running sum of window sizes. */
__global__ void k_rank(const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi,
                       const uint64_t *wprefix /* [R+1] */, uint64_t total, uint64_t *order,
                       uint8_t *shadowed /* [total], by rank position */,
                       const uint64_t *bt_off, const uint64_t *bt, int bt_shift)
{
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < total;
         t += gridDim.x * (uint64_t)blockDim.x) {
        int r = 0;
        while (wprefix[r + 1] <= t)
            r++;
        uint64_t i = lo[r] + (t - wprefix[r]);
        uint64_t kl;
        const uint8_t *k = run_key(runs[r], i, &kl);
        uint64_t rank = i - lo[r];
        int shadow = 0;
        const uint64_t *b0 = nullptr, *b1 = nullptr;
        if (bt) {
            uint64_t j = (i - lo[r]) >> bt_shift;
            b0 = bt + (bt_off[r] + j) * (uint64_t)R;
            b1 = bt + (bt_off[r] + j + 1) * (uint64_t)R;
        }
        for (int q = 0; q < R; q++) {
            if (q == r)
                continue;
            uint64_t qlo = b0 ? b0[q] : lo[q];
            uint64_t qhi = b1 ? b1[q] : hi[q];
            if (q > r) {
                uint64_t ub = dev_upper_bound(runs[q], k, kl, qlo, qhi);
                if (!shadow && ub > lo[q]) {
                    uint64_t pl;
                    const uint8_t *pk = run_key(runs[q], ub - 1, &pl);
                    if (dev_key_cmp(pk, pl, k, kl) == 0)
                        shadow = 1;
                }
                rank += ub - lo[q];
            } else {
                rank += dev_lower_bound(runs[q], k, kl, qlo, qhi) - lo[q];
            }
        }
        order[rank] = ((uint64_t)r << 40) | i;
        shadowed[rank] = (uint8_t)shadow;
    }
}

/* LDS tail-word staged variant of k_rank (view builds for scans) — same
 * staging and eligibility as k_rank_compact_ldst */
__global__ void __launch_bounds__(BLOCK) k_rank_ldst(
    const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi, const uint64_t *wprefix,
    uint64_t total, uint64_t *order, uint8_t *shadowed, const uint64_t *bt_off,
    const uint64_t *bt, int bt_shift)
{
    __shared__ uint64_t s_tails[LDST_CAP];
    __shared__ uint64_t s_winlo[LDST_MAXR], s_winhi[LDST_MAXR], s_base[LDST_MAXR];
    __shared__ int s_fallback;

    uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
    int lane = threadIdx.x % WAVE;
    if (threadIdx.x == 0)
        s_fallback = (bt == nullptr || R > LDST_MAXR) ? 1 : 0;
    if (threadIdx.x < (unsigned)(R < LDST_MAXR ? R : LDST_MAXR)) {
        s_winlo[threadIdx.x] = ~0ull;
        s_winhi[threadIdx.x] = 0;
    }
    __syncthreads();
    int r = 0;
    uint64_t i = 0, qw = 0;
    const uint64_t *b0 = nullptr, *b1 = nullptr;
    if (t < total) {
        while (wprefix[r + 1] <= t)
            r++;
        i = lo[r] + (t - wprefix[r]);
        qw = runs[r].tails[i];
        if (bt) {
            uint64_t j = (i - lo[r]) >> bt_shift;
            b0 = bt + (bt_off[r] + j) * (uint64_t)R;
            b1 = bt + (bt_off[r] + j + 1) * (uint64_t)R;
        }
    }
    if (!s_fallback) {
        for (int q = 0; q < R; q++) {
            uint64_t wl = (t < total && q != r && b0) ? b0[q] : ~0ull;
            uint64_t wh = (t < total && q != r && b1) ? b1[q] : 0;
            for (int d = WAVE / 2; d; d >>= 1) {
                uint64_t o1 = __shfl_xor(wl, d);
                if (o1 < wl)
                    wl = o1;
                uint64_t o2 = __shfl_xor(wh, d);
                if (o2 > wh)
                    wh = o2;
            }
            if (lane == 0 && wh > 0) {
                atomicMin((unsigned long long *)&s_winlo[q], (unsigned long long)wl);
                atomicMax((unsigned long long *)&s_winhi[q], (unsigned long long)wh);
            }
        }
    }
    __syncthreads();
    if (threadIdx.x == 0 && !s_fallback) {
        uint64_t tot = 0;
        for (int q = 0; q < R; q++) {
            s_base[q] = tot;
            if (s_winhi[q] > s_winlo[q])
                tot += s_winhi[q] - s_winlo[q];
        }
        if (tot > LDST_CAP)
            s_fallback = 1;
    }
    __syncthreads();
    if (!s_fallback) {
        for (int q = 0; q < R; q++) {
            uint64_t wl = s_winlo[q], wh = s_winhi[q];
            if (wh <= wl)
                continue;
            const uint64_t *tq = runs[q].tails;
            uint64_t base = s_base[q];
            for (uint64_t j = wl + threadIdx.x; j < wh; j += blockDim.x)
                s_tails[base + (j - wl)] = tq[j];
        }
    }
    __syncthreads();
    if (t < total) {
        uint64_t rank = i - lo[r];
        int shadow = 0;
        bool fb = s_fallback != 0;
        for (int q = 0; q < R; q++) {
            if (q == r)
                continue;
            uint64_t qlo = b0 ? b0[q] : lo[q];
            uint64_t qhi = b1 ? b1[q] : hi[q];
            uint64_t bnd;
            if (!fb) {
                uint64_t off = s_base[q] - s_winlo[q];
                uint64_t l = qlo, h = qhi;
                if (q > r) {
                    while (l < h) {
                        uint64_t mid = (l + h) >> 1;
                        if (s_tails[off + mid] <= qw)
                            l = mid + 1;
                        else
                            h = mid;
                    }
                } else {
                    while (l < h) {
                        uint64_t mid = (l + h) >> 1;
                        if (s_tails[off + mid] < qw)
                            l = mid + 1;
                        else
                            h = mid;
                    }
                }
                bnd = l;
            } else {
                uint64_t kl;
                const uint8_t *k = run_key(runs[r], i, &kl);
                bnd = (q > r) ? dev_upper_bound(runs[q], k, kl, qlo, qhi)
                              : dev_lower_bound(runs[q], k, kl, qlo, qhi);
            }
            if (q > r && !shadow && bnd > lo[q] && runs[q].tails[bnd - 1] == qw)
                shadow = 1;
            rank += bnd - lo[q];
        }
        order[rank] = ((uint64_t)r << 40) | i;
        shadowed[rank] = (uint8_t)shadow;
    }
}

void launch_rank_ldst(const DevRun *d_runs, int R, const uint64_t *d_lo, const uint64_t *d_hi,
                      const uint64_t *d_wprefix, uint64_t total, uint64_t *d_order,
                      uint8_t *d_shadow, const uint64_t *d_bt_off, const uint64_t *d_bt,
                      int bt_shift, hipStream_t s)
{
    uint64_t blocks = (total + BLOCK - 1) / BLOCK;
    if (blocks == 0)
        blocks = 1;
    k_rank_ldst<<<dim3((uint32_t)blocks), dim3(BLOCK), 0, s>>>(
        d_runs, R, d_lo, d_hi, d_wprefix, total, d_order, d_shadow, d_bt_off, d_bt, bt_shift);
}

/* visible flag: newest version of its key (first of the equal-key group in
 * (key asc, run desc) order) and not a tombstone */
__global__ void k_visible(const DevRun *runs, const uint64_t *order, const uint8_t *shadowed,
                          uint64_t m, uint64_t *flags)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < m;
         p += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t id = order[p];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        flags[p] = (!shadowed[p] && !(r.sk[i] & 1)) ? 1 : 0;
    }
}

/* gather order entries whose flag is set, by exclusive prefix positions */
__global__ void k_gather(const uint64_t *order, const uint64_t *flags, const uint64_t *pos,
                         uint64_t m, uint64_t *out)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < m;
         p += gridDim.x * (uint64_t)blockDim.x) {
        if (flags[p])
            out[pos[p]] = order[p];
    }
}

/* ================= prefix sum (u64, exclusive) ================= */
__global__ void k_psum1(const uint64_t *in, uint64_t *out, uint64_t *blocksums, uint64_t n)
{
    __shared__ uint64_t lds[BLOCK];
    uint64_t base = (uint64_t)blockIdx.x * PSUM_BLOCK_ITEMS;
    uint64_t tbase = base + (uint64_t)threadIdx.x * PSUM_ITEMS_PER_THREAD;
    uint64_t loc[PSUM_ITEMS_PER_THREAD];
    uint64_t s = 0;
    for (int j = 0; j < PSUM_ITEMS_PER_THREAD; j++) {
        uint64_t idx = tbase + j;
        loc[j] = s;
        s += (idx < n) ? in[idx] : 0;
    }
    lds[threadIdx.x] = s;
    __syncthreads();
    /* Hillis-Steele inclusive scan over 256 thread sums */
    for (int off = 1; off < BLOCK; off <<= 1) {
        uint64_t v = (threadIdx.x >= (unsigned)off) ? lds[threadIdx.x - off] : 0;
        __syncthreads();
        lds[threadIdx.x] += v;
        __syncthreads();
    }
    uint64_t texcl = (threadIdx.x == 0) ? 0 : lds[threadIdx.x - 1];
    for (int j = 0; j < PSUM_ITEMS_PER_THREAD; j++) {
        uint64_t idx = tbase + j;
        if (idx < n)
            out[idx] = texcl + loc[j];
    }
    if (threadIdx.x == BLOCK - 1 && blocksums)
        blocksums[blockIdx.x] = lds[BLOCK - 1];
}

__global__ void k_psum_add(uint64_t *out, const uint64_t *blockoffs, uint64_t n)
{
    uint64_t b = blockIdx.x;
    uint64_t base = b * PSUM_BLOCK_ITEMS;
    uint64_t add = blockoffs[b];
    for (uint64_t i = base + threadIdx.x; i < base + PSUM_BLOCK_ITEMS && i < n; i += blockDim.x)
        out[i] += add;
}

/* ================= point get =================
 * on_get / DB::Get equivalent (pegasus_server_impl.cpp:441): search newest
 * run first; first hit wins (run seqno ranges are ordered). */
__global__ void k_get(const DevRun *runs, int R, const uint8_t *qkeys, const uint64_t *qoffs,
                      uint64_t nq, uint32_t epoch_now, uint32_t dv, int32_t *status,
                      uint64_t *hit, uint64_t *ulen /* user-data len */, uint32_t *expire_out)
{
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < nq;
         t += gridDim.x * (uint64_t)blockDim.x) {
        const uint8_t *k = qkeys + qoffs[t];
        uint64_t kl = qoffs[t + 1] - qoffs[t];
        int32_t st = 1; /* NotFound */
        uint64_t h = 0, ul = 0;
        uint32_t eo = 0;
        uint64_t bh = bloom_hash(k, kl);
        for (int r = R - 1; r >= 0; r--) {
            if (!bloom_maybe_has(runs[r], bh))
                continue; /* bloom negative: run cannot contain the key */
            uint64_t i = dev_lower_bound(runs[r], k, kl, 0, runs[r].n);
            if (i >= runs[r].n)
                continue;
            uint64_t ml;
            const uint8_t *mk = run_key(runs[r], i, &ml);
            if (dev_key_cmp(mk, ml, k, kl) != 0)
                continue;
            /* newest version */
            if (runs[r].sk[i] & 1)
                break; /* tombstone -> NotFound */
            uint64_t vl;
            const uint8_t *v = run_val(runs[r], i, &vl);
            uint32_t e = dev_expire_ts(dv, v);
            if (dev_ts_expired(epoch_now, e))
                break; /* expired -> NotFound (on_get:444) */
            st = 0;
            h = ((uint64_t)r << 40) | i;
            ul = vl - dev_hdr_len(dv);
            eo = e;
            break;
        }
        status[t] = st;
        hit[t] = h;
        ulen[t] = ul;
        if (expire_out)
            expire_out[t] = eo;
    }
}

/* emit user data for selected hits (wave per record, lane-strided bytes) */
__global__ void k_emit_values(const DevRun *runs, const uint64_t *hit, const int32_t *status,
                              uint64_t nq, uint32_t dv, const uint64_t *voffs, uint8_t *vout)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    uint32_t hdr = dev_hdr_len(dv);
    for (uint64_t t = wave; t < nq; t += nwaves) {
        if (status[t] != 0)
            continue;
        uint64_t id = hit[t];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull, vl;
        const uint8_t *v = run_val(r, i, &vl);
        wave_copy(vout + voffs[t], v + hdr, vl - hdr, lane);
    }
}

/* ================= scan state =================
 * validate_key_value_for_scan (pegasus_server_impl.cpp:2382-2432) applied to
 * a window of the visible view; also writes out key/value sizes. */
__global__ void k_scan_state(const DevRun *runs, const uint64_t *view, uint64_t w, ScanParams sp,
                             uint8_t *state, uint64_t *ksz, uint64_t *vsz)
{
    uint32_t hdr = dev_hdr_len(sp.data_version);
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < w;
         p += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t id = view[p];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        uint64_t kl = r.fixed_klen ? r.fixed_klen : (r.koff[i + 1] - r.koff[i]);
        uint64_t vl = r.fixed_vlen ? r.fixed_vlen : (r.voff[i + 1] - r.voff[i]);
        const uint8_t *k = r.keys + (r.fixed_klen ? i * r.fixed_klen : r.koff[i]);
        uint32_t ets;
        if (r.meta) {
            ets = (uint32_t)(r.meta[i] >> 32);
        } else {
            uint64_t vtmp;
            ets = dev_expire_ts(sp.data_version, run_val(r, i, &vtmp));
        }
        int st = ST_NORMAL;
        if (dev_ts_expired(sp.epoch_now, ets)) {
            st = ST_EXPIRED;
        } else if (sp.validate_hash &&
                   (sp.partition_version < 0 || sp.pidx > sp.partition_version ||
                    (int64_t)(dev_key_hash(k, kl) & (uint64_t)sp.partition_version) !=
                        (int64_t)sp.pidx)) {
            st = ST_HASH_INVALID;
        } else if (sp.hk_ft != 0 || sp.sk_ft != 0) {
            uint32_t hklen = ((uint32_t)k[0] << 8) | k[1];
            const uint8_t *hk = k + 2;
            const uint8_t *skp = k + 2 + hklen;
            uint64_t sklen = kl - 2 - hklen;
            if (sp.hk_ft != 0 &&
                !dev_validate_filter(sp.hk_ft, sp.hk_pat, sp.hk_pat_len, hk, hklen))
                st = ST_FILTERED;
            else if (sp.sk_ft != 0 &&
                     !dev_validate_filter(sp.sk_ft, sp.sk_pat, sp.sk_pat_len, skp, sklen))
                st = ST_FILTERED;
        }
        state[p] = (uint8_t)st;
        ksz[p] = (st == ST_NORMAL) ? (kl - sp.hash_key_skip) : 0;
        vsz[p] = (st == ST_NORMAL && !sp.no_value) ? (vl - hdr) : 0;
    }
}

/* normal-flag (1/0) from state, zeroed at/after the cutoff */
__global__ void k_normal_flags(const uint8_t *state, uint64_t w, uint64_t *flags)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < w;
         p += gridDim.x * (uint64_t)blockDim.x)
        flags[p] = (state[p] == ST_NORMAL) ? 1 : 0;
}

/* zero sizes for entries at/after consumed cutoff or non-normal */
__global__ void k_cut_sizes(const uint8_t *state, uint64_t w, uint64_t consumed, uint64_t *ksz,
                            uint64_t *vsz)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < w;
         p += gridDim.x * (uint64_t)blockDim.x) {
        if (p >= consumed || state[p] != ST_NORMAL) {
            ksz[p] = 0;
            vsz[p] = 0;
        }
    }
}

/* first p with nprefix[p] >= batch_count (inclusive-scan semantics handled by
 * caller passing exclusive prefix + flags) — single thread, tiny */
__global__ void k_cutoff(const uint64_t *nprefix_excl, const uint64_t *flags, uint64_t w,
                         uint64_t batch_count, uint64_t *out)
{
    if (blockIdx.x != 0 || threadIdx.x != 0)
        return;
    /* consumed = index just after the batch_count-th normal record, or w */
    uint64_t lo = 0, hi = w;
    /* nprefix_excl[p] + flags[p] = inclusive count at p */
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1;
        uint64_t incl = nprefix_excl[mid] + flags[mid];
        if (incl < batch_count)
            lo = mid + 1;
        else
            hi = mid;
    }
    out[0] = (lo < w) ? lo + 1 : w;
    /* also export the number of normals consumed */
    out[1] = (lo < w) ? batch_count : (w ? nprefix_excl[w - 1] + flags[w - 1] : 0);
}

/* emit scan kvs: for each consumed normal record, copy (key minus
 * hash_key_skip) and user-data value into packed buffers, and write offsets +
 * optional expire_ts.  Wave per record. */
__global__ void k_emit_scan(const DevRun *runs, const uint64_t *view, uint64_t w,
                            const uint8_t *state, uint64_t consumed, const uint64_t *npos,
                            const uint64_t *koffs, const uint64_t *voffs, ScanParams sp,
                            uint8_t *kout, uint8_t *vout, uint64_t *kout_offs,
                            uint64_t *vout_offs, int32_t *ets_out, uint64_t n_out)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    uint32_t hdr = dev_hdr_len(sp.data_version);
    for (uint64_t p = wave; p < consumed; p += nwaves) {
        if (state[p] != ST_NORMAL)
            continue;
        uint64_t o = npos[p]; /* output row index */
        uint64_t id = view[p];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        uint64_t kl, vl;
        const uint8_t *k = run_key(r, i, &kl);
        const uint8_t *v = run_val(r, i, &vl);
        const uint8_t *ksrc = k + sp.hash_key_skip;
        uint64_t kn = kl - sp.hash_key_skip;
        wave_copy(kout + koffs[p], ksrc, kn, lane);
        if (!sp.no_value) {
            wave_copy(vout + voffs[p], v + hdr, vl - hdr, lane);
        }
        if (lane == 0) {
            kout_offs[o] = koffs[p];
            vout_offs[o] = voffs[p];
            if (ets_out)
                ets_out[o] = (int32_t)dev_expire_ts(sp.data_version, v);
            if (o == n_out - 1) {
                kout_offs[n_out] = koffs[p] + kn;
                vout_offs[n_out] = voffs[p] + ((!sp.no_value) ? (vl - hdr) : 0);
            }
        }
    }
}

/* emit an explicit host-selected list of view rows (multi_get caps are
 * applied host-side over a <=3000-entry window, mirroring the reference's
 * limiter loop exactly).  rows[j] = view position; koffs/voffs per j. */
__global__ void k_emit_rows(const DevRun *runs, const uint64_t *view, const uint64_t *rows,
                            uint64_t n_rows, const uint64_t *koffs, const uint64_t *voffs,
                            ScanParams sp, uint8_t *kout, uint8_t *vout)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    uint32_t hdr = dev_hdr_len(sp.data_version);
    for (uint64_t j = wave; j < n_rows; j += nwaves) {
        uint64_t id = view[rows[j]];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        uint64_t kl, vl;
        const uint8_t *k = run_key(r, i, &kl);
        const uint8_t *v = run_val(r, i, &vl);
        const uint8_t *ksrc = k + sp.hash_key_skip;
        uint64_t kn = kl - sp.hash_key_skip;
        wave_copy(kout + koffs[j], ksrc, kn, lane);
        if (!sp.no_value) {
            wave_copy(vout + voffs[j], v + hdr, vl - hdr, lane);
        }
    }
}

/* compare the key of view[0] with a given key; out[0]=1 if equal */
__global__ void k_first_eq(const DevRun *runs, const uint64_t *view, uint64_t n,
                           const uint8_t *key, uint64_t klen, uint32_t *out)
{
    if (blockIdx.x != 0 || threadIdx.x != 0)
        return;
    out[0] = 0;
    if (n == 0)
        return;
    uint64_t id = view[0];
    const DevRun &r = runs[id >> 40];
    uint64_t i = id & 0xFFFFFFFFFFull, kl;
    const uint8_t *k = run_key(r, i, &kl);
    out[0] = dev_key_cmp(k, kl, key, klen) == 0;
}

/* it->Valid() restatement for multi_get's limit-exit (on_multi_get:777-788):
 * does any rocksdb-iterator-visible record exist beyond the range boundary?
 * Visible = the newest version of its key group is a PUT (tombstone groups
 * are merged away by the iterator; expiry is app-level and does NOT hide
 * records here).  Walks merged key groups outward from the boundary —
 * expected O(1) groups (the first group is usually a live PUT).
 * forward: first group with key >= bound; reverse: first with key < bound. */
__device__ static int dev_valid_beyond(const DevRun *runs, int R, const uint8_t *bound,
                                       uint64_t blen, int reverse)
{
    uint64_t cur[RRDB_MAX_RUNS];
    if (!reverse) {
        for (int q = 0; q < R; q++)
            cur[q] = dev_lower_bound(runs[q], bound, blen, 0, runs[q].n);
        for (;;) {
            int best = -1;
            const uint8_t *bk = nullptr;
            uint64_t bl = 0;
            for (int q = R - 1; q >= 0; q--) { /* ties: newest (highest q) first */
                if (cur[q] >= runs[q].n)
                    continue;
                uint64_t kl;
                const uint8_t *k = run_key(runs[q], cur[q], &kl);
                if (best < 0 || dev_key_cmp(k, kl, bk, bl) < 0) {
                    best = q;
                    bk = k;
                    bl = kl;
                }
            }
            if (best < 0)
                return 0; /* exhausted the DB: iterator invalid */
            if (!(runs[best].sk[cur[best]] & 1))
                return 1; /* newest version is a PUT: iterator lands here */
            for (int q = 0; q < R; q++) /* tombstone group: skip the key */
                cur[q] = dev_upper_bound(runs[q], bk, bl, cur[q], runs[q].n);
        }
    }
    for (int q = 0; q < R; q++)
        cur[q] = dev_lower_bound(runs[q], bound, blen, 0, runs[q].n);
    for (;;) {
        int best = -1;
        const uint8_t *bk = nullptr;
        uint64_t bl = 0;
        for (int q = R - 1; q >= 0; q--) {
            if (cur[q] == 0)
                continue;
            uint64_t kl;
            const uint8_t *k = run_key(runs[q], cur[q] - 1, &kl);
            if (best < 0 || dev_key_cmp(k, kl, bk, bl) > 0) {
                best = q;
                bk = k;
                bl = kl;
            }
        }
        if (best < 0)
            return 0;
        if (!(runs[best].sk[cur[best] - 1] & 1))
            return 1;
        for (int q = 0; q < R; q++)
            cur[q] = dev_lower_bound(runs[q], bk, bl, 0, cur[q]);
    }
}

__global__ void k_valid_beyond(const DevRun *runs, int R, const uint8_t *bound, uint64_t blen,
                               int reverse, uint32_t *out)
{
    if (blockIdx.x != 0 || threadIdx.x != 0)
        return;
    out[0] = (uint32_t)dev_valid_beyond(runs, R, bound, blen, reverse);
}

void launch_valid_beyond(const DevRun *runs, int R, const uint8_t *bound, uint64_t blen,
                         int reverse, uint32_t *out, hipStream_t s)
{
    k_valid_beyond<<<1, 1, 0, s>>>(runs, R, bound, blen, reverse, out);
}

/* ================= compaction =================
 * Disposition of every record in the full-range order[] array:
 * KeyWithTTLCompactionFilter::Filter (key_ttl_compaction_filter.h:55-92) on
 * the surviving newest PUT; tombstones and shadowed versions dropped
 * (bottommost CompactRange, do_manual_compact:3389). */
/* per-record disposition (stats aggregated per wave via ballot — a single
 * shared atomic per record serializes at ~88 adds/us and was 85% of the
 * compaction pass before this) */
enum { D_NONE = 0, D_KEEP, D_SHADOWED, D_TOMBSTONE, D_EXPIRED, D_FILTERED };

/* record-level KeyWithTTLCompactionFilter::Filter restatement
 * (key_ttl_compaction_filter.h:55-92); returns D_*, fills outputs on KEEP.
 * Hot path reads only the 8B meta column ((expire<<32)|kind, built at run
 * creation) — key bytes are touched only under user rules / hash
 * validation, value bytes never (no rule inspects value data). */
__device__ static int dev_disposition(const DevRun &r, uint64_t i, const CompactParams &cp,
                                      int shadow, uint8_t *changed_out, uint32_t *new_ts_out,
                                      uint64_t *kl_out, uint64_t *vl_out,
                                      const uint64_t *mword = nullptr)
{
    *changed_out = 0;
    *new_ts_out = 0;
    *kl_out = 0;
    *vl_out = 0;
    if (shadow)
        return D_SHADOWED; /* newest-wins, decided during ranking */
    uint32_t expire_ts;
    if (mword || r.meta) {
        uint64_t m = mword ? *mword : r.meta[i];
        if (m & 1)
            return D_TOMBSTONE;
        expire_ts = (uint32_t)(m >> 32);
    } else {
        if (r.sk[i] & 1)
            return D_TOMBSTONE;
        uint64_t vl0;
        expire_ts = dev_expire_ts(cp.data_version, run_val(r, i, &vl0));
    }
    uint64_t kl =
        r.fixed_klen ? r.fixed_klen : (r.koff[i + 1] - r.koff[i]);
    uint64_t vl =
        r.fixed_vlen ? r.fixed_vlen : (r.voff[i + 1] - r.voff[i]);
    int drop = 0, value_changed = 0;
    uint32_t new_ts_val = 0;
    uint32_t eff_expire = expire_ts; /* value_view's expire after default-ttl */
    if (kl >= 2) {
        if (cp.default_ttl != 0 && expire_ts == 0) {
            expire_ts = cp.epoch_now + cp.default_ttl;
            eff_expire = expire_ts;
            value_changed = 1;
            new_ts_val = expire_ts;
        }
        if (cp.n_ops > 0) {
            const uint8_t *k = r.keys + (r.fixed_klen ? i * r.fixed_klen : r.koff[i]);
            uint32_t hklen = ((uint32_t)k[0] << 8) | k[1];
            const uint8_t *hk = k + 2;
            const uint8_t *skp = k + 2 + hklen;
            uint64_t sklen = kl - 2 - hklen;
            /* value_view fixed at loop entry: rules see eff_expire (post
             * default-ttl), not later op rewrites (filter :82-89) */
            for (int oi = 0; oi < cp.n_ops && !drop; oi++) {
                const DevOp &op = cp.ops[oi];
                int all = (op.n_rules > 0);
                for (int ri = 0; ri < op.n_rules && all; ri++) {
                    const DevRule &rule = cp.rules[op.rule_off + ri];
                    if (rule.type == DFR_TTL_RANGE) {
                        uint32_t e = eff_expire;
                        all = (e == 0 && rule.start_ttl == 0 && rule.stop_ttl == 0) ||
                              ((uint32_t)(rule.start_ttl + cp.epoch_now) <= e &&
                               (uint32_t)(rule.stop_ttl + cp.epoch_now) >= e);
                    } else if (rule.type == DFR_HASHKEY) {
                        all = dev_pattern_match(hk, hklen, rule.match_type,
                                                cp.pats + rule.pat_off, rule.pat_len);
                    } else {
                        all = dev_pattern_match(skp, sklen, rule.match_type,
                                                cp.pats + rule.pat_off, rule.pat_len);
                    }
                }
                if (!all)
                    continue;
                if (op.type == DOP_DELETE) {
                    drop = 1;
                    break;
                }
                /* update_ttl (compaction_operation.cpp:78-113) */
                uint32_t nts = 0;
                int apply = 1;
                switch (op.ut_type) {
                case DUT_FROM_NOW:
                    nts = cp.epoch_now + op.ut_value;
                    break;
                case DUT_FROM_CURRENT:
                    if (eff_expire == 0)
                        apply = 0;
                    else
                        nts = op.ut_value + eff_expire;
                    break;
                case DUT_TIMESTAMP:
                    nts = op.ut_value - 1451606400u;
                    break;
                default:
                    apply = 0;
                    break;
                }
                if (apply) {
                    value_changed = 1;
                    new_ts_val = nts;
                }
            }
        }
        if (drop)
            return D_FILTERED;
        /* final keep/drop on local expire_ts + stale split hash (:91,114-121) */
        if (dev_ts_expired(cp.epoch_now, expire_ts))
            return D_EXPIRED;
        if (cp.validate_hash && cp.partition_version >= 0 && cp.pidx <= cp.partition_version) {
            uint64_t kl2;
            const uint8_t *k2 = run_key(r, i, &kl2);
            if ((int64_t)(dev_key_hash(k2, kl2) & (uint64_t)cp.partition_version) !=
                (int64_t)cp.pidx)
                return D_FILTERED;
        }
    }
    *changed_out = (uint8_t)value_changed;
    *new_ts_out = new_ts_val;
    *kl_out = kl;
    *vl_out = vl;
    return D_KEEP;
}

/* fused rank + newest-wins + compaction filter: one pass over the input
 * records computes each record's merged position, shadow state and filter
 * disposition, writing keep/changed/new-expire/sizes scattered by rank.
 * Replaces the separate flags pass (which re-gathered every record through
 * the order array). */
__global__ void k_rank_compact(const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi,
                               const uint64_t *wprefix, uint64_t total, CompactParams cp,
                               uint64_t *order, uint64_t *keepw, uint8_t *changed,
                               uint32_t *new_expire, uint64_t *ksz, uint64_t *vsz,
                               uint64_t *rank_of /* [total] by input index, may be null */,
                               const uint64_t *bt_off, const uint64_t *bt, int bt_shift,
                               CompactStatsDev *stats)
{
    uint64_t tid = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
    uint64_t stride = gridDim.x * (uint64_t)blockDim.x;
    uint64_t iters = (total + stride - 1) / stride;
    int lane = threadIdx.x % WAVE;
    for (uint64_t it = 0; it < iters; it++) {
        uint64_t t = tid + it * stride;
        int disp = D_NONE;
        if (t < total) {
            int r = 0;
            while (wprefix[r + 1] <= t)
                r++;
            uint64_t i = lo[r] + (t - wprefix[r]);
            uint64_t kl;
            const uint8_t *k = run_key(runs[r], i, &kl);
            uint64_t rank = i - lo[r];
            int shadow = 0;
            const uint64_t *b0 = nullptr, *b1 = nullptr;
            if (bt) {
                uint64_t j = (i - lo[r]) >> bt_shift;
                b0 = bt + (bt_off[r] + j) * (uint64_t)R;
                b1 = bt + (bt_off[r] + j + 1) * (uint64_t)R;
            }
            for (int q = 0; q < R; q++) {
                if (q == r)
                    continue;
                uint64_t qlo = b0 ? b0[q] : lo[q];
                uint64_t qhi = b1 ? b1[q] : hi[q];
                if (q > r) {
                    uint64_t ub = dev_upper_bound(runs[q], k, kl, qlo, qhi);
                    if (!shadow && ub > lo[q]) {
                        uint64_t pl;
                        const uint8_t *pk = run_key(runs[q], ub - 1, &pl);
                        if (dev_key_cmp(pk, pl, k, kl) == 0)
                            shadow = 1;
                    }
                    rank += ub - lo[q];
                } else {
                    rank += dev_lower_bound(runs[q], k, kl, qlo, qhi) - lo[q];
                }
            }
            uint8_t ch;
            uint32_t nts;
            uint64_t okl, ovl;
            disp = dev_disposition(runs[r], i, cp, shadow, &ch, &nts, &okl, &ovl);
            order[rank] = ((uint64_t)r << 40) | i;
            keepw[rank] = (disp == D_KEEP) ? 1 : 0;
            if (changed)
                changed[rank] = ch;
            if (new_expire)
                new_expire[rank] = nts;
            if (ksz)
                ksz[rank] = okl;
            if (vsz)
                vsz[rank] = ovl;
            if (rank_of)
                rank_of[t] = rank;
        }
        unsigned long long b;
        b = __ballot(disp == D_SHADOWED);
        if (lane == 0 && b)
            atomicAdd(&stats->shadowed, (unsigned long long)__popcll(b));
        b = __ballot(disp == D_TOMBSTONE);
        if (lane == 0 && b)
            atomicAdd(&stats->tombstones, (unsigned long long)__popcll(b));
        b = __ballot(disp == D_EXPIRED);
        if (lane == 0 && b)
            atomicAdd(&stats->expired, (unsigned long long)__popcll(b));
        b = __ballot(disp == D_FILTERED);
        if (lane == 0 && b)
            atomicAdd(&stats->filtered, (unsigned long long)__popcll(b));
        b = __ballot(disp == D_KEEP);
        if (lane == 0 && b)
            atomicAdd(&stats->output_records, (unsigned long long)__popcll(b));
    }
}

/* ================= LDS-tails cooperative rank (rank_mode=3) =================
 * Same per-record rank+filter algorithm as k_rank_compact, but each 256-thread
 * workgroup stages the tail words of all windows its 256-record span probes
 * into LDS first.  Eligible only when every run is in single-word probe mode
 * with one shared cross-run prefix (host checks; see engine.cpp), so a probe
 * is one u64 compare and LDS holds ~2.3K words (~18KB) per workgroup — the
 * earlier full-key LDS experiment died at 1 wg/CU; this one keeps ~6.  Probes
 * whose first midpoints coincide across lanes become LDS broadcasts. */

__global__ void __launch_bounds__(BLOCK) k_rank_compact_ldst(
    const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi, const uint64_t *wprefix,
    uint64_t total, CompactParams cp, uint64_t *order, uint64_t *keepw, uint8_t *changed,
    uint32_t *new_expire, uint64_t *ksz, uint64_t *vsz, uint64_t *rank_of,
    const uint64_t *bt_off, const uint64_t *bt, int bt_shift, CompactStatsDev *stats)
{
    __shared__ uint64_t s_tails[LDST_CAP];
    __shared__ uint64_t s_winlo[LDST_MAXR], s_winhi[LDST_MAXR], s_base[LDST_MAXR];
    __shared__ int s_fallback;

    uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
    int lane = threadIdx.x % WAVE;

    if (threadIdx.x == 0)
        s_fallback = (bt == nullptr || R > LDST_MAXR) ? 1 : 0;
    if (threadIdx.x < (unsigned)(R < LDST_MAXR ? R : LDST_MAXR)) {
        s_winlo[threadIdx.x] = ~0ull;
        s_winhi[threadIdx.x] = 0;
    }
    __syncthreads();

    /* per-record identity + window pointers (kept in registers for phase 2) */
    int r = 0;
    uint64_t i = 0, qw = 0;
    const uint64_t *b0 = nullptr, *b1 = nullptr;
    if (t < total) {
        while (wprefix[r + 1] <= t)
            r++;
        i = lo[r] + (t - wprefix[r]);
        qw = runs[r].tails[i];
        if (bt) {
            uint64_t j = (i - lo[r]) >> bt_shift;
            b0 = bt + (bt_off[r] + j) * (uint64_t)R;
            b1 = bt + (bt_off[r] + j + 1) * (uint64_t)R;
        }
    }
    if (!s_fallback) { /* wave-reduced window union: one atomic per wave per run */
        for (int q = 0; q < R; q++) {
            uint64_t wl = (t < total && q != r && b0) ? b0[q] : ~0ull;
            uint64_t wh = (t < total && q != r && b1) ? b1[q] : 0;
            for (int d = WAVE / 2; d; d >>= 1) {
                uint64_t o1 = __shfl_xor(wl, d);
                if (o1 < wl)
                    wl = o1;
                uint64_t o2 = __shfl_xor(wh, d);
                if (o2 > wh)
                    wh = o2;
            }
            if (lane == 0 && wh > 0) {
                atomicMin((unsigned long long *)&s_winlo[q], (unsigned long long)wl);
                atomicMax((unsigned long long *)&s_winhi[q], (unsigned long long)wh);
            }
        }
    }
    __syncthreads();
    if (threadIdx.x == 0 && !s_fallback) {
        uint64_t tot = 0;
        for (int q = 0; q < R; q++) {
            s_base[q] = tot;
            if (s_winhi[q] > s_winlo[q])
                tot += s_winhi[q] - s_winlo[q];
        }
        if (tot > LDST_CAP)
            s_fallback = 1;
    }
    __syncthreads();
    if (!s_fallback) { /* stage the windows' tail words, coalesced per run */
        for (int q = 0; q < R; q++) {
            uint64_t wl = s_winlo[q], wh = s_winhi[q];
            if (wh <= wl)
                continue;
            const uint64_t *tq = runs[q].tails;
            uint64_t base = s_base[q];
            for (uint64_t j = wl + threadIdx.x; j < wh; j += blockDim.x)
                s_tails[base + (j - wl)] = tq[j];
        }
    }
    __syncthreads();

    int disp = D_NONE;
    if (t < total) {
        uint64_t rank = i - lo[r];
        int shadow = 0;
        bool fb = s_fallback != 0;
        for (int q = 0; q < R; q++) {
            if (q == r)
                continue;
            uint64_t qlo = b0 ? b0[q] : lo[q];
            uint64_t qhi = b1 ? b1[q] : hi[q];
            uint64_t bnd;
            if (!fb) {
                uint64_t off = s_base[q] - s_winlo[q];
                uint64_t l = qlo, h = qhi;
                if (q > r) {
                    while (l < h) {
                        uint64_t mid = (l + h) >> 1;
                        if (s_tails[off + mid] <= qw)
                            l = mid + 1;
                        else
                            h = mid;
                    }
                } else {
                    while (l < h) {
                        uint64_t mid = (l + h) >> 1;
                        if (s_tails[off + mid] < qw)
                            l = mid + 1;
                        else
                            h = mid;
                    }
                }
                bnd = l;
            } else {
                uint64_t kl;
                const uint8_t *k = run_key(runs[r], i, &kl);
                bnd = (q > r) ? dev_upper_bound(runs[q], k, kl, qlo, qhi)
                              : dev_lower_bound(runs[q], k, kl, qlo, qhi);
            }
            if (q > r) {
                if (!shadow && bnd > lo[q] && runs[q].tails[bnd - 1] == qw)
                    shadow = 1;
                rank += bnd - lo[q];
            } else {
                rank += bnd - lo[q];
            }
        }
        uint8_t ch;
        uint32_t nts;
        uint64_t okl, ovl;
        disp = dev_disposition(runs[r], i, cp, shadow, &ch, &nts, &okl, &ovl);
        order[rank] = ((uint64_t)r << 40) | i;
        keepw[rank] = (disp == D_KEEP) ? 1 : 0;
        if (changed)
            changed[rank] = ch;
        if (new_expire)
            new_expire[rank] = nts;
        if (ksz)
            ksz[rank] = okl;
        if (vsz)
            vsz[rank] = ovl;
        if (rank_of)
            rank_of[t] = rank;
    }
    unsigned long long b;
    b = __ballot(disp == D_SHADOWED);
    if (lane == 0 && b)
        atomicAdd(&stats->shadowed, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_TOMBSTONE);
    if (lane == 0 && b)
        atomicAdd(&stats->tombstones, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_EXPIRED);
    if (lane == 0 && b)
        atomicAdd(&stats->expired, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_FILTERED);
    if (lane == 0 && b)
        atomicAdd(&stats->filtered, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_KEEP);
    if (lane == 0 && b)
        atomicAdd(&stats->output_records, (unsigned long long)__popcll(b));
}

/* ================= group-streaming rank (rank_mode=4, default) =============
 * Round-2 replacement for the windowed-staging ldst rank.  The windowed
 * design staged every run's search windows once per 256-record span of EVERY
 * OTHER run, fetching each tail word ~R times (PMC r01: ~196 B/record
 * against a ~30 B algorithmic budget).  Here the merged output is cut into
 * GROUPS at anchor keys (every (1<<gs)-th key of one anchor run, full
 * comparator: key asc, run desc), which partitions every run's window into
 * DISJOINT segments: group g owns [anch[g][q], anch[g+1][q]) of run q.  One
 * workgroup streams its group's segments through LDS exactly once — global
 * traffic becomes one 8B tail read per record — and every element's exact
 * global rank is  base(g) + local rank,  base(g) = sum_q (anch[g][q]-lo[q]).
 * Local ranks come from in-LDS binary searches of the sibling segments
 * (expected ~R*log2(group/R) LDS probes, wave-coherent).  Output writes land
 * in the contiguous rank range [base, base+gsize): coalesced, unlike the
 * old full-range scatter.
 * Shadow (newest-wins) stays exact across group splits of an equal-key set:
 * a newer same-key version is either inside the group (found by the search)
 * or the LAST element of the previous group's segment (s_btail below).
 * Oversized groups (duplicate-heavy data) fall back per element to global
 * tail searches inside the same segment windows — same math, no staging. */

/* anchor table: rows g=0..n_groups; row g holds every run's boundary for
 * group g (row 0 = lo, row n_groups = hi).  Anchor key = run q0's record
 * lo[q0] + (g<<gs); searched in the other runs' packed tail words (word-
 * probe eligibility makes tail order == key order). */
__global__ void k_anchor_rows(const DevRun *runs, int R, int q0, const uint64_t *lo,
                              const uint64_t *hi, int gs, uint64_t n_groups, uint64_t *anch)
{
    uint64_t total = (n_groups + 1) * (uint64_t)R;
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < total;
         t += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t g = t / R;
        int q = (int)(t % R);
        uint64_t *cell = anch + t;
        if (g == 0) {
            *cell = lo[q];
            continue;
        }
        if (g == n_groups) {
            *cell = hi[q];
            continue;
        }
        uint64_t i = lo[q0] + (g << gs);
        if (i >= hi[q0]) {
            *cell = hi[q];
            continue;
        }
        if (q == q0) {
            *cell = i;
            continue;
        }
        uint64_t key_t = runs[q0].tails[i];
        const uint64_t *tq = runs[q].tails;
        uint64_t l = lo[q], h = hi[q];
        if (q > q0) { /* equal keys in newer runs rank BEFORE the anchor */
            while (l < h) {
                uint64_t mid = (l + h) >> 1;
                if (tq[mid] <= key_t)
                    l = mid + 1;
                else
                    h = mid;
            }
        } else {
            while (l < h) {
                uint64_t mid = (l + h) >> 1;
                if (tq[mid] < key_t)
                    l = mid + 1;
                else
                    h = mid;
            }
        }
        *cell = l;
    }
}

void launch_anchor_rows(const DevRun *runs, int R, int q0, const uint64_t *d_lo,
                        const uint64_t *d_hi, int gs, uint64_t n_groups, uint64_t *d_anch,
                        hipStream_t s)
{
    uint64_t total = (n_groups + 1) * (uint64_t)R;
    k_anchor_rows<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(runs, R, q0, d_lo, d_hi, gs,
                                                           n_groups, d_anch);
}

static_assert(GRP_CAP <= (1 << GRP_ORG_SHIFT), "org segpos bits");
static_assert(LDST_MAXR <= (1 << (16 - GRP_ORG_SHIFT)), "org run bits");

/* (tail asc, run desc) strict order on (tail, org) pairs — tails are unique
 * within a run, so ties are always cross-run and org's run bits decide */
__device__ static inline bool grp_lt(uint64_t ta, uint16_t oa, uint64_t tb, uint16_t ob)
{
    return ta < tb || (ta == tb && (oa >> GRP_ORG_SHIFT) > (ob >> GRP_ORG_SHIFT));
}

/* per-thread disposition tallies, flushed ONCE per block at kernel end into
 * one of 8 stat banks (atomicAdds to a single CompactStatsDev serialized at
 * ~88 adds/us and put a ~1.5ms floor under every fused rank kernel) */
struct GrpTally {
    uint32_t shadowed = 0, tombstones = 0, expired = 0, filtered = 0, keep = 0;
    __device__ void add(int disp)
    {
        shadowed += (disp == D_SHADOWED);
        tombstones += (disp == D_TOMBSTONE);
        expired += (disp == D_EXPIRED);
        filtered += (disp == D_FILTERED);
        keep += (disp == D_KEEP);
    }
    __device__ void flush(CompactStatsDev *banks /* [8] */)
    {
        /* wave-reduce each counter, lane 0 adds to the block's bank */
        uint64_t a = ((uint64_t)shadowed << 32) | tombstones;
        uint64_t b = ((uint64_t)expired << 32) | filtered;
        uint64_t c = keep;
        for (int d = WAVE / 2; d; d >>= 1) {
            a += __shfl_xor(a, d);
            b += __shfl_xor(b, d);
            c += __shfl_xor(c, d);
        }
        if ((threadIdx.x % WAVE) == 0) {
            CompactStatsDev *st = banks + (blockIdx.x & 7);
            if (a >> 32)
                atomicAdd(&st->shadowed, (unsigned long long)(a >> 32));
            if (a & 0xFFFFFFFFull)
                atomicAdd(&st->tombstones, (unsigned long long)(a & 0xFFFFFFFFull));
            if (b >> 32)
                atomicAdd(&st->expired, (unsigned long long)(b >> 32));
            if (b & 0xFFFFFFFFull)
                atomicAdd(&st->filtered, (unsigned long long)(b & 0xFFFFFFFFull));
            if (c)
                atomicAdd(&st->output_records, (unsigned long long)c);
        }
    }
};


/* countable record for the fused count scan (validate_key_value_for_scan,
 * pegasus_server_impl.cpp:2382-2432, count-only flavor): visible PUT, not
 * expired, passes hash validation + hash/sort-key filters */
__device__ static inline int dev_count_ok(const DevRun &r, uint64_t i, const ScanParams &sp)
{
    uint32_t ets;
    int kind;
    if (r.meta) {
        uint64_t m = r.meta[i];
        kind = (int)(m & 1);
        ets = (uint32_t)(m >> 32);
    } else {
        kind = (int)(r.sk[i] & 1);
        uint64_t vtmp;
        ets = kind ? 0 : dev_expire_ts(sp.data_version, run_val(r, i, &vtmp));
    }
    if (kind)
        return 0; /* tombstone: invisible to the iterator */
    if (dev_ts_expired(sp.epoch_now, ets))
        return 0;
    if (sp.validate_hash || sp.hk_ft != 0 || sp.sk_ft != 0) {
        uint64_t kl;
        const uint8_t *k = run_key(r, i, &kl);
        if (sp.validate_hash &&
            (sp.partition_version < 0 || sp.pidx > sp.partition_version ||
             (int64_t)(dev_key_hash(k, kl) & (uint64_t)sp.partition_version) !=
                 (int64_t)sp.pidx))
            return 0;
        if (sp.hk_ft != 0 || sp.sk_ft != 0) {
            uint32_t hklen = ((uint32_t)k[0] << 8) | k[1];
            const uint8_t *hk = k + 2;
            const uint8_t *skp = k + 2 + hklen;
            uint64_t sklen = kl - 2 - hklen;
            if (sp.hk_ft != 0 &&
                !dev_validate_filter(sp.hk_ft, sp.hk_pat, sp.hk_pat_len, hk, hklen))
                return 0;
            if (sp.sk_ft != 0 &&
                !dev_validate_filter(sp.sk_ft, sp.sk_pat, sp.sk_pat_len, skp, sklen))
                return 0;
        }
    }
    return 1;
}

/* per-element epilogue of the group rank: MODE 0 = compaction outputs,
 * MODE 1 = view (order + shadowed), MODE 2 = fused count (no arrays) */
template <int MODE>
__device__ static inline int grp_epilogue(const DevRun *runs, int q, uint64_t i, int shadow,
                                          uint64_t rank, const CompactParams &cp,
                                          const ScanParams &sp, uint64_t *order,
                                          uint64_t *keepw, uint8_t *changed,
                                          uint32_t *new_expire, uint64_t *ksz, uint64_t *vsz,
                                          uint8_t *shadowed, const uint64_t *smeta)
{
    if (MODE == 2) {
        if (shadow)
            return D_NONE;
        if (smeta) { /* staged meta: LDS fast path for the common case */
            uint64_t m = *smeta;
            if ((m & 1) || dev_ts_expired(sp.epoch_now, (uint32_t)(m >> 32)))
                return D_NONE;
            if (!sp.validate_hash && sp.hk_ft == 0 && sp.sk_ft == 0)
                return D_KEEP;
        }
        return dev_count_ok(runs[q], i, sp) ? D_KEEP : D_NONE;
    }
    order[rank] = ((uint64_t)q << 40) | i;
    if (MODE == 1) {
        shadowed[rank] = (uint8_t)shadow;
        return D_NONE;
    }
    uint8_t ch;
    uint32_t nts;
    uint64_t okl, ovl;
    int disp = dev_disposition(runs[q], i, cp, shadow, &ch, &nts, &okl, &ovl, smeta);
    keepw[rank] = (disp == D_KEEP) ? 1 : 0;
    if (changed)
        changed[rank] = ch;
    if (new_expire)
        new_expire[rank] = nts;
    if (ksz)
        ksz[rank] = okl;
    if (vsz)
        vsz[rank] = ovl;
    return disp;
}

/* MODE=0: compaction (fused KeyWithTTLCompactionFilter disposition outputs);
 * MODE=1: scan view build (order + shadowed only);
 * MODE=2: fused count scan (tallies only — the count_data hot path).
 * Staged groups rank by PAIRWISE MERGE-PATH ROUNDS in LDS (log2(R) rounds;
 * each round every element moves once, each thread owns a fixed output-slot
 * range found by one diagonal search) — ~5-10x fewer instructions per
 * element than per-element binary searches, which measured issue-bound. */
/* 256-thread workgroups: 512 was tried (8 waves/WG, double co-residency at
 * the same LDS) and REGRESSED (rank 0.32 -> 0.43ms): wider barriers and
 * the serial thread-0 sections stall twice the waves */
#define GRP_BLOCK 256
template <int MODE>
__global__ void __launch_bounds__(GRP_BLOCK, 6) k_rank_grp(
    const DevRun *runs, int R, const uint64_t *lo, const uint64_t *anch, uint64_t n_groups,
    CompactParams cp, ScanParams sp, uint64_t *order, uint64_t *keepw, uint8_t *changed,
    uint32_t *new_expire, uint64_t *ksz, uint64_t *vsz, uint64_t *rank_of, uint8_t *shadowed,
    CompactStatsDev *stats)
{
    __shared__ uint64_t s_ta[GRP_CAP], s_tb[GRP_CAP]; /* tails ping-pong */
    __shared__ uint16_t s_oa[GRP_CAP], s_ob[GRP_CAP]; /* (q<<12)|segpos */
    /* staged disposition column — only the count mode pays its LDS */
    constexpr int MCAP = (MODE == 2) ? GRP_CAP : 1;
    __shared__ uint64_t s_meta[MCAP];
    __shared__ uint64_t s_a0[LDST_MAXR], s_seglen[LDST_MAXR], s_segoff[LDST_MAXR + 1];
    __shared__ uint64_t s_btail[LDST_MAXR];
    __shared__ uint64_t s_loff[LDST_MAXR + 1]; /* merge-list offsets (ping) */
    __shared__ uint64_t s_loff2[LDST_MAXR + 1]; /* merge-list offsets (pong) */
    __shared__ uint64_t s_base, s_gsize;
    __shared__ uint32_t s_bmask;
    __shared__ int s_nl, s_allmeta;

    GrpTally tally;
    for (uint64_t g = blockIdx.x; g < n_groups; g += gridDim.x) {
        if (threadIdx.x < (unsigned)R) {
            int q = (int)threadIdx.x;
            uint64_t a0 = anch[g * R + q], a1 = anch[(g + 1) * R + q];
            s_a0[q] = a0;
            s_seglen[q] = a1 - a0;
            s_btail[q] = (a0 > lo[q]) ? runs[q].tails[a0 - 1] : 0;
            if (q == 0)
                s_bmask = 0;
        }
        __syncthreads();
        if (threadIdx.x < (unsigned)R) {
            int q = (int)threadIdx.x;
            if (s_a0[q] > lo[q])
                atomicOr(&s_bmask, 1u << q);
        }
        if (threadIdx.x == 0) {
            uint64_t t = 0, b = 0;
            int am = 1;
            for (int q = 0; q < R; q++) {
                s_segoff[q] = t;
                s_loff[q] = t;
                t += s_seglen[q];
                b += s_a0[q] - lo[q];
                if (!runs[q].meta)
                    am = 0;
            }
            s_segoff[R] = t;
            s_loff[R] = t;
            s_gsize = t;
            s_base = b;
            s_nl = R;
            s_allmeta = am;
        }
        __syncthreads();
        uint64_t gsize = s_gsize;
        bool staged = gsize <= GRP_CAP;
        if (staged) {
            /* stream each segment through LDS exactly once, org alongside */
            for (int q = 0; q < R; q++) {
                uint64_t len = s_seglen[q], off = s_segoff[q], a0 = s_a0[q];
                const uint64_t *tq = runs[q].tails;
                /* stage the disposition column only for the count mode:
                 * compaction measured faster with direct gathers (the extra
                 * 8KB LDS costs a workgroup of occupancy) */
                const uint64_t *mq =
                    (MODE == 2 && s_allmeta) ? runs[q].meta : nullptr;
                for (uint64_t j = threadIdx.x; j < len; j += blockDim.x) {
                    s_ta[off + j] = tq[a0 + j];
                    s_oa[off + j] = (uint16_t)((q << GRP_ORG_SHIFT) | (uint32_t)j);
                    if (mq)
                        s_meta[off + j] = mq[a0 + j];
                }
            }
            __syncthreads();
            /* pairwise merge rounds; ping-pong a->b->a->...  The list-offset
             * table ping-pongs too: the next round's offsets are a pure
             * function of this round's (loff'[x] = loff[2x]), so a few lanes
             * write them into the pong table DURING the merge loop and each
             * round needs ONE barrier (the former thread-0 rebuild +
             * second barrier measured as part of the ~88%-parked SQ time) */
            uint64_t E = (gsize + blockDim.x - 1) / blockDim.x;
            int cur = 0;
            int nl = s_nl;
            while (nl > 1) {
                const uint64_t *st = cur ? s_tb : s_ta;
                const uint16_t *so = cur ? s_ob : s_oa;
                uint64_t *dt = cur ? s_ta : s_tb;
                uint16_t *do_ = cur ? s_oa : s_ob;
                const uint64_t *lofc = cur ? s_loff2 : s_loff;
                uint64_t *lofn = cur ? s_loff : s_loff2;
                int nn = (nl + 1) / 2;
                if (threadIdx.x <= (unsigned)nn)
                    lofn[threadIdx.x] =
                        (2 * (int)threadIdx.x <= nl) ? lofc[2 * threadIdx.x] : gsize;
                uint64_t slot = (uint64_t)threadIdx.x * E;
                uint64_t send = slot + E;
                if (send > gsize)
                    send = gsize;
                while (slot < send) {
                    /* pair j covering this slot */
                    int j = 0;
                    while (2 * j + 2 < nl + 1 && lofc[2 * j + 2] <= slot)
                        j++;
                    uint64_t ka = lofc[2 * j];
                    uint64_t kb = (2 * j + 1 <= nl) ? lofc[2 * j + 1] : gsize;
                    uint64_t ke = (2 * j + 2 <= nl) ? lofc[2 * j + 2] : gsize;
                    uint64_t la = kb - ka, lb = ke - kb;
                    uint64_t k = slot - ka;
                    /* diagonal: i = #A elements among the first k of the pair */
                    uint64_t l = k > lb ? k - lb : 0, h = k < la ? k : la;
                    while (l < h) {
                        uint64_t mid = (l + h) >> 1;
                        if (grp_lt(st[ka + mid], so[ka + mid], st[kb + k - mid - 1],
                                   so[kb + k - mid - 1]))
                            l = mid + 1;
                        else
                            h = mid;
                    }
                    uint64_t ia = ka + l, ib = kb + (k - l);
                    uint64_t end = send < ke ? send : ke;
                    for (; slot < end; slot++) {
                        bool ta = (ib >= ke) ||
                                  (ia < kb && grp_lt(st[ia], so[ia], st[ib], so[ib]));
                        if (ta) {
                            dt[slot] = st[ia];
                            do_[slot] = so[ia++];
                        } else {
                            dt[slot] = st[ib];
                            do_[slot] = so[ib++];
                        }
                    }
                }
                __syncthreads();
                cur ^= 1;
                nl = nn;
            }
            const uint64_t *ft = cur ? s_tb : s_ta;
            const uint16_t *fo = cur ? s_ob : s_oa;
            /* MODE 0: the retired ping-pong buffer stages the meta column
             * (coalesced segment-major loads instead of per-element gathers
             * in merged order) — costs no extra LDS */
            uint64_t *s_mt = cur ? s_ta : s_tb;
            if (MODE == 0 && s_allmeta) {
                for (int q = 0; q < R; q++) {
                    uint64_t len = s_seglen[q], off = s_segoff[q], a0 = s_a0[q];
                    const uint64_t *mq = runs[q].meta;
                    for (uint64_t j = threadIdx.x; j < len; j += blockDim.x)
                        s_mt[off + j] = mq[a0 + j];
                }
                __syncthreads();
            }
            /* output phase: position p in the merged group == local rank */
            uint64_t iters = (gsize + blockDim.x - 1) / blockDim.x;
            for (uint64_t it = 0; it < iters; it++) {
                uint64_t p = it * blockDim.x + threadIdx.x;
                int disp = D_NONE;
                if (p < gsize) {
                    uint64_t myt = ft[p];
                    uint16_t org = fo[p];
                    int q = org >> GRP_ORG_SHIFT;
                    uint64_t i = s_a0[q] + (org & ((1u << GRP_ORG_SHIFT) - 1));
                    int shadow = 0;
                    if (p > 0) {
                        shadow = (ft[p - 1] == myt);
                    } else {
                        /* equal-key set split at the group boundary: a newer
                         * version ended the previous group's segment */
                        uint32_t bm = s_bmask;
                        for (int q2 = q + 1; q2 < R; q2++)
                            if (((bm >> q2) & 1) && s_btail[q2] == myt) {
                                shadow = 1;
                                break;
                            }
                    }
                    uint64_t rank = s_base + p;
                    const uint64_t *sm = nullptr;
                    if (MODE == 2 && s_allmeta)
                        sm = &s_meta[s_segoff[q] + (org & ((1u << GRP_ORG_SHIFT) - 1))];
                    else if (MODE == 0 && s_allmeta)
                        sm = &s_mt[s_segoff[q] + (org & ((1u << GRP_ORG_SHIFT) - 1))];
                    disp = grp_epilogue<MODE>(runs, q, i, shadow, rank, cp, sp, order, keepw,
                                              changed, new_expire, ksz, vsz, shadowed, sm);
                }
                if (MODE != 1)
                    tally.add(disp);
            }
        } else {
            /* oversized group (duplicate-clustered data): per-element global
             * tail probes inside the same disjoint segment windows */
            uint64_t iters = (gsize + blockDim.x - 1) / blockDim.x;
            for (uint64_t it = 0; it < iters; it++) {
                uint64_t e = it * blockDim.x + threadIdx.x;
                int disp = D_NONE;
                if (e < gsize) {
                    int q = 0;
                    while (s_segoff[q + 1] <= e)
                        q++;
                    uint64_t segpos = e - s_segoff[q];
                    uint64_t i = s_a0[q] + segpos;
                    uint64_t myt = runs[q].tails[i];
                    uint64_t lrank = segpos;
                    int shadow = 0;
                    uint32_t bm = s_bmask;
                    for (int p = 0; p < R; p++) {
                        if (p == q)
                            continue;
                        const uint64_t *tp = runs[p].tails;
                        uint64_t a0 = s_a0[p];
                        uint64_t l = a0, h = a0 + s_seglen[p];
                        if (p > q) {
                            while (l < h) {
                                uint64_t mid = (l + h) >> 1;
                                if (tp[mid] <= myt)
                                    l = mid + 1;
                                else
                                    h = mid;
                            }
                        } else {
                            while (l < h) {
                                uint64_t mid = (l + h) >> 1;
                                if (tp[mid] < myt)
                                    l = mid + 1;
                                else
                                    h = mid;
                            }
                        }
                        uint64_t b = l - a0;
                        if (p > q && !shadow && b > 0 && tp[a0 + b - 1] == myt)
                            shadow = 1;
                        lrank += b;
                        if (p > q && !shadow && ((bm >> p) & 1) && s_btail[p] == myt)
                            shadow = 1;
                    }
                    uint64_t rank = s_base + lrank;
                    disp = grp_epilogue<MODE>(runs, q, i, shadow, rank, cp, sp, order, keepw,
                                              changed, new_expire, ksz, vsz, shadowed, nullptr);
                }
                if (MODE != 1)
                    tally.add(disp);
            }
        }
        __syncthreads(); /* LDS reused by the next group */
    }
    if (MODE != 1)
        tally.flush(stats);
}

void launch_rank_grp_compact(const DevRun *d_runs, int R, const uint64_t *d_lo,
                             const uint64_t *d_anch, uint64_t n_groups, const CompactParams &cp,
                             uint64_t *d_order, uint64_t *d_keepw, uint8_t *d_changed,
                             uint32_t *d_new_expire, uint64_t *d_ksz, uint64_t *d_vsz,
                             CompactStatsDev *d_stats, int block_cap, hipStream_t s)
{
    uint64_t blocks = n_groups;
    if (blocks == 0)
        blocks = 1;
    if (blocks > (uint64_t)block_cap)
        blocks = (uint64_t)block_cap;
    ScanParams sp{};
    k_rank_grp<0><<<dim3((uint32_t)blocks), dim3(GRP_BLOCK), 0, s>>>(
        d_runs, R, d_lo, d_anch, n_groups, cp, sp, d_order, d_keepw, d_changed, d_new_expire,
        d_ksz, d_vsz, nullptr, nullptr, d_stats);
}

void launch_rank_grp_view(const DevRun *d_runs, int R, const uint64_t *d_lo,
                          const uint64_t *d_anch, uint64_t n_groups, uint64_t *d_order,
                          uint8_t *d_shadow, hipStream_t s)
{
    /* grid-stride with a capped grid: one-block-per-group paid measurable
     * workgroup setup/LDS churn (rank 0.48 -> 0.37 ms at 3.5M records) */
    uint64_t blocks = n_groups;
    if (blocks == 0)
        blocks = 1;
    if (blocks > 3584)
        blocks = 3584;
    CompactParams cp{};
    ScanParams sp{};
    k_rank_grp<1><<<dim3((uint32_t)blocks), dim3(GRP_BLOCK), 0, s>>>(
        d_runs, R, d_lo, d_anch, n_groups, cp, sp, d_order, nullptr, nullptr, nullptr, nullptr,
        nullptr, nullptr, d_shadow, nullptr);
}

/* single-run count: no merge, no shadow — a grid-stride pass of
 * dev_count_ok over the window (the common post-compaction state) */
__global__ void k_count_single(const DevRun *runs, const uint64_t *lo, const uint64_t *hi,
                               ScanParams sp, CompactStatsDev *stats)
{
    GrpTally tally;
    uint64_t l = lo[0], h = hi[0];
    for (uint64_t i = l + blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < h;
         i += gridDim.x * (uint64_t)blockDim.x)
        tally.add(dev_count_ok(runs[0], i, sp) ? D_KEEP : D_NONE);
    tally.flush(stats);
}

void launch_count_single(const DevRun *d_runs, const uint64_t *d_lo, const uint64_t *d_hi,
                         const ScanParams &sp, CompactStatsDev *d_stats, uint64_t n_max,
                         hipStream_t s)
{
    k_count_single<<<grid_for(n_max, BLOCK), BLOCK, 0, s>>>(d_runs, d_lo, d_hi, sp, d_stats);
}

/* fused count scan: shadow + countability evaluated in-kernel, the count
 * lands in stats->output_records (banked); zero intermediate arrays */
void launch_rank_grp_count(const DevRun *d_runs, int R, const uint64_t *d_lo,
                           const uint64_t *d_anch, uint64_t n_groups, const ScanParams &sp,
                           CompactStatsDev *d_stats, hipStream_t s)
{
    uint64_t blocks = n_groups;
    if (blocks == 0)
        blocks = 1;
    if (blocks > 3584)
        blocks = 3584;
    CompactParams cp{};
    k_rank_grp<2><<<dim3((uint32_t)blocks), dim3(GRP_BLOCK), 0, s>>>(
        d_runs, R, d_lo, d_anch, n_groups, cp, sp, nullptr, nullptr, nullptr, nullptr, nullptr,
        nullptr, nullptr, nullptr, d_stats);
}

void launch_rank_compact_ldst(const DevRun *d_runs, int R, const uint64_t *d_lo,
                              const uint64_t *d_hi, const uint64_t *d_wprefix, uint64_t total,
                              const CompactParams &cp, uint64_t *d_order, uint64_t *d_keepw,
                              uint8_t *d_changed, uint32_t *d_new_expire, uint64_t *d_ksz,
                              uint64_t *d_vsz, uint64_t *d_rank_of, const uint64_t *d_bt_off,
                              const uint64_t *d_bt, int bt_shift, CompactStatsDev *d_stats,
                              hipStream_t s)
{
    uint64_t blocks = (total + BLOCK - 1) / BLOCK;
    if (blocks == 0)
        blocks = 1;
    k_rank_compact_ldst<<<dim3((uint32_t)blocks), dim3(BLOCK), 0, s>>>(
        d_runs, R, d_lo, d_hi, d_wprefix, total, cp, d_order, d_keepw, d_changed, d_new_expire,
        d_ksz, d_vsz, d_rank_of, d_bt_off, d_bt, bt_shift, d_stats);
}

/* emit the merged run (wave per record); values copied whole, expire header
 * patched in-place for changed records */
__global__ void k_emit_compact(const DevRun *runs, const uint64_t *order, uint64_t m,
                               const uint64_t *keepw, const uint8_t *changed,
                               const uint32_t *new_expire, const uint64_t *kpos,
                               const uint64_t *koffs, const uint64_t *voffs, uint32_t dv,
                               uint8_t *kout, uint8_t *vout, uint64_t *okoff, uint64_t *ovoff,
                               uint64_t *osk, uint64_t n_out)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    for (uint64_t p = wave; p < m; p += nwaves) {
        if (!keepw[p])
            continue;
        uint64_t o = kpos[p];
        uint64_t id = order[p];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        uint64_t kl, vl;
        const uint8_t *k = run_key(r, i, &kl);
        const uint8_t *v = run_val(r, i, &vl);
        wave_copy(kout + koffs[p], k, kl, lane);
        uint8_t *vdst = vout + voffs[p];
        wave_copy(vdst, v, vl, lane);
        __builtin_amdgcn_wave_barrier(); /* patch below must see the copy */
        if (lane == 0) {
            if (changed[p]) {
                uint32_t ts = new_expire[p];
                uint32_t off = (dv == 2) ? 1 : 0;
                vdst[off] = (uint8_t)(ts >> 24);
                vdst[off + 1] = (uint8_t)(ts >> 16);
                vdst[off + 2] = (uint8_t)(ts >> 8);
                vdst[off + 3] = (uint8_t)ts;
            }
            okoff[o] = koffs[p];
            ovoff[o] = voffs[p];
            osk[o] = r.sk[i];
            if (o == n_out - 1) {
                okoff[n_out] = koffs[p] + kl;
                ovoff[n_out] = voffs[p] + vl;
            }
        }
    }
}

/* chunked emit (the fast path, measured 2.8 TB/s on the probe vs 1.1 for
 * wave-per-record): one scatter pass builds output-row metadata, then pure
 * 16B-chunk-parallel copies with an offset binary search per chunk. */
/* fk/fv nonzero = all-fixed-stride mode: row offsets and source addresses
 * derive from the keep count, so the per-rank ksz/vsz arrays and two of the
 * three full-length prefix sums are never built */
__global__ void k_compact_gather_meta(const DevRun *runs, const uint64_t *order, uint64_t m,
                                      const uint64_t *keepw, const uint8_t *changed,
                                      const uint32_t *new_expire, const uint64_t *kpos,
                                      const uint64_t *koffs, const uint64_t *voffs,
                                      uint64_t *row_koff, uint64_t *row_voff, uint64_t *row_ksrc,
                                      uint64_t *row_vsrc, uint32_t *row_patch, uint64_t *osk,
                                      uint64_t n_out, uint64_t kbytes, uint64_t vbytes,
                                      uint64_t fk, uint64_t fv)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < m;
         p += gridDim.x * (uint64_t)blockDim.x) {
        if (!keepw[p])
            continue;
        uint64_t o = kpos[p];
        uint64_t id = order[p];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        row_koff[o] = fk ? o * fk : koffs[p];
        row_voff[o] = fk ? o * fv : voffs[p];
        row_ksrc[o] = (uint64_t)(fk ? r.keys + i * fk : r.keys + r.koff[i]);
        /* input offsets stay voff-addressed: tombstones make input value
         * strides irregular even when every PUT shares one length */
        row_vsrc[o] = (uint64_t)(r.vals + r.voff[i]);
        row_patch[o] = (changed && changed[p]) ? 3 : 0;
        osk[o] = r.sk[i];
        if (o == n_out - 1) {
            row_koff[n_out] = kbytes;
            row_voff[n_out] = vbytes;
        }
        /* stash the patched expire per row (reuse row_patch upper bits is not
         * enough; separate array written below) */
    }
}

/* per-row patched expire value (only meaningful when changed) */
__global__ void k_compact_gather_expire(const uint64_t *keepw, const uint64_t *kpos,
                                        const uint8_t *changed, const uint32_t *new_expire,
                                        uint64_t m, uint32_t *row_expire)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < m;
         p += gridDim.x * (uint64_t)blockDim.x) {
        if (!keepw[p])
            continue;
        row_expire[kpos[p]] = changed[p] ? new_expire[p] : 0xFFFFFFFFu;
    }
}

/* copy 16B chunks of the packed output; chunk -> row via binary search on the
 * row offset array (hot in L1/L2: consecutive chunks hit consecutive rows) */
/* every 64th chunk's (1 KB of output) starting row: narrows the per-chunk
 * row search from log2(n_rows) L2 probes to ~3 probes in an 8-row window */
__global__ void k_chunk_anchors(const uint64_t *row_off, uint64_t n_rows, uint64_t n_anchors,
                                uint64_t *anchors)
{
    for (uint64_t a = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; a < n_anchors;
         a += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t pos = a << 10; /* 64 chunks x 16B */
        uint64_t lo = 0, hi = n_rows;
        while (lo + 1 < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (row_off[mid] <= pos)
                lo = mid;
            else
                hi = mid;
        }
        anchors[a] = lo;
    }
}

/* fixed-stride copy: every row is `stride` bytes with stride % 16 == 0, so
 * chunk -> row is a division — no row-offset reads, no anchor search */
__global__ void k_copy_chunks_fixed(const uint64_t *row_src, uint64_t n_rows, uint64_t stride,
                                    uint8_t *dst)
{
    uint64_t cpr = stride >> 4; /* chunks per row */
    uint64_t n_chunks = n_rows * cpr;
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < n_chunks;
         t += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t row = t / cpr;
        uint64_t off = (t - row * cpr) << 4;
        const uint8_t *src = (const uint8_t *)row_src[row] + off;
        uint32_t w[4];
        __builtin_memcpy(w, src, 16);
        __builtin_memcpy(dst + row * stride + off, w, 16);
    }
}

/* fixed-stride copy for strides that are NOT 16B multiples (e.g. 18B
 * keys): one thread copies one whole row in u64 pieces — consecutive
 * threads write consecutive stride-length spans, which the coalescer folds
 * into contiguous segments (the 16B-chunk path degraded to byte loops at
 * every straddled row boundary) */
__global__ void k_copy_rows_fixed(const uint64_t *row_src, uint64_t n_rows, uint64_t stride,
                                  uint8_t *dst)
{
    for (uint64_t r = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; r < n_rows;
         r += gridDim.x * (uint64_t)blockDim.x) {
        const uint8_t *src = (const uint8_t *)row_src[r];
        uint8_t *d = dst + r * stride;
        uint64_t b = 0;
        for (; b + 8 <= stride; b += 8) {
            uint64_t w;
            __builtin_memcpy(&w, src + b, 8);
            __builtin_memcpy(d + b, &w, 8);
        }
        for (; b < stride; b++)
            d[b] = src[b];
    }
}

__global__ void k_copy_chunks(const uint64_t *row_off /* [n_rows+1] */,
                              const uint64_t *row_src /* device ptrs */, uint64_t n_rows,
                              uint64_t total_bytes, const uint64_t *anchors,
                              uint64_t n_anchors, uint8_t *dst)
{
    uint64_t n_chunks = (total_bytes + 15) >> 4;
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < n_chunks;
         t += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t pos = t << 4;
        uint64_t end = pos + 16 <= total_bytes ? pos + 16 : total_bytes;
        /* row = last row with row_off[row] <= pos */
        uint64_t ab = t >> 6;
        uint64_t lo = anchors[ab];
        uint64_t hi = (ab + 1 < n_anchors) ? anchors[ab + 1] + 1 : n_rows;
        while (lo + 1 < hi) {
            uint64_t mid = (lo + hi) >> 1;
            if (row_off[mid] <= pos)
                lo = mid;
            else
                hi = mid;
        }
        uint64_t row = lo;
        while (pos < end) {
            uint64_t row_end = row_off[row + 1];
            const uint8_t *s = (const uint8_t *)row_src[row] + (pos - row_off[row]);
            uint64_t nb = (end < row_end ? end : row_end) - pos;
            uint8_t *d = dst + pos;
            if (nb == 16) {
                uint32_t w[4];
                __builtin_memcpy(w, s, 16);
                __builtin_memcpy(d, w, 16);
            } else {
                for (uint64_t b = 0; b < nb; b++)
                    d[b] = s[b];
            }
            pos += nb;
            row++;
        }
    }
}

/* patch rewritten expire headers in the packed values */
__global__ void k_patch_expire(const uint64_t *row_voff, const uint32_t *row_expire,
                               uint64_t n_rows, uint32_t dv, uint8_t *vout)
{
    uint32_t off = (dv == 2) ? 1 : 0;
    for (uint64_t o = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; o < n_rows;
         o += gridDim.x * (uint64_t)blockDim.x) {
        uint32_t ts = row_expire[o];
        if (ts == 0xFFFFFFFFu)
            continue;
        uint8_t *v = vout + row_voff[o] + off;
        v[0] = (uint8_t)(ts >> 24);
        v[1] = (uint8_t)(ts >> 16);
        v[2] = (uint8_t)(ts >> 8);
        v[3] = (uint8_t)ts;
    }
}

/* input-major emit: waves sweep each run's records in storage order
 * (sequential coalesced reads); the write side follows the monotone output
 * offsets of the kept records.  A/B alternative to the rank-major emit. */
__global__ void k_emit_compact_inmajor(const DevRun *runs, int R, const uint64_t *wprefix,
                                       uint64_t total, const uint64_t *rank_of,
                                       const uint64_t *keepw, const uint8_t *changed,
                                       const uint32_t *new_expire, const uint64_t *kpos,
                                       const uint64_t *koffs, const uint64_t *voffs, uint32_t dv,
                                       uint8_t *kout, uint8_t *vout, uint64_t *okoff,
                                       uint64_t *ovoff, uint64_t *osk, uint64_t n_out)
{
    uint64_t ws, we;
    int lane;
    wave_chunk(total, &ws, &we, &lane);
    for (uint64_t t = ws; t < we; t++) {
        uint64_t rank = rank_of[t];
        if (!keepw[rank])
            continue;
        int r = 0;
        while (wprefix[r + 1] <= t)
            r++;
        uint64_t i = t - wprefix[r];
        const DevRun &run = runs[r];
        uint64_t kl, vl;
        const uint8_t *k = run_key(run, i, &kl);
        const uint8_t *v = run_val(run, i, &vl);
        uint64_t o = kpos[rank];
        wave_copy(kout + koffs[rank], k, kl, lane);
        uint8_t *vdst = vout + voffs[rank];
        wave_copy(vdst, v, vl, lane);
        __builtin_amdgcn_wave_barrier();
        if (lane == 0) {
            if (changed[rank]) {
                uint32_t ts = new_expire[rank];
                uint32_t off = (dv == 2) ? 1 : 0;
                vdst[off] = (uint8_t)(ts >> 24);
                vdst[off + 1] = (uint8_t)(ts >> 16);
                vdst[off + 2] = (uint8_t)(ts >> 8);
                vdst[off + 3] = (uint8_t)ts;
            }
            okoff[o] = koffs[rank];
            ovoff[o] = voffs[rank];
            osk[o] = run.sk[i];
            if (o == n_out - 1) {
                okoff[n_out] = koffs[rank] + kl;
                ovoff[n_out] = voffs[rank] + vl;
            }
        }
    }
}

/* ================= launch wrappers (host) ================= */
extern "C++" {

void launch_crc64_table_init(const uint64_t *host_table)
{
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(d_crc64_table), host_table, 256 * sizeof(uint64_t)));
}

void launch_bounds(const DevRun *d_runs, int R, const uint8_t *d_key, uint64_t klen,
                   uint64_t *d_out, int upper, hipStream_t s)
{
    k_bounds<<<(R + 63) / 64, 64, 0, s>>>(d_runs, R, d_key, klen, d_out, upper);
}

void launch_rank(const DevRun *d_runs, int R, const uint64_t *d_lo, const uint64_t *d_hi,
                 const uint64_t *d_wprefix, uint64_t total, uint64_t *d_order,
                 uint8_t *d_shadowed, const uint64_t *d_bt_off, const uint64_t *d_bt,
                 int bt_shift, hipStream_t s)
{
    k_rank<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(d_runs, R, d_lo, d_hi, d_wprefix, total,
                                                    d_order, d_shadowed, d_bt_off, d_bt,
                                                    bt_shift);
}

void launch_bound_table(const DevRun *d_runs, int R, const uint64_t *d_lo, const uint64_t *d_hi,
                        const uint64_t *d_bt_off, uint64_t total_rows, int bt_shift,
                        uint64_t *d_bt, const uint64_t *d_cbt_off, const uint64_t *d_cbt,
                        int cbt_shift, hipStream_t s)
{
    k_bound_table<<<grid_for(total_rows, BLOCK), BLOCK, 0, s>>>(d_runs, R, d_lo, d_hi, d_bt_off,
                                                                total_rows, bt_shift, d_bt,
                                                                d_cbt_off, d_cbt, cbt_shift);
}

void launch_visible(const DevRun *d_runs, const uint64_t *d_order, const uint8_t *d_shadowed,
                    uint64_t m, uint64_t *d_flags, hipStream_t s)
{
    k_visible<<<grid_for(m, BLOCK), BLOCK, 0, s>>>(d_runs, d_order, d_shadowed, m, d_flags);
}

void launch_gather(const uint64_t *d_order, const uint64_t *d_flags, const uint64_t *d_pos,
                   uint64_t m, uint64_t *d_out, hipStream_t s)
{
    k_gather<<<grid_for(m, BLOCK), BLOCK, 0, s>>>(d_order, d_flags, d_pos, m, d_out);
}

/* exclusive scan; caller provides scratch of psum_scratch_elems(n) u64s
 * (engine-owned arena — no per-call allocation) */
uint64_t psum_scratch_elems(uint64_t n)
{
    uint64_t tot = 0;
    while (n > 1) {
        uint64_t nb = (n + PSUM_BLOCK_ITEMS - 1) / PSUM_BLOCK_ITEMS;
        tot += 2 * nb;
        n = nb;
    }
    return tot + 4;
}

static void psum_rec(const uint64_t *d_in, uint64_t *d_out, uint64_t n, uint64_t *scratch,
                     hipStream_t s)
{
    uint64_t nb = (n + PSUM_BLOCK_ITEMS - 1) / PSUM_BLOCK_ITEMS;
    if (nb == 0)
        nb = 1;
    uint64_t *d_bs = scratch;
    k_psum1<<<(uint32_t)nb, BLOCK, 0, s>>>(d_in, d_out, d_bs, n);
    if (nb > 1) {
        uint64_t *d_bo = scratch + nb;
        psum_rec(d_bs, d_bo, nb, scratch + 2 * nb, s);
        k_psum_add<<<(uint32_t)nb, BLOCK, 0, s>>>(d_out, d_bo, n);
    }
}

void launch_psum(const uint64_t *d_in, uint64_t *d_out, uint64_t n, uint64_t *d_scratch,
                 hipStream_t s)
{
    psum_rec(d_in, d_out, n, d_scratch, s);
}

void launch_get(const DevRun *d_runs, int R, const uint8_t *d_qkeys, const uint64_t *d_qoffs,
                uint64_t nq, uint32_t epoch_now, uint32_t dv, int32_t *d_status, uint64_t *d_hit,
                uint64_t *d_ulen, uint32_t *d_expire, hipStream_t s)
{
    k_get<<<grid_for(nq, BLOCK), BLOCK, 0, s>>>(d_runs, R, d_qkeys, d_qoffs, nq, epoch_now, dv,
                                                d_status, d_hit, d_ulen, d_expire);
}

void launch_emit_values(const DevRun *d_runs, const uint64_t *d_hit, const int32_t *d_status,
                        uint64_t nq, uint32_t dv, const uint64_t *d_voffs, uint8_t *d_vout,
                        hipStream_t s)
{
    k_emit_values<<<grid_for(nq * WAVE, BLOCK), BLOCK, 0, s>>>(d_runs, d_hit, d_status, nq, dv,
                                                               d_voffs, d_vout);
}

void launch_scan_state(const DevRun *d_runs, const uint64_t *d_view, uint64_t w,
                       const ScanParams &sp, uint8_t *d_state, uint64_t *d_ksz, uint64_t *d_vsz,
                       hipStream_t s)
{
    k_scan_state<<<grid_for(w, BLOCK), BLOCK, 0, s>>>(d_runs, d_view, w, sp, d_state, d_ksz,
                                                      d_vsz);
}

void launch_normal_flags(const uint8_t *d_state, uint64_t w, uint64_t *d_flags, hipStream_t s)
{
    k_normal_flags<<<grid_for(w, BLOCK), BLOCK, 0, s>>>(d_state, w, d_flags);
}

void launch_cutoff(const uint64_t *d_nprefix, const uint64_t *d_flags, uint64_t w,
                   uint64_t batch_count, uint64_t *d_out2, hipStream_t s)
{
    k_cutoff<<<1, 1, 0, s>>>(d_nprefix, d_flags, w, batch_count, d_out2);
}

void launch_cut_sizes(const uint8_t *d_state, uint64_t w, uint64_t consumed, uint64_t *d_ksz,
                      uint64_t *d_vsz, hipStream_t s)
{
    k_cut_sizes<<<grid_for(w, BLOCK), BLOCK, 0, s>>>(d_state, w, consumed, d_ksz, d_vsz);
}

void launch_emit_scan(const DevRun *d_runs, const uint64_t *d_view, uint64_t w,
                      const uint8_t *d_state, uint64_t consumed, const uint64_t *d_npos,
                      const uint64_t *d_koffs, const uint64_t *d_voffs, const ScanParams &sp,
                      uint8_t *d_kout, uint8_t *d_vout, uint64_t *d_kout_offs,
                      uint64_t *d_vout_offs, int32_t *d_ets, uint64_t n_out, hipStream_t s)
{
    k_emit_scan<<<grid_for(consumed * WAVE, BLOCK), BLOCK, 0, s>>>(
        d_runs, d_view, w, d_state, consumed, d_npos, d_koffs, d_voffs, sp, d_kout, d_vout,
        d_kout_offs, d_vout_offs, d_ets, n_out);
}

void launch_emit_rows(const DevRun *d_runs, const uint64_t *d_view, const uint64_t *d_rows,
                      uint64_t n_rows, const uint64_t *d_koffs, const uint64_t *d_voffs,
                      const ScanParams &sp, uint8_t *d_kout, uint8_t *d_vout, hipStream_t s)
{
    k_emit_rows<<<grid_for(n_rows * WAVE, BLOCK), BLOCK, 0, s>>>(d_runs, d_view, d_rows, n_rows,
                                                                 d_koffs, d_voffs, sp, d_kout,
                                                                 d_vout);
}

void launch_first_eq(const DevRun *d_runs, const uint64_t *d_view, uint64_t n,
                     const uint8_t *d_key, uint64_t klen, uint32_t *d_out, hipStream_t s)
{
    k_first_eq<<<1, 1, 0, s>>>(d_runs, d_view, n, d_key, klen, d_out);
}

void launch_rank_compact(const DevRun *d_runs, int R, const uint64_t *d_lo, const uint64_t *d_hi,
                         const uint64_t *d_wprefix, uint64_t total, const CompactParams &cp,
                         uint64_t *d_order, uint64_t *d_keepw, uint8_t *d_changed,
                         uint32_t *d_new_expire, uint64_t *d_ksz, uint64_t *d_vsz,
                         uint64_t *d_rank_of, const uint64_t *d_bt_off, const uint64_t *d_bt,
                         int bt_shift, CompactStatsDev *d_stats, hipStream_t s)
{
    k_rank_compact<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(d_runs, R, d_lo, d_hi, d_wprefix,
                                                            total, cp, d_order, d_keepw,
                                                            d_changed, d_new_expire, d_ksz,
                                                            d_vsz, d_rank_of, d_bt_off, d_bt,
                                                            bt_shift, d_stats);
}

void launch_emit_compact_chunked(const DevRun *d_runs, const uint64_t *d_order, uint64_t m,
                                 const uint64_t *d_keepw, const uint8_t *d_changed,
                                 const uint32_t *d_new_expire, const uint64_t *d_kpos,
                                 const uint64_t *d_koffs, const uint64_t *d_voffs, uint32_t dv,
                                 uint64_t n_out, uint64_t kbytes, uint64_t vbytes,
                                 uint64_t *d_row_ksrc, uint64_t *d_row_vsrc,
                                 uint32_t *d_row_patch, uint32_t *d_row_expire, uint8_t *d_kout,
                                 uint8_t *d_vout, uint64_t *d_okoff /* also the key row offsets */,
                                 uint64_t *d_ovoff /* also the value row offsets */,
                                 uint64_t *d_osk, uint64_t *d_kanchor, uint64_t *d_vanchor,
                                 uint64_t fk, uint64_t fv, hipStream_t s)
{
    k_compact_gather_meta<<<grid_for(m, BLOCK), BLOCK, 0, s>>>(
        d_runs, d_order, m, d_keepw, d_changed, d_new_expire, d_kpos, d_koffs, d_voffs, d_okoff,
        d_ovoff, d_row_ksrc, d_row_vsrc, d_row_patch, d_osk, n_out, kbytes, vbytes, fk, fv);
    if (d_changed)
        k_compact_gather_expire<<<grid_for(m, BLOCK), BLOCK, 0, s>>>(
            d_keepw, d_kpos, d_changed, d_new_expire, m, d_row_expire);
    bool kfix = fk > 0 && (fk & 15) == 0, vfix = fk > 0 && (fv & 15) == 0;
    if (kfix) {
        k_copy_chunks_fixed<<<grid_for((kbytes + 15) >> 4, BLOCK), BLOCK, 0, s>>>(
            d_row_ksrc, n_out, fk, d_kout);
    } else if (fk > 0) {
        k_copy_rows_fixed<<<grid_for(n_out, BLOCK), BLOCK, 0, s>>>(d_row_ksrc, n_out, fk,
                                                                   d_kout);
    } else {
        uint64_t kanch = ((kbytes + 15) >> 4 >> 6) + 1;
        k_chunk_anchors<<<grid_for(kanch, BLOCK), BLOCK, 0, s>>>(d_okoff, n_out, kanch,
                                                                d_kanchor);
        k_copy_chunks<<<grid_for((kbytes + 15) >> 4, BLOCK), BLOCK, 0, s>>>(
            d_okoff, d_row_ksrc, n_out, kbytes, d_kanchor, kanch, d_kout);
    }
    if (vfix) {
        k_copy_chunks_fixed<<<grid_for((vbytes + 15) >> 4, BLOCK), BLOCK, 0, s>>>(
            d_row_vsrc, n_out, fv, d_vout);
    } else if (fk > 0) {
        k_copy_rows_fixed<<<grid_for(n_out, BLOCK), BLOCK, 0, s>>>(d_row_vsrc, n_out, fv,
                                                                   d_vout);
    } else {
        uint64_t vanch = ((vbytes + 15) >> 4 >> 6) + 1;
        k_chunk_anchors<<<grid_for(vanch, BLOCK), BLOCK, 0, s>>>(d_ovoff, n_out, vanch,
                                                                d_vanchor);
        k_copy_chunks<<<grid_for((vbytes + 15) >> 4, BLOCK), BLOCK, 0, s>>>(
            d_ovoff, d_row_vsrc, n_out, vbytes, d_vanchor, vanch, d_vout);
    }
    if (d_changed)
        k_patch_expire<<<grid_for(n_out, BLOCK), BLOCK, 0, s>>>(d_ovoff, d_row_expire, n_out,
                                                                dv, d_vout);
}

void launch_emit_compact_inmajor(const DevRun *d_runs, int R, const uint64_t *d_wprefix,
                                 uint64_t total, const uint64_t *d_rank_of,
                                 const uint64_t *d_keepw, const uint8_t *d_changed,
                                 const uint32_t *d_new_expire, const uint64_t *d_kpos,
                                 const uint64_t *d_koffs, const uint64_t *d_voffs, uint32_t dv,
                                 uint8_t *d_kout, uint8_t *d_vout, uint64_t *d_okoff,
                                 uint64_t *d_ovoff, uint64_t *d_osk, uint64_t n_out,
                                 hipStream_t s)
{
    k_emit_compact_inmajor<<<grid_for(total * WAVE, BLOCK), BLOCK, 0, s>>>(
        d_runs, R, d_wprefix, total, d_rank_of, d_keepw, d_changed, d_new_expire, d_kpos,
        d_koffs, d_voffs, dv, d_kout, d_vout, d_okoff, d_ovoff, d_osk, n_out);
}

void launch_emit_compact(const DevRun *d_runs, const uint64_t *d_order, uint64_t m,
                         const uint64_t *d_keepw, const uint8_t *d_changed,
                         const uint32_t *d_new_expire, const uint64_t *d_kpos,
                         const uint64_t *d_koffs, const uint64_t *d_voffs, uint32_t dv,
                         uint8_t *d_kout, uint8_t *d_vout, uint64_t *d_okoff, uint64_t *d_ovoff,
                         uint64_t *d_osk, uint64_t n_out, hipStream_t s)
{
    k_emit_compact<<<grid_for(m * WAVE, BLOCK), BLOCK, 0, s>>>(
        d_runs, d_order, m, d_keepw, d_changed, d_new_expire, d_kpos, d_koffs, d_voffs, dv,
        d_kout, d_vout, d_okoff, d_ovoff, d_osk, n_out);
}

} /* extern C++ */

/* ================= fused small multi_get =================
 * One launch for the whole on_multi_get range path when the hashkey's row
 * count is small (the common YCSB-E case): per-run bounds, in-window rank
 * merge, TTL/sortkey-filter, limiter caps (count/iteration/size,
 * forward/reverse, first-exclusive) and output packing — one block, one
 * result blob, no intermediate syncs.  Semantics mirror
 * on_multi_get:540-778 exactly; ranges larger than the scratch budget set
 * out_hdr[0]=-1 and the host falls back to the general path. */
/* cooperative lower_bound: MG_BND_L consecutive lanes probe that many split
 * points per round, so the cold-HBM pointer chase of a serving-path binary
 * search becomes L overlapped loads (5 rounds instead of ~22 dependent
 * probes at 3.5M-record runs).  All L lanes return the same result. */
#define MG_BND_L 16
__device__ static uint64_t coop_lower_bound(const DevRun &r, const uint8_t *key, uint64_t klen,
                                            int j, int sub)
{
    uint64_t lo = 0, hi = r.n;
    while (hi - lo > MG_BND_L) {
        uint64_t step = (hi - lo) / (MG_BND_L + 1);
        uint64_t pos = lo + step * (uint64_t)(j + 1);
        uint64_t kl;
        const uint8_t *k = run_key(r, pos, &kl);
        int ge = dev_key_cmp(k, kl, key, klen) >= 0;
        uint64_t b = __ballot(ge);
        uint32_t sub_bits = (uint32_t)((b >> (sub * MG_BND_L)) & 0xFFFFu);
        if (sub_bits == 0) {
            lo = lo + step * MG_BND_L + 1;
        } else {
            int f = __ffs(sub_bits) - 1; /* first probe with key >= target */
            uint64_t newhi = lo + step * (uint64_t)(f + 1);
            if (f > 0)
                lo = lo + step * (uint64_t)f + 1;
            hi = newhi;
        }
    }
    uint64_t res = hi;
    int ge2 = 0;
    uint64_t pos2 = lo + (uint64_t)j;
    if (pos2 < hi) {
        uint64_t kl;
        const uint8_t *k = run_key(r, pos2, &kl);
        ge2 = dev_key_cmp(k, kl, key, klen) >= 0;
    }
    uint64_t b2 = __ballot(ge2);
    uint32_t sb = (uint32_t)((b2 >> (sub * MG_BND_L)) & 0xFFFFu);
    if (sb)
        res = lo + (uint64_t)(__ffs(sb) - 1);
    return res;
}

__device__ static void mg_core(const DevRun *runs, int R, const MgFusedArgs &a,
                               const uint8_t *mg_start, uint64_t mg_start_len,
                               const uint8_t *mg_stop, uint64_t mg_stop_len,
                               uint64_t hash_key_skip, int64_t *out_hdr, uint8_t *out_blob,
                               uint64_t blob_cap)
{
    __shared__ uint64_t s_lo[RRDB_MAX_RUNS], s_hi[RRDB_MAX_RUNS], s_wp[RRDB_MAX_RUNS + 1];
    __shared__ uint64_t s_total;
    __shared__ uint32_t s_state[MG_MAX_ROWS];   /* 0 normal,1 skip,2 tomb/shadow */
    __shared__ uint16_t s_klen[MG_MAX_ROWS];
    __shared__ uint16_t s_vlen_lo[MG_MAX_ROWS]; /* value len (<=64KB supported in fused path) */
    __shared__ uint64_t s_id[MG_MAX_ROWS];      /* (run<<40)|idx by rank */
    int tid = threadIdx.x;
    /* phase 1: cooperative bounds; the hashkey-prefix bloom skips runs that
     * cannot contain this hashkey (every in-range key shares the
     * [u16 len][hashkey] prefix = the first hash_key_skip bytes of start) */
    __shared__ uint64_t s_pfxh;
    if (tid == 0)
        s_pfxh = bloom_hash(mg_start, hash_key_skip <= mg_start_len ? hash_key_skip
                                                                    : mg_start_len);
    __syncthreads();
    {
        int j = tid % MG_BND_L;
        int sub = (tid % WAVE) / MG_BND_L;
        int per_block = blockDim.x / MG_BND_L;
        for (int sidx = tid / MG_BND_L; sidx < 2 * R; sidx += per_block) {
            int q = sidx >> 1;
            int is_stop = sidx & 1;
            const DevRun &r = runs[q];
            if (!pfx_bloom_maybe_has(r, s_pfxh)) {
                if (j == 0) {
                    if (is_stop)
                        s_hi[q] = 0;
                    else
                        s_lo[q] = 0;
                }
                continue;
            }
            uint64_t res = coop_lower_bound(r, is_stop ? mg_stop : mg_start,
                                            is_stop ? mg_stop_len : mg_start_len, j, sub);
            if (j == 0) {
                if (is_stop)
                    s_hi[q] = res;
                else
                    s_lo[q] = res;
            }
        }
    }
    __syncthreads();
    if (tid < R && s_hi[tid] < s_lo[tid])
        s_hi[tid] = s_lo[tid];
    __syncthreads();
    if (tid == 0) {
        uint64_t t = 0;
        for (int r = 0; r < R; r++) {
            s_wp[r] = t;
            t += s_hi[r] - s_lo[r];
        }
        s_wp[R] = t;
        s_total = t;
        if (t > MG_MAX_ROWS)
            out_hdr[0] = -1; /* fallback */
    }
    __syncthreads();
    uint64_t total = s_total;
    if (total > MG_MAX_ROWS)
        return;
    /* phase 2: rank + visibility + per-row state/sizes (merged order in LDS) */
    for (uint64_t t = tid; t < total; t += blockDim.x) {
        int r = 0;
        while (s_wp[r + 1] <= t)
            r++;
        uint64_t i = s_lo[r] + (t - s_wp[r]);
        uint64_t kl;
        const uint8_t *k = run_key(runs[r], i, &kl);
        uint64_t rank = i - s_lo[r];
        int shadow = 0;
        for (int q = 0; q < R; q++) {
            if (q == r)
                continue;
            if (q > r) {
                uint64_t ub = dev_upper_bound(runs[q], k, kl, s_lo[q], s_hi[q]);
                if (!shadow && ub > s_lo[q]) {
                    uint64_t pl;
                    const uint8_t *pk = run_key(runs[q], ub - 1, &pl);
                    if (dev_key_cmp(pk, pl, k, kl) == 0)
                        shadow = 1;
                }
                rank += ub - s_lo[q];
            } else {
                rank += dev_lower_bound(runs[q], k, kl, s_lo[q], s_hi[q]) - s_lo[q];
            }
        }
        uint32_t st;
        uint16_t klen_out = 0, vlen_out = 0;
        if (shadow || (runs[r].sk[i] & 1)) {
            st = 2; /* invisible: not iterated at all */
        } else {
            uint64_t vl;
            const uint8_t *v = run_val(runs[r], i, &vl);
            uint32_t hdr = dev_hdr_len(a.data_version);
            if (dev_ts_expired(a.epoch_now, dev_expire_ts(a.data_version, v))) {
                st = 1; /* kExpired: iterated but skipped */
            } else {
                const uint8_t *skp = k + hash_key_skip;
                uint64_t sklen = kl - hash_key_skip;
                if (a.sk_ft != 0 &&
                    !dev_validate_filter(a.sk_ft, a.sk_pat, a.sk_pat_len, skp, sklen)) {
                    st = 1; /* kFiltered */
                } else if (kl - hash_key_skip > 0xFFFF || vl - hdr > 0xFFFF) {
                    st = 3; /* oversize for fused path -> fallback */
                } else {
                    st = 0;
                    klen_out = (uint16_t)(kl - hash_key_skip);
                    vlen_out = a.no_value ? 0 : (uint16_t)(vl - hdr);
                }
            }
        }
        /* indexed by RANK: the limiter walk below runs in merged order */
        s_state[rank] = st;
        s_klen[rank] = klen_out;
        s_vlen_lo[rank] = vlen_out;
        s_id[rank] = ((uint64_t)r << 40) | i;
    }
    __syncthreads();
    /* phase 3 (thread 0): limiter walk, exactly on_multi_get:616-778 */
    __shared__ uint64_t s_sel[MG_MAX_ROWS]; /* selected merged positions */
    __shared__ uint64_t s_nsel, s_kbytes;
    if (tid == 0) {
        int fallback = 0;
        uint64_t nsel = 0, kb = 0, vb = 0;
        uint64_t count = 0, iteration = 0;
        int64_t size = 0;
        int complete = 0;
        /* reverse + start-exclusive: records == start are OUT OF RANGE for
         * the reverse walk (on_multi_get:697-700 — c==0 && !start_inclusive
         * -> complete); they sit at the lowest merged positions.  oor_exists
         * notes whether the iterator would land on one (it->Valid()). */
        uint64_t t_floor = 0;
        int oor_exists = 0;
        if (a.reverse && !a.start_inclusive) {
            for (uint64_t t = 0; t < total; t++) {
                uint64_t id = s_id[t];
                const DevRun &r = runs[id >> 40];
                uint64_t i = id & 0xFFFFFFFFFFull, kl;
                const uint8_t *k = run_key(r, i, &kl);
                if (dev_key_cmp(k, kl, mg_start, mg_start_len) != 0)
                    break;
                if (s_state[t] != 2)
                    oor_exists = 1; /* iterator-visible == start */
                t_floor = t + 1;
            }
        }
        uint64_t countable = 0;
        /* countable = entries the reference iterator would visit (visible) */
        for (uint64_t t = t_floor; t < total; t++)
            if (s_state[t] != 2 && s_state[t] != 3)
                countable++;
            else if (s_state[t] == 3)
                fallback = 1;
        int skipped_first = 0;
        uint64_t visited = 0;
        uint64_t span = total - t_floor;
        for (uint64_t s = 0; s < span && !fallback; s++) {
            uint64_t t = a.reverse ? (total - 1 - s) : t_floor + s;
            if (s_state[t] == 2)
                continue; /* invisible to the iterator */
            if (count >= a.max_kv_count || iteration >= a.max_iteration_count ||
                size >= a.max_iteration_size)
                break;
            if (visited == 0) {
                /* first-exclusive boundary skip (:636-643 / :700-707) */
                int check = (!a.reverse && !a.start_inclusive) ||
                            (a.reverse && !a.stop_inclusive);
                if (check) {
                    uint64_t id = s_id[t];
                    const DevRun &r = runs[id >> 40];
                    uint64_t i = id & 0xFFFFFFFFFFull, kl;
                    const uint8_t *k = run_key(r, i, &kl);
                    const uint8_t *bound = a.reverse ? mg_stop : mg_start;
                    uint64_t blen = a.reverse ? mg_stop_len : mg_start_len;
                    visited = 1;
                    if (dev_key_cmp(k, kl, bound, blen) == 0) {
                        skipped_first = 1;
                        continue;
                    }
                } else {
                    visited = 1;
                }
            }
            iteration++;
            if (s_state[t] == 1)
                continue;
            s_sel[nsel++] = t;
            count++;
            kb += s_klen[t];
            vb += s_vlen_lo[t];
            size += (int64_t)s_klen[t] + (int64_t)s_vlen_lo[t];
        }
        /* completion (on_multi_get:777-788): kIncomplete iff the iterator is
         * still Valid() after a limit exit — even when the remaining records
         * lie past the range.  Exception: the post-append c==0
         * break (:668-672/:739-744) completes when the LAST iterated record
         * equals the inclusive far boundary, regardless of limits. */
        {
            int consumed_all = (iteration >= countable - (uint64_t)skipped_first);
            int boundary_hit = 0;
            if (consumed_all && iteration > 0 &&
                ((!a.reverse && a.stop_inclusive) || (a.reverse && a.start_inclusive))) {
                /* far-end iterated record: forward = highest non-invisible t,
                 * reverse = lowest at/above t_floor */
                int64_t t_far = -1;
                if (!a.reverse) {
                    for (int64_t t = (int64_t)total - 1; t >= (int64_t)t_floor; t--)
                        if (s_state[t] != 2) {
                            t_far = t;
                            break;
                        }
                } else {
                    for (uint64_t t = t_floor; t < total; t++)
                        if (s_state[t] != 2) {
                            t_far = (int64_t)t;
                            break;
                        }
                }
                if (t_far >= 0) {
                    uint64_t id = s_id[t_far];
                    const DevRun &r = runs[id >> 40];
                    uint64_t i = id & 0xFFFFFFFFFFull, kl;
                    const uint8_t *k = run_key(r, i, &kl);
                    /* original stop = mg_stop minus the '\0' the host
                     * appended for the inclusive exclusive-bound */
                    const uint8_t *bnd = a.reverse ? mg_start : mg_stop;
                    uint64_t blen = a.reverse ? mg_start_len : mg_stop_len - 1;
                    boundary_hit = dev_key_cmp(k, kl, bnd, blen) == 0;
                }
            }
            int limit_exit = !boundary_hit &&
                             (count >= a.max_kv_count || iteration >= a.max_iteration_count ||
                              size >= a.max_iteration_size);
            if (!consumed_all)
                complete = 0; /* stopped mid-range */
            else if (!limit_exit)
                complete = 1; /* walked past the range end */
            else if (a.reverse)
                complete = !(oor_exists ||
                             dev_valid_beyond(runs, R, mg_start, mg_start_len, 1));
            else
                complete = !dev_valid_beyond(runs, R, mg_stop, mg_stop_len, 0);
        }
        if (2 * (nsel + 1) * 8 + kb + vb > blob_cap)
            fallback = 1;
        if (fallback) {
            out_hdr[0] = -1;
            s_nsel = (uint64_t)-1;
        } else {
            /* ascending output (reverse selected from the top) */
            if (a.reverse) {
                for (uint64_t x = 0; x < nsel / 2; x++) {
                    uint64_t tmp = s_sel[x];
                    s_sel[x] = s_sel[nsel - 1 - x];
                    s_sel[nsel - 1 - x] = tmp;
                }
            }
            /* blob layout: [koff (nsel+1)*8][voff (nsel+1)*8][keys][vals] */
            uint64_t *okoff = (uint64_t *)out_blob;
            uint64_t *ovoff = okoff + (nsel + 1);
            uint64_t ko = 0, vo = 0;
            for (uint64_t x = 0; x < nsel; x++) {
                okoff[x] = ko;
                ovoff[x] = vo;
                ko += s_klen[s_sel[x]];
                vo += s_vlen_lo[s_sel[x]];
            }
            okoff[nsel] = ko;
            ovoff[nsel] = vo;
            out_hdr[0] = (int64_t)nsel;
            out_hdr[1] = complete;
            out_hdr[2] = (int64_t)ko;
            out_hdr[3] = (int64_t)vo;
            s_nsel = nsel;
            s_kbytes = ko;
        }
    }
    __syncthreads();
    if (s_nsel == (uint64_t)-1)
        return;
    /* phase 4: copy selected rows (wave per row) into the blob */
    uint64_t nsel = s_nsel;
    const uint64_t *okoff = (const uint64_t *)out_blob;
    const uint64_t *ovoff = okoff + (nsel + 1);
    uint8_t *keys_out = out_blob + 2 * (nsel + 1) * 8;
    uint8_t *vals_out = keys_out + s_kbytes;
    int lane = tid % WAVE;
    int wid = tid / WAVE;
    int nw = blockDim.x / WAVE;
    uint32_t hdr = dev_hdr_len(a.data_version);
    for (uint64_t x = wid; x < nsel; x += nw) {
        uint64_t t = s_sel[x];
        uint64_t id = s_id[t];
        const DevRun &r = runs[id >> 40];
        uint64_t i = id & 0xFFFFFFFFFFull;
        uint64_t kl, vl;
        const uint8_t *k = run_key(r, i, &kl);
        wave_copy(keys_out + okoff[x], k + hash_key_skip, s_klen[t], lane);
        if (!a.no_value && s_vlen_lo[t]) {
            const uint8_t *v = run_val(r, i, &vl);
            wave_copy(vals_out + ovoff[x], v + hdr, s_vlen_lo[t], lane);
        }
    }
}

__global__ void __launch_bounds__(256) k_multi_get_small(const DevRun *runs, int R,
                                                         MgFusedArgs a)
{
    mg_core(runs, R, a, a.start, a.start_len, a.stop, a.stop_len, a.hash_key_skip, a.out_hdr,
            a.out_blob, MG_BLOB_BYTES);
}

/* graph-capturable variant: all per-call state read from the d_in slice */
__global__ void __launch_bounds__(256) k_multi_get_graph(const DevRun *runs, int R,
                                                         const uint8_t *d_in, uint8_t *d_out)
{
    const MgGraphHdr *h = (const MgGraphHdr *)d_in;
    MgFusedArgs a{};
    const uint8_t *p = d_in + sizeof(MgGraphHdr);
    a.start = p;
    a.start_len = h->start_len;
    a.stop = p + h->start_len;
    a.stop_len = h->stop_len;
    a.sk_pat = p + h->start_len + h->stop_len;
    a.sk_pat_len = h->sk_pat_len;
    a.start_inclusive = h->start_inclusive;
    a.stop_inclusive = h->stop_inclusive;
    a.reverse = h->reverse;
    a.no_value = h->no_value;
    a.max_kv_count = h->max_kv_count;
    a.max_iteration_count = h->max_iteration_count;
    a.max_iteration_size = h->max_iteration_size;
    a.sk_ft = h->sk_ft;
    a.epoch_now = h->epoch_now;
    a.data_version = h->data_version;
    a.hash_key_skip = h->hash_key_skip;
    mg_core(runs, R, a, a.start, a.start_len, a.stop, a.stop_len, a.hash_key_skip,
            (int64_t *)d_out, d_out + 32, MG_BLOB_BYTES);
}

void launch_multi_get_graph(const DevRun *runs, int R, const uint8_t *d_in, uint8_t *d_out,
                            hipStream_t s)
{
    k_multi_get_graph<<<1, 256, 0, s>>>(runs, R, d_in, d_out);
}

/* batched full-range multi_get: one workgroup per request (hashkey); start =
 * [u16 len][hk], stop = pegasus_generate_next_blob(hk)
 * (pegasus_key_schema.h:64-81) built in-kernel */
__global__ void __launch_bounds__(256) k_multi_get_batch(const DevRun *runs, int R,
                                                         MgFusedArgs a, const uint8_t *hks,
                                                         const uint64_t *hk_offs,
                                                         uint64_t n_req, int64_t *hdrs,
                                                         uint8_t *blobs, uint64_t blob_stride)
{
    __shared__ uint8_t s_start[2 + 4096], s_stop[2 + 4096];
    __shared__ uint64_t s_slen, s_stlen;
    uint64_t req = blockIdx.x;
    if (req >= n_req)
        return;
    if (threadIdx.x == 0) {
        uint64_t hl = hk_offs[req + 1] - hk_offs[req];
        if (hl == 0 || hl > 4096) {
            hdrs[req * 4] = -1; /* host fallback */
            s_slen = (uint64_t)-1;
        } else {
            const uint8_t *hk = hks + hk_offs[req];
            s_start[0] = (uint8_t)(hl >> 8);
            s_start[1] = (uint8_t)hl;
            for (uint64_t b = 0; b < hl; b++)
                s_start[2 + b] = hk[b];
            s_slen = 2 + hl;
            for (uint64_t b = 0; b < s_slen; b++)
                s_stop[b] = s_start[b];
            uint64_t p = s_slen - 1;
            while (s_stop[p] == 0xFF)
                p--;
            s_stop[p]++;
            s_stlen = p + 1;
        }
    }
    __syncthreads();
    if (s_slen == (uint64_t)-1)
        return;
    mg_core(runs, R, a, s_start, s_slen, s_stop, s_stlen, s_slen /* 2+hklen */,
            hdrs + req * 4, blobs + req * blob_stride, blob_stride);
}

/* ================= persistent serving kernel =================
 * One resident workgroup serves single-request multi_gets from a pinned
 * mailbox: no per-call kernel dispatch (the ~15-20us launch-to-retire floor
 * of a one-workgroup kernel dominated the 40us serving path).  Exit is
 * BOUNDED: the poll loop leaves after MG_SRV_IDLE_MS without a request, on
 * the quit flag, always — a device-wide sync can stall at most the idle
 * window, never hang.  Memory ordering follows the producer-release /
 * consumer-acquire recipe at system scope (host <-> device over pinned). */
__global__ void __launch_bounds__(256) k_mg_server(const DevRun *runs, int R, MgMailbox *mb,
                                                   uint8_t *d_out)
{
    __shared__ uint64_t s_req;
    __shared__ int s_stop;
    int tid = threadIdx.x;
    /* adopt the current sequence so a pre-launch doorbell value is never
     * misread as a fresh request */
    uint64_t last =
        __hip_atomic_load(&mb->req_seq, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
    /* ~100 MHz constant wall clock; idle budget in ticks */
    const long long IDLE = (long long)MG_SRV_IDLE_MS * 100000;
    long long idle_start = (long long)wall_clock64();
    if (tid == 0)
        __hip_atomic_store(&mb->alive, 1u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
    for (;;) {
        if (tid == 0) {
            uint64_t r = last;
            int stop = 0;
            for (;;) {
                if (__hip_atomic_load(&mb->quit, __ATOMIC_ACQUIRE,
                                      __HIP_MEMORY_SCOPE_SYSTEM)) {
                    stop = 1;
                    break;
                }
                r = __hip_atomic_load(&mb->req_seq, __ATOMIC_ACQUIRE,
                                      __HIP_MEMORY_SCOPE_SYSTEM);
                if (r != last)
                    break;
                if ((long long)wall_clock64() - idle_start > IDLE) {
                    stop = 1;
                    break;
                }
            }
            s_req = r;
            s_stop = stop;
        }
        __syncthreads();
        if (s_stop)
            break;
        uint64_t seq = s_req;
        /* request slice layout = the graph lane's: [MgGraphHdr][start][stop][pat] */
        const MgGraphHdr *h = (const MgGraphHdr *)mb->req;
        MgFusedArgs a{};
        const uint8_t *p = mb->req + sizeof(MgGraphHdr);
        a.start = p;
        a.start_len = h->start_len;
        a.stop = p + h->start_len;
        a.stop_len = h->stop_len;
        a.sk_pat = p + h->start_len + h->stop_len;
        a.sk_pat_len = h->sk_pat_len;
        a.start_inclusive = h->start_inclusive;
        a.stop_inclusive = h->stop_inclusive;
        a.reverse = h->reverse;
        a.no_value = h->no_value;
        a.max_kv_count = h->max_kv_count;
        a.max_iteration_count = h->max_iteration_count;
        a.max_iteration_size = h->max_iteration_size;
        a.sk_ft = h->sk_ft;
        a.epoch_now = h->epoch_now;
        a.data_version = h->data_version;
        a.hash_key_skip = h->hash_key_skip;
        mg_core(runs, R, a, a.start, a.start_len, a.stop, a.stop_len, a.hash_key_skip,
                (int64_t *)d_out, d_out + 32, MG_BLOB_BYTES);
        __syncthreads();
        /* response prefix -> pinned: hdr + the used blob bytes (caller does
         * a device read of the tail for oversized results) */
        const int64_t *hdr = (const int64_t *)d_out;
        uint64_t blob_n = 0;
        if (hdr[0] >= 0)
            blob_n = 2 * ((uint64_t)hdr[0] + 1) * 8 + (uint64_t)hdr[2] + (uint64_t)hdr[3];
        uint64_t n = 32 + (blob_n < (uint64_t)(16 << 10) ? blob_n : (uint64_t)(16 << 10));
        for (uint64_t b = tid * 8; b + 8 <= n; b += blockDim.x * 8) {
            uint64_t w;
            __builtin_memcpy(&w, d_out + b, 8);
            __builtin_memcpy(mb->resp + b, &w, 8);
        }
        if (tid == 0)
            for (uint64_t b = n & ~7ull; b < n; b++)
                mb->resp[b] = d_out[b];
        __threadfence_system();
        __syncthreads();
        if (tid == 0)
            __hip_atomic_store(&mb->done_seq, seq, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_SYSTEM);
        last = seq;
        idle_start = (long long)wall_clock64();
    }
    __syncthreads();
    if (tid == 0)
        __hip_atomic_store(&mb->alive, 0u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

void launch_mg_server(const DevRun *runs, int R, MgMailbox *mb, uint8_t *d_out, hipStream_t s)
{
    k_mg_server<<<1, 256, 0, s>>>(runs, R, mb, d_out);
}

/* gather each request's used blob bytes into one packed buffer (one D2H) */
__global__ void k_pack_blobs(const uint8_t *blobs, uint64_t blob_stride, uint64_t n_req,
                             const uint64_t *used /* [n_req] */,
                             const uint64_t *pack_off /* [n_req+1] */, uint8_t *packed)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    for (uint64_t i = wave; i < n_req; i += nwaves)
        wave_copy(packed + pack_off[i], blobs + i * blob_stride, used[i], lane);
}

void launch_multi_get_small(const DevRun *d_runs, int R, const MgFusedArgs &a, hipStream_t s)
{
    k_multi_get_small<<<1, 256, 0, s>>>(d_runs, R, a);
}

void launch_multi_get_batch(const DevRun *d_runs, int R, const MgFusedArgs &a,
                            const uint8_t *d_hks, const uint64_t *d_hk_offs, uint64_t n_req,
                            int64_t *d_hdrs, uint8_t *d_blobs, uint64_t blob_stride,
                            hipStream_t s)
{
    k_multi_get_batch<<<(uint32_t)n_req, 256, 0, s>>>(d_runs, R, a, d_hks, d_hk_offs, n_req,
                                                      d_hdrs, d_blobs, blob_stride);
}

void launch_pack_blobs(const uint8_t *d_blobs, uint64_t blob_stride, uint64_t n_req,
                       const uint64_t *d_used, const uint64_t *d_pack_off, uint8_t *d_packed,
                       hipStream_t s)
{
    k_pack_blobs<<<grid_for(n_req * WAVE, BLOCK), BLOCK, 0, s>>>(d_blobs, blob_stride, n_req,
                                                                 d_used, d_pack_off, d_packed);
}

/* ================= LDS-staged rank (compaction) =================
 * One 256-thread workgroup ranks one 256-record block of one run.  The
 * block's search windows in every run (from the shift-8 bound table) are
 * staged into LDS once — packed key bytes + local offsets — and all
 * cross-run binary searches then run on-chip.  Workgroups whose windows
 * exceed the LDS budget (heavy skew) fall back to global searches inline;
 * results are identical either way. */
#define LRK_BLK 256
#define LRK_MAX_WIN 12288      /* total staged records across runs */
#define LRK_KEY_BYTES (96 * 1024)

__device__ static inline uint64_t lds_lower_bound(const uint32_t *skoff, const uint8_t *skeys,
                                                  uint64_t lo, uint64_t hi, const uint8_t *key,
                                                  uint64_t klen)
{
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1;
        const uint8_t *mk = skeys + skoff[mid];
        uint64_t ml = skoff[mid + 1] - skoff[mid];
        if (dev_key_cmp(mk, ml, key, klen) < 0)
            lo = mid + 1;
        else
            hi = mid;
    }
    return lo;
}
__device__ static inline uint64_t lds_upper_bound(const uint32_t *skoff, const uint8_t *skeys,
                                                  uint64_t lo, uint64_t hi, const uint8_t *key,
                                                  uint64_t klen)
{
    while (lo < hi) {
        uint64_t mid = (lo + hi) >> 1;
        const uint8_t *mk = skeys + skoff[mid];
        uint64_t ml = skoff[mid + 1] - skoff[mid];
        if (dev_key_cmp(mk, ml, key, klen) <= 0)
            lo = mid + 1;
        else
            hi = mid;
    }
    return lo;
}

__global__ void __launch_bounds__(LRK_BLK, 1) k_rank_compact_lds(
    const DevRun *runs, int R, const uint64_t *lo, const uint64_t *hi,
    const uint64_t *blk_prefix /* [R+1] workgroup counts */, CompactParams cp,
    const uint64_t *bt8_off, const uint64_t *bt8 /* shift-8 bound table */, uint64_t *order,
    uint64_t *keepw, uint8_t *changed, uint32_t *new_expire, uint64_t *ksz, uint64_t *vsz,
    CompactStatsDev *stats)
{
    __shared__ uint32_t skoff[LRK_MAX_WIN + RRDB_MAX_RUNS]; /* +1 per staged window */
    __shared__ uint8_t skeys[LRK_KEY_BYTES];
    __shared__ uint32_t s_w0[RRDB_MAX_RUNS];  /* window start (global idx) */
    __shared__ uint32_t s_wn[RRDB_MAX_RUNS];  /* window length */
    __shared__ uint32_t s_obase[RRDB_MAX_RUNS]; /* offset-array base */
    __shared__ int s_fallback;
    uint64_t wg = blockIdx.x;
    int r = 0;
    while (blk_prefix[r + 1] <= wg)
        r++;
    uint64_t j = wg - blk_prefix[r];
    uint64_t rec0 = lo[r] + (j << 8);
    uint64_t rec_end = rec0 + LRK_BLK < hi[r] ? rec0 + LRK_BLK : hi[r];
    int tid = threadIdx.x;
    /* window extents from the shift-8 table rows j and j+1 */
    const uint64_t *b0 = bt8 + (bt8_off[r] + j) * (uint64_t)R;
    const uint64_t *b1 = bt8 + (bt8_off[r] + j + 1) * (uint64_t)R;
    if (tid == 0) {
        uint64_t tot = 0, kb = 0, obase = 0;
        s_fallback = 0;
        for (int q = 0; q < R; q++) {
            uint64_t w0 = (q == r) ? rec0 : b0[q];
            uint64_t w1 = (q == r) ? rec_end : b1[q];
            if (w1 < w0)
                w1 = w0;
            uint64_t wn = w1 - w0;
            uint64_t kbq = runs[q].koff[w1] - runs[q].koff[w0];
            s_w0[q] = (uint32_t)w0;
            s_wn[q] = (uint32_t)wn;
            s_obase[q] = (uint32_t)(obase + (uint64_t)q); /* +q: one extra offset per window */
            obase += wn;
            tot += wn;
            kb += kbq;
            if (wn > 0xFFFFFFF0ull || w1 > 0xFFFFFFF0ull)
                s_fallback = 1;
        }
        if (tot > LRK_MAX_WIN || kb > LRK_KEY_BYTES)
            s_fallback = 1;
    }
    __syncthreads();
    int fallback = s_fallback;
    if (!fallback) {
        /* stage windows: offsets then packed key bytes (coalesced) */
        uint64_t key_base = 0;
        for (int q = 0; q < R; q++) {
            uint64_t w0 = s_w0[q], wn = s_wn[q];
            uint64_t gk0 = runs[q].koff[w0];
            uint32_t ob = s_obase[q];
            for (uint64_t t = tid; t <= wn; t += LRK_BLK)
                skoff[ob + t] = (uint32_t)(runs[q].koff[w0 + t] - gk0 + key_base);
            uint64_t kbq = runs[q].koff[w0 + wn] - gk0;
            const uint8_t *src = runs[q].keys + gk0;
            for (uint64_t b = tid * 4; b + 4 <= kbq; b += LRK_BLK * 4) {
                uint32_t w;
                __builtin_memcpy(&w, src + b, 4);
                __builtin_memcpy(&skeys[key_base + b], &w, 4);
            }
            /* byte tail */
            uint64_t tail = kbq & ~3ull;
            for (uint64_t b = tail + tid; b < kbq; b += LRK_BLK)
                skeys[key_base + b] = src[b];
            key_base += kbq;
        }
        __syncthreads();
    }
    /* one record per thread */
    uint64_t i = rec0 + tid;
    int disp = D_NONE;
    if (i < rec_end) {
        uint64_t kl;
        const uint8_t *k = run_key(runs[r], i, &kl);
        uint64_t rank = i - lo[r];
        int shadow = 0;
        for (int q = 0; q < R; q++) {
            if (q == r)
                continue;
            uint64_t pos; /* count of window records before our key */
            if (!fallback) {
                uint32_t ob = s_obase[q], wn = s_wn[q];
                uint64_t p = (q > r)
                                 ? lds_upper_bound(skoff + ob, skeys, 0, wn, k, kl)
                                 : lds_lower_bound(skoff + ob, skeys, 0, wn, k, kl);
                if (q > r && !shadow) {
                    if (p > 0) {
                        const uint8_t *pk = skeys + skoff[ob + p - 1];
                        uint64_t pl = skoff[ob + p] - skoff[ob + p - 1];
                        if (dev_key_cmp(pk, pl, k, kl) == 0)
                            shadow = 1;
                    } else if ((uint64_t)s_w0[q] > lo[q]) {
                        /* equal key may sit just before the window */
                        uint64_t pl;
                        const uint8_t *pk = run_key(runs[q], s_w0[q] - 1, &pl);
                        if (dev_key_cmp(pk, pl, k, kl) == 0)
                            shadow = 1;
                    }
                }
                pos = (uint64_t)s_w0[q] + p;
            } else {
                uint64_t ub = (q > r) ? dev_upper_bound(runs[q], k, kl, lo[q], hi[q])
                                      : dev_lower_bound(runs[q], k, kl, lo[q], hi[q]);
                if (q > r && !shadow && ub > lo[q]) {
                    uint64_t pl;
                    const uint8_t *pk = run_key(runs[q], ub - 1, &pl);
                    if (dev_key_cmp(pk, pl, k, kl) == 0)
                        shadow = 1;
                }
                pos = ub;
            }
            rank += pos - lo[q];
        }
        uint8_t ch;
        uint32_t nts;
        uint64_t okl, ovl;
        disp = dev_disposition(runs[r], i, cp, shadow, &ch, &nts, &okl, &ovl);
        order[rank] = ((uint64_t)r << 40) | i;
        keepw[rank] = (disp == D_KEEP) ? 1 : 0;
        if (changed)
            changed[rank] = ch;
        if (new_expire)
            new_expire[rank] = nts;
        if (ksz)
            ksz[rank] = okl;
        if (vsz)
            vsz[rank] = ovl;
    }
    /* wave-aggregated stats */
    int lane = tid % WAVE;
    unsigned long long b;
    b = __ballot(disp == D_SHADOWED);
    if (lane == 0 && b)
        atomicAdd(&stats->shadowed, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_TOMBSTONE);
    if (lane == 0 && b)
        atomicAdd(&stats->tombstones, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_EXPIRED);
    if (lane == 0 && b)
        atomicAdd(&stats->expired, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_FILTERED);
    if (lane == 0 && b)
        atomicAdd(&stats->filtered, (unsigned long long)__popcll(b));
    b = __ballot(disp == D_KEEP);
    if (lane == 0 && b)
        atomicAdd(&stats->output_records, (unsigned long long)__popcll(b));
}

void launch_rank_compact_lds(const DevRun *d_runs, int R, const uint64_t *d_lo,
                             const uint64_t *d_hi, const uint64_t *d_blk_prefix,
                             uint64_t n_blocks, const CompactParams &cp,
                             const uint64_t *d_bt8_off, const uint64_t *d_bt8,
                             uint64_t *d_order, uint64_t *d_keepw, uint8_t *d_changed,
                             uint32_t *d_new_expire, uint64_t *d_ksz, uint64_t *d_vsz,
                             CompactStatsDev *d_stats, hipStream_t s)
{
    k_rank_compact_lds<<<(uint32_t)n_blocks, LRK_BLK, 0, s>>>(
        d_runs, R, d_lo, d_hi, d_blk_prefix, cp, d_bt8_off, d_bt8, d_order, d_keepw, d_changed,
        d_new_expire, d_ksz, d_vsz, d_stats);
}

/* ================= per-run blocked bloom (point gets, §8(f)3) =================
 * 64B blocks (one cache line), 6 probe bits from one 64-bit FNV-1a hash;
 * ~10 bits/key like the reference's FullFilter profile
 * (pegasus_server_impl_init.cpp:816-841).  Lossless for correctness: a
 * negative skips the run's binary search, a positive falls through to it. */
__device__ __host__ static inline uint64_t bloom_hash(const uint8_t *k, uint64_t n)
{
    uint64_t h = 0xcbf29ce484222325ull;
    for (uint64_t i = 0; i < n; i++)
        h = (h ^ k[i]) * 0x100000001b3ull;
    return h;
}

__global__ void k_bloom_build(const DevRun run, uint64_t *bloom, uint64_t n_blocks)
{
    for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < run.n;
         i += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t kl = run.koff[i + 1] - run.koff[i];
        uint64_t h = bloom_hash(run.keys + run.koff[i], kl);
        uint64_t blk = (h >> 32) % n_blocks;
        uint64_t *base = bloom + blk * 8; /* 8 u64 = 64B block */
        uint32_t x = (uint32_t)h;
        for (int j = 0; j < 6; j++) {
            uint32_t bit = (x >> (j * 5)) & 31; /* bit within word */
            uint32_t word = ((x >> (j * 5 + 3)) ^ (x >> 27)) & 7;
            atomicOr((unsigned long long *)&base[word], 1ull << (bit + ((x >> j) & 1) * 32));
        }
    }
}

__device__ static inline int bloom_maybe_has(const DevRun &r, uint64_t h)
{
    if (!r.bloom)
        return 1;
    uint64_t blk = (h >> 32) % r.bloom_blocks;
    const uint64_t *base = r.bloom + blk * 8;
    uint32_t x = (uint32_t)h;
    for (int j = 0; j < 6; j++) {
        uint32_t bit = (x >> (j * 5)) & 31;
        uint32_t word = ((x >> (j * 5 + 3)) ^ (x >> 27)) & 7;
        if (!(base[word] & (1ull << (bit + ((x >> j) & 1) * 32))))
            return 0;
    }
    return 1;
}

void launch_bloom_build(const DevRun &run, uint64_t *d_bloom, uint64_t n_blocks, hipStream_t s)
{
    k_bloom_build<<<grid_for(run.n, BLOCK), BLOCK, 0, s>>>(run, d_bloom, n_blocks);
}

/* hashkey-prefix bloom: same block/probe scheme hashed over the
 * [u16 len][hashkey] prefix of each record */
__global__ void k_bloom_pfx_build(const DevRun run, uint64_t *bloom, uint64_t n_blocks)
{
    for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < run.n;
         i += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t kl = run.koff[i + 1] - run.koff[i];
        const uint8_t *k = run.keys + run.koff[i];
        uint64_t plen = kl >= 2 ? 2 + (((uint64_t)k[0] << 8) | k[1]) : kl;
        if (plen > kl)
            plen = kl;
        uint64_t h = bloom_hash(k, plen);
        uint64_t blk = (h >> 32) % n_blocks;
        uint64_t *base = bloom + blk * 8;
        uint32_t x = (uint32_t)h;
        for (int j = 0; j < 6; j++) {
            uint32_t bit = (x >> (j * 5)) & 31;
            uint32_t word = ((x >> (j * 5 + 3)) ^ (x >> 27)) & 7;
            atomicOr((unsigned long long *)&base[word], 1ull << (bit + ((x >> j) & 1) * 32));
        }
    }
}

__device__ static inline int pfx_bloom_maybe_has(const DevRun &r, uint64_t h)
{
    if (!r.pfx_bloom)
        return 1;
    uint64_t blk = (h >> 32) % r.pfx_bloom_blocks;
    const uint64_t *base = r.pfx_bloom + blk * 8;
    uint32_t x = (uint32_t)h;
    for (int j = 0; j < 6; j++) {
        uint32_t bit = (x >> (j * 5)) & 31;
        uint32_t word = ((x >> (j * 5 + 3)) ^ (x >> 27)) & 7;
        if (!(base[word] & (1ull << (bit + ((x >> j) & 1) * 32))))
            return 0;
    }
    return 1;
}

void launch_bloom_pfx_build(const DevRun &run, uint64_t *d_bloom, uint64_t n_blocks,
                            hipStream_t s)
{
    k_bloom_pfx_build<<<grid_for(run.n, BLOCK), BLOCK, 0, s>>>(run, d_bloom, n_blocks);
}
