/* engine.cpp — host side of the MI355X rrdb engine (C-ABI in
 * include/rrdb_engine.h).  One handle per partition; all record data lives in
 * HBM; every per-record data-path operation runs in the HIP kernels of
 * kernels.hip.  There is NO CPU fallback: rrdb_open fails without a GPU.
 *
 * Host responsibilities only: run registry, request parsing (incl. the
 * user_specified_compaction JSON of compaction_operation.cpp:160-190),
 * range-bound construction (pegasus_key_schema.h:41-122 byte logic), scan
 * contexts (pegasus_scan_context.h:33-140 equivalent), result marshalling.
 *
 * Reference semantics restated per handler: see the comment on each function
 * and include/rrdb_engine.h.
 */
#include <hip/hip_runtime.h>

#include <algorithm>
#include <map>
#include <chrono>
#include <cstdint>
#include <cstring>
#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/rrdb_engine.h"
#include "engine_common.h"

/* ---- launchers (kernels.hip) ---- */
void launch_crc64_table_init(const uint64_t *host_table);
void launch_bounds(const DevRun *, int, const uint8_t *, uint64_t, uint64_t *, int, hipStream_t);
void launch_rank(const DevRun *, int, const uint64_t *, const uint64_t *, const uint64_t *,
                 uint64_t, uint64_t *, uint8_t *, const uint64_t *, const uint64_t *, int,
                 hipStream_t);
void launch_bound_table(const DevRun *, int, const uint64_t *, const uint64_t *,
                        const uint64_t *, uint64_t, int, uint64_t *, const uint64_t *,
                        const uint64_t *, int, hipStream_t);
void launch_visible(const DevRun *, const uint64_t *, const uint8_t *, uint64_t, uint64_t *,
                    hipStream_t);
void launch_gather(const uint64_t *, const uint64_t *, const uint64_t *, uint64_t, uint64_t *,
                   hipStream_t);
void launch_psum(const uint64_t *, uint64_t *, uint64_t, uint64_t *, hipStream_t);
uint64_t psum_scratch_elems(uint64_t);
void launch_get(const DevRun *, int, const uint8_t *, const uint64_t *, uint64_t, uint32_t,
                uint32_t, int32_t *, uint64_t *, uint64_t *, uint32_t *, hipStream_t);
void launch_emit_values(const DevRun *, const uint64_t *, const int32_t *, uint64_t, uint32_t,
                        const uint64_t *, uint8_t *, hipStream_t);
void launch_scan_state(const DevRun *, const uint64_t *, uint64_t, const ScanParams &, uint8_t *,
                       uint64_t *, uint64_t *, hipStream_t);
void launch_normal_flags(const uint8_t *, uint64_t, uint64_t *, hipStream_t);
void launch_cutoff(const uint64_t *, const uint64_t *, uint64_t, uint64_t, uint64_t *,
                   hipStream_t);
void launch_cut_sizes(const uint8_t *, uint64_t, uint64_t, uint64_t *, uint64_t *, hipStream_t);
void launch_emit_scan(const DevRun *, const uint64_t *, uint64_t, const uint8_t *, uint64_t,
                      const uint64_t *, const uint64_t *, const uint64_t *, const ScanParams &,
                      uint8_t *, uint8_t *, uint64_t *, uint64_t *, int32_t *, uint64_t,
                      hipStream_t);
void launch_emit_rows(const DevRun *, const uint64_t *, const uint64_t *, uint64_t,
                      const uint64_t *, const uint64_t *, const ScanParams &, uint8_t *,
                      uint8_t *, hipStream_t);
void launch_multi_get_small(const DevRun *, int, const MgFusedArgs &, hipStream_t);
void launch_multi_get_graph(const DevRun *, int, const uint8_t *, uint8_t *, hipStream_t);
void launch_mg_server(const DevRun *, int, MgMailbox *, uint8_t *, hipStream_t);
void launch_bloom_build(const DevRun &, uint64_t *, uint64_t, hipStream_t);
void launch_bloom_pfx_build(const DevRun &, uint64_t *, uint64_t, hipStream_t);
void launch_build_tails(const uint8_t *, uint64_t, uint64_t, uint64_t *, hipStream_t);
void launch_build_meta(const uint8_t *, const uint64_t *, uint64_t, const uint64_t *, uint64_t,
                       uint32_t, uint64_t *, hipStream_t);
void launch_rank_compact_lds(const DevRun *, int, const uint64_t *, const uint64_t *,
                             const uint64_t *, uint64_t, const CompactParams &,
                             const uint64_t *, const uint64_t *, uint64_t *, uint64_t *,
                             uint8_t *, uint32_t *, uint64_t *, uint64_t *, CompactStatsDev *,
                             hipStream_t);
void launch_valid_beyond(const DevRun *, int, const uint8_t *, uint64_t, int, uint32_t *,
                         hipStream_t);
void launch_first_eq(const DevRun *, const uint64_t *, uint64_t, const uint8_t *, uint64_t,
                     uint32_t *, hipStream_t);
void launch_rank_ldst(const DevRun *, int, const uint64_t *, const uint64_t *,
                      const uint64_t *, uint64_t, uint64_t *, uint8_t *, const uint64_t *,
                      const uint64_t *, int, hipStream_t);
void launch_anchor_rows(const DevRun *, int, int, const uint64_t *, const uint64_t *, int,
                        uint64_t, uint64_t *, hipStream_t);
void launch_rank_grp_compact(const DevRun *, int, const uint64_t *, const uint64_t *, uint64_t,
                             const CompactParams &, uint64_t *, uint64_t *, uint8_t *,
                             uint32_t *, uint64_t *, uint64_t *, CompactStatsDev *, int,
                             hipStream_t);
void launch_rank_grp_view(const DevRun *, int, const uint64_t *, const uint64_t *, uint64_t,
                          uint64_t *, uint8_t *, hipStream_t);
void launch_rank_grp_count(const DevRun *, int, const uint64_t *, const uint64_t *, uint64_t,
                           const ScanParams &, CompactStatsDev *, hipStream_t);
void launch_count_single(const DevRun *, const uint64_t *, const uint64_t *,
                         const ScanParams &, CompactStatsDev *, uint64_t, hipStream_t);
void launch_rank_compact_ldst(const DevRun *, int, const uint64_t *, const uint64_t *,
                              const uint64_t *, uint64_t, const CompactParams &, uint64_t *,
                              uint64_t *, uint8_t *, uint32_t *, uint64_t *, uint64_t *,
                              uint64_t *, const uint64_t *, const uint64_t *, int,
                              CompactStatsDev *, hipStream_t);
void launch_rank_compact(const DevRun *, int, const uint64_t *, const uint64_t *,
                         const uint64_t *, uint64_t, const CompactParams &, uint64_t *,
                         uint64_t *, uint8_t *, uint32_t *, uint64_t *, uint64_t *, uint64_t *,
                         const uint64_t *, const uint64_t *, int, CompactStatsDev *,
                         hipStream_t);
void launch_emit_compact_inmajor(const DevRun *, int, const uint64_t *, uint64_t,
                                 const uint64_t *, const uint64_t *, const uint8_t *,
                                 const uint32_t *, const uint64_t *, const uint64_t *,
                                 const uint64_t *, uint32_t, uint8_t *, uint8_t *, uint64_t *,
                                 uint64_t *, uint64_t *, uint64_t, hipStream_t);
void launch_emit_compact_chunked(const DevRun *, const uint64_t *, uint64_t, const uint64_t *,
                                 const uint8_t *, const uint32_t *, const uint64_t *,
                                 const uint64_t *, const uint64_t *, uint32_t, uint64_t,
                                 uint64_t, uint64_t, uint64_t *, uint64_t *, uint32_t *,
                                 uint32_t *, uint8_t *, uint8_t *, uint64_t *, uint64_t *,
                                 uint64_t *, uint64_t *, uint64_t *,
                                 uint64_t, uint64_t, hipStream_t);
void launch_emit_compact(const DevRun *, const uint64_t *, uint64_t, const uint64_t *,
                         const uint8_t *, const uint32_t *, const uint64_t *, const uint64_t *,
                         const uint64_t *, uint32_t, uint8_t *, uint8_t *, uint64_t *, uint64_t *,
                         uint64_t *, uint64_t, hipStream_t);

#define HIP_OK(x)                                                                                  \
    do {                                                                                           \
        hipError_t e_ = (x);                                                                       \
        if (e_ != hipSuccess) {                                                                    \
            fprintf(stderr, "rrdb-hip fatal: %s at %s:%d\n", hipGetErrorString(e_), __FILE__,      \
                    __LINE__);                                                                     \
            abort();                                                                               \
        }                                                                                          \
    } while (0)

namespace {

/* ================ crc64 table (host build; same construction as
 * reference crc.cpp:236-295, restated) ================ */
uint64_t host_crc64_table[256];
void build_crc64_table()
{
    static const int bits[] = {63, 61, 59, 58, 56, 55, 52, 49, 48, 47, 46, 44, 41, 37, 36, 34,
                               32, 31, 28, 26, 23, 22, 19, 16, 13, 12, 10, 9,  6,  4,  3,  0};
    uint64_t poly = 0;
    for (int b : bits)
        poly |= 1ull << (63 - b);
    for (int i = 0; i < 256; i++) {
        uint64_t k = (uint64_t)i;
        for (int j = 0; j < 8; j++)
            k = (k & 1) ? ((k >> 1) ^ poly) : (k >> 1);
        host_crc64_table[i] = k;
    }
}

/* host crc64 over a buffer (same table); used for checkpoint-file integrity */
static uint64_t host_crc64(const void *p, uint64_t n)
{
    const uint8_t *b = (const uint8_t *)p;
    uint64_t crc = ~0ull;
    for (uint64_t i = 0; i < n; i++)
        crc = host_crc64_table[(uint8_t)(crc ^ b[i])] ^ (crc >> 8);
    return ~crc;
}

/* ================ result arena ================ */
struct Arena {
    std::vector<void *> host_blocks;
    std::vector<void *> dev_ptrs;
    void *alloc(size_t n)
    {
        void *p = malloc(n ? n : 1);
        host_blocks.push_back(p);
        return p;
    }
    ~Arena()
    {
        for (void *p : host_blocks)
            free(p);
        for (void *p : dev_ptrs)
            (void)hipFree(p);
    }
};

Arena *result_init(rrdb_result *r)
{
    memset(r, 0, sizeof(*r));
    Arena *a = new Arena();
    r->_arena = a;
    return a;
}

/* ================ tiny JSON parser (independent of the oracle's) ======== */
struct Json {
    enum Type { NUL, BOOL, NUM, STR, ARR, OBJ } type = NUL;
    bool b = false;
    long long num = 0;
    std::string str;
    std::vector<Json> arr;
    std::vector<std::pair<std::string, Json>> obj;
    const Json *get(const std::string &k) const
    {
        for (auto &kv : obj)
            if (kv.first == k)
                return &kv.second;
        return nullptr;
    }
};

struct JParse {
    const char *s, *e;
    bool ok = true;
    void ws()
    {
        while (s < e && (*s == ' ' || *s == '\t' || *s == '\n' || *s == '\r'))
            s++;
    }
    bool lit(char c)
    {
        ws();
        if (s < e && *s == c) {
            s++;
            return true;
        }
        return false;
    }
    std::string pstr()
    {
        ws();
        std::string out;
        if (s >= e || *s != '"') {
            ok = false;
            return out;
        }
        s++;
        while (s < e && *s != '"') {
            char c = *s++;
            if (c == '\\' && s < e) {
                char esc = *s++;
                switch (esc) {
                case 'n': c = '\n'; break;
                case 't': c = '\t'; break;
                case 'r': c = '\r'; break;
                case 'b': c = '\b'; break;
                case 'f': c = '\f'; break;
                case 'u': {
                    if (e - s < 4) { ok = false; return out; }
                    unsigned v = 0;
                    for (int i = 0; i < 4; i++) {
                        char h = s[i];
                        v <<= 4;
                        if (h >= '0' && h <= '9') v |= h - '0';
                        else if (h >= 'a' && h <= 'f') v |= h - 'a' + 10;
                        else if (h >= 'A' && h <= 'F') v |= h - 'A' + 10;
                        else { ok = false; return out; }
                    }
                    s += 4;
                    c = (char)v;
                    break;
                }
                default: c = esc; break;
                }
            }
            out.push_back(c);
        }
        if (s >= e) {
            ok = false;
            return out;
        }
        s++;
        return out;
    }
    Json value()
    {
        Json j;
        ws();
        if (s >= e) {
            ok = false;
            return j;
        }
        char c = *s;
        if (c == '"') {
            j.type = Json::STR;
            j.str = pstr();
        } else if (c == '{') {
            s++;
            j.type = Json::OBJ;
            ws();
            if (lit('}'))
                return j;
            for (;;) {
                std::string k = pstr();
                if (!ok || !lit(':')) {
                    ok = false;
                    return j;
                }
                j.obj.emplace_back(std::move(k), value());
                if (!ok)
                    return j;
                if (lit('}'))
                    return j;
                if (!lit(',')) {
                    ok = false;
                    return j;
                }
            }
        } else if (c == '[') {
            s++;
            j.type = Json::ARR;
            ws();
            if (lit(']'))
                return j;
            for (;;) {
                j.arr.push_back(value());
                if (!ok)
                    return j;
                if (lit(']'))
                    return j;
                if (!lit(',')) {
                    ok = false;
                    return j;
                }
            }
        } else if (c == 't' && e - s >= 4 && !strncmp(s, "true", 4)) {
            j.type = Json::BOOL;
            j.b = true;
            s += 4;
        } else if (c == 'f' && e - s >= 5 && !strncmp(s, "false", 5)) {
            j.type = Json::BOOL;
            s += 5;
        } else if (c == 'n' && e - s >= 4 && !strncmp(s, "null", 4)) {
            s += 4;
        } else if (c == '-' || (c >= '0' && c <= '9')) {
            j.type = Json::NUM;
            bool neg = c == '-';
            if (neg)
                s++;
            if (s >= e || *s < '0' || *s > '9') {
                ok = false;
                return j;
            }
            long long v = 0;
            while (s < e && *s >= '0' && *s <= '9')
                v = v * 10 + (*s++ - '0');
            j.num = neg ? -v : v;
        } else {
            ok = false;
        }
        return j;
    }
};

bool json_parse(const std::string &text, Json &out)
{
    JParse p{text.c_str(), text.c_str() + text.size()};
    out = p.value();
    return p.ok;
}

/* host-side parsed rules/ops (mirrors compaction_operation.cpp:116-190) */
struct HostRule {
    int type = -1;       /* DFR_* */
    int match_type = DSM_INVALID;
    uint32_t start_ttl = 0, stop_ttl = 0;
    std::string pattern;
};
struct HostOp {
    int type = -1; /* DOP_* */
    int ut_type = DUT_INVALID;
    uint32_t ut_value = 0;
    std::vector<HostRule> rules;
};

int enum_smt(const std::string &s)
{
    if (s == "SMT_MATCH_ANYWHERE") return DSM_ANYWHERE;
    if (s == "SMT_MATCH_PREFIX") return DSM_PREFIX;
    if (s == "SMT_MATCH_POSTFIX") return DSM_POSTFIX;
    return DSM_INVALID;
}
int enum_frt(const std::string &s)
{
    if (s == "FRT_HASHKEY_PATTERN") return DFR_HASHKEY;
    if (s == "FRT_SORTKEY_PATTERN") return DFR_SORTKEY;
    if (s == "FRT_TTL_RANGE") return DFR_TTL_RANGE;
    return -1;
}
int enum_cot(const std::string &s)
{
    if (s == "COT_UPDATE_TTL") return DOP_UPDATE_TTL;
    if (s == "COT_DELETE") return DOP_DELETE;
    return -1;
}
int enum_utot(const std::string &s)
{
    if (s == "UTOT_FROM_NOW") return DUT_FROM_NOW;
    if (s == "UTOT_FROM_CURRENT") return DUT_FROM_CURRENT;
    if (s == "UTOT_TIMESTAMP") return DUT_TIMESTAMP;
    return DUT_INVALID;
}

/* invalid rules skipped; op kept only with >=1 valid rule and valid params
 * (compaction_operation.cpp:160-190; all DEFINE_JSON fields required) */
bool parse_user_ops(const std::string &text, std::vector<HostOp> &out)
{
    out.clear();
    Json root;
    if (!json_parse(text, root) || root.type != Json::OBJ)
        return false;
    const Json *ops = root.get("ops");
    if (!ops || ops->type != Json::ARR)
        return true; /* decodes, but no ops */
    for (const Json &jop : ops->arr) {
        if (jop.type != Json::OBJ)
            return false;
        const Json *jt = jop.get("type");
        const Json *jp = jop.get("params");
        const Json *jr = jop.get("rules");
        /* a missing "params" decodes as "" (json_helper.h:136-143
         * JSON_TRY_DECODE_ENTRY tolerates absent members): delete_key
         * ignores params, update_ttl's own params decode then fails and
         * drops the op */
        if (!jt || jt->type != Json::STR || (jp && jp->type != Json::STR) || !jr ||
            jr->type != Json::ARR)
            continue;
        std::string op_params = jp ? jp->str : std::string();
        HostOp op;
        op.type = enum_cot(jt->str);
        for (const Json &jrule : jr->arr) {
            if (jrule.type != Json::OBJ)
                continue;
            const Json *rt = jrule.get("type");
            const Json *rp = jrule.get("params");
            if (!rt || rt->type != Json::STR || !rp || rp->type != Json::STR)
                continue;
            HostRule rule;
            rule.type = enum_frt(rt->str);
            if (rule.type < 0)
                continue;
            Json rparams;
            if (!json_parse(rp->str, rparams) || rparams.type != Json::OBJ)
                continue;
            if (rule.type == DFR_TTL_RANGE) {
                const Json *a = rparams.get("start_ttl");
                const Json *b = rparams.get("stop_ttl");
                if (!a || a->type != Json::NUM || !b || b->type != Json::NUM)
                    continue;
                rule.start_ttl = (uint32_t)a->num;
                rule.stop_ttl = (uint32_t)b->num;
            } else {
                const Json *pat = rparams.get("pattern");
                const Json *mt = rparams.get("match_type");
                if (!pat || pat->type != Json::STR || !mt || mt->type != Json::STR)
                    continue;
                rule.pattern = pat->str;
                rule.match_type = enum_smt(mt->str);
            }
            op.rules.push_back(std::move(rule));
        }
        if (op.rules.empty() || op.type < 0)
            continue;
        if (op.type == DOP_UPDATE_TTL) {
            Json oparams;
            if (!json_parse(op_params, oparams) || oparams.type != Json::OBJ)
                continue;
            const Json *t = oparams.get("type");
            const Json *v = oparams.get("value");
            /* both fields required: the reference's tolerant decode would
             * accept one missing and read the uninitialized member
             * (compaction_operation.cpp:75 initializes neither) — UB with
             * no semantics to restate, so the op is dropped instead */
            if (!t || t->type != Json::STR || !v || v->type != Json::NUM)
                continue;
            op.ut_type = enum_utot(t->str);
            op.ut_value = (uint32_t)v->num;
        }
        out.push_back(std::move(op));
    }
    return true;
}

/* ================ key byte helpers (pegasus_key_schema.h) ================ */
/* the 2-byte length prefix caps hash keys at 64KiB-1; the reference
 * CHECK_LTs (pegasus_key_schema.h:43) — as a library we reject with
 * kInvalidArgument at the C-ABI boundary instead of aborting */
static inline bool hklen_ok(uint64_t hklen) { return hklen < 0xFFFFull; }
std::string make_key(const uint8_t *hk, uint64_t hklen, const uint8_t *sk, uint64_t sklen)
{
    std::string k;
    k.reserve(2 + hklen + sklen);
    k.push_back((char)(hklen >> 8));
    k.push_back((char)(hklen & 0xFF));
    k.append((const char *)hk, hklen);
    if (sk)
        k.append((const char *)sk, sklen);
    return k;
}
std::string next_blob(std::string k)
{
    size_t p = k.size() - 1;
    while ((uint8_t)k[p] == 0xFF)
        p--;
    k[p] = (char)((uint8_t)k[p] + 1);
    k.resize(p + 1);
    return k;
}
int key_cmp(const std::string &a, const std::string &b)
{
    int c = memcmp(a.data(), b.data(), std::min(a.size(), b.size()));
    if (c)
        return c;
    return a.size() < b.size() ? -1 : (a.size() > b.size() ? 1 : 0);
}

[[maybe_unused]] uint32_t hdr_len(uint32_t ver) { return ver == 0 ? 4u : (ver == 1 ? 12u : 13u); }

/* sorted order => every key in a run shares lcp(first key, last key); device
 * searches skip that many leading bytes (whole 8B words, capped at 16) */
static uint32_t lcp_exact32(const uint8_t *a, uint64_t al, const uint8_t *b, uint64_t bl)
{
    uint32_t m = (uint32_t)std::min<uint64_t>(std::min(al, bl), 32);
    uint32_t i = 0;
    while (i < m && a[i] == b[i])
        i++;
    return i;
}
static void set_pfx(uint32_t lcp, uint32_t *pfx_skip, uint32_t *lcp_exact)
{
    *lcp_exact = lcp;
    *pfx_skip = std::min(lcp, 16u) & ~7u;
}

/* constant key stride lets device searches skip the offset-pair loads */
/* constant encoded-value stride (0 = variable); mirrors detect_fixed_klen */
static uint32_t detect_fixed_vlen(const uint64_t *voff, uint64_t n)
{
    if (n == 0)
        return 0;
    uint64_t s = voff[1] - voff[0];
    if (s == 0 || s > UINT32_MAX)
        return 0;
    for (uint64_t i = 1; i < n; i++)
        if (voff[i + 1] - voff[i] != s)
            return 0;
    return (uint32_t)s;
}

/* constant encoded-value stride over PUT records only (tombstones carry
 * empty values and break the plain stride): the compaction OUTPUT contains
 * only surviving PUTs, so this is the output stride the fixed-emit mode
 * needs even when inputs hold tombstones */
static uint32_t detect_fixed_vlen_put(const uint64_t *voff, const uint64_t *sk, uint64_t n)
{
    uint64_t s = 0;
    for (uint64_t i = 0; i < n; i++) {
        if (sk[i] & 1)
            continue;
        uint64_t d = voff[i + 1] - voff[i];
        if (d == 0 || d > UINT32_MAX)
            return 0;
        if (s == 0)
            s = d;
        else if (d != s)
            return 0;
    }
    return (uint32_t)s;
}

static uint32_t detect_fixed_klen(const uint64_t *koff, uint64_t n)
{
    if (n == 0)
        return 0;
    uint64_t s = koff[1] - koff[0];
    if (s == 0 || s > UINT32_MAX)
        return 0;
    for (uint64_t i = 1; i < n; i++)
        if (koff[i + 1] - koff[i] != s)
            return 0;
    return (uint32_t)s;
}

/* ================ engine ================ */
struct RunBuf {
    uint8_t *keys = nullptr;
    uint64_t *koff = nullptr;
    uint8_t *vals = nullptr;
    uint64_t *voff = nullptr;
    uint64_t *sk = nullptr;
    uint64_t n = 0;
    uint64_t *bloom = nullptr; /* blocked bloom over full keys (may be null) */
    uint64_t bloom_blocks = 0;
    uint32_t fixed_klen = 0; /* nonzero when every key in the run has this length */
    uint32_t pfx_skip = 0;   /* lcp(first,last) floored to 8B words, <=16 */
    uint32_t lcp_exact = 0;  /* exact lcp(first,last), capped at 32 */
    uint64_t *tails = nullptr; /* packed BE tail words (word-probe mode) */
    uint64_t *pfx_bloom = nullptr; /* hashkey-prefix blocked bloom */
    uint64_t pfx_bloom_blocks = 0;
    uint64_t *meta = nullptr;  /* (expire_ts<<32)|kind disposition column */
    uint32_t fixed_vlen = 0;   /* constant encoded-value stride (0=variable) */
    uint32_t fixed_vlen_put = 0; /* constant PUT-value stride (emit output) */
};

struct HipScanCtx {
    int64_t id = 0;
    uint32_t parked_at = 0; /* epoch_now when (re-)parked; 5-min GC */
    uint64_t *d_view = nullptr;
    uint64_t view_n = 0;
    uint64_t cursor = 0;
    int32_t batch_size = -1;
    uint8_t no_value = 0, validate_hash_req = 1, return_expire_ts = 0, only_return_count = 0,
            on_device_out = 0;
    int32_t hk_ft = 0, sk_ft = 0;
    uint8_t *d_hk_pat = nullptr;
    uint64_t hk_pat_len = 0;
    uint8_t *d_sk_pat = nullptr;
    uint64_t sk_pat_len = 0;
    ~HipScanCtx()
    {
        if (d_view)
            (void)hipFree(d_view);
        if (d_hk_pat)
            (void)hipFree(d_hk_pat);
        if (d_sk_pat)
            (void)hipFree(d_sk_pat);
    }
};

struct HipEngine {
    int32_t app_id = 0, pidx = 0, partition_version = -1;
    int device = 0;
    hipStream_t stream = nullptr;
    uint32_t data_version = 1, default_ttl = 0;
    bool validate_hash = false, manual_compact_disabled = false;
    bool bloom_enabled = true; /* env "rocksdb.filter_type": common(default)|none */
    uint32_t max_iter_count = 1000, mg_max_iter_count = 3000;
    uint64_t mg_max_iter_size = 30ull << 20, iter_time_ms = 30000;
    std::vector<RunBuf> runs;
    DevRun *d_runs = nullptr;
    bool d_runs_dirty = true;
    int ldst_elig_cache = -1; /* -1 unknown; recomputed when the run set changes */
    /* phase-1 state of a split compaction (rrdb_manual_compact_begin):
     * device pointers are arena memory, valid until the next scratch_reset */
    struct PendingCompact {
        bool active = false, trivial = false, keep_inputs = false;
        int R = 0;
        uint64_t total = 0;
        DevRun *dr = nullptr;
        uint64_t *d_wp = nullptr, *d_order = nullptr, *d_keepw = nullptr;
        uint8_t *d_changed = nullptr;
        uint32_t *d_new_expire = nullptr;
        uint64_t *d_kpos = nullptr, *d_koffs = nullptr, *d_voffs = nullptr;
        uint64_t *d_rank_of = nullptr;
        CompactStatsDev *d_stats = nullptr;
        uint64_t fk = 0, fv = 0; /* all-fixed-stride emit (ksz/vsz skipped) */
        bool no_rewrite = false;  /* changed/new_expire arrays skipped */
        hipEvent_t ev[6] = {};
        rrdb_compact_stats st{};
    } pend;
    uint64_t *pend_sizes = nullptr; /* pinned [6]: output sizes d2h target */
    CompactStatsDev *pend_stats_h = nullptr; /* pinned [8]: stats d2h target */
    /* events of the last compaction pass, resolved lazily by phase_ms() so
     * the keep_inputs finish can return without draining the emit */
    hipEvent_t tev[6] = {};
    bool tev_live = false, tev_have_emit = false;
    void resolve_phase_events()
    {
        if (!tev_live)
            return;
        tev_live = false;
        float ms;
        HIP_OK(hipEventSynchronize(tev[1]));
        HIP_OK(hipEventElapsedTime(&ms, tev[0], tev[1]));
        phase_ms["compact_rank"] = ms;
        phase_ms["compact_flags"] = 0.0;
        if (tev_have_emit) {
            HIP_OK(hipEventSynchronize(tev[4]));
            HIP_OK(hipEventElapsedTime(&ms, tev[3], tev[4]));
            phase_ms["compact_emit"] = ms;
            HIP_OK(hipEventElapsedTime(&ms, tev[0], tev[4]));
            phase_ms["compact_total"] = ms;
        }
    }
    /* serving-lane staging for the fused multi_get: one pinned H2D of
     * [start|stop|pattern], one pinned D2H of hdr + blob prefix (small-op
     * latency; pageable copies + two syncs dominated the 77us/call path) */
    static constexpr uint64_t MG_IN_CAP = 64 << 10;
    static constexpr uint64_t MG_OUT_PREFIX = 16 << 10;
    uint8_t *mg_hin = nullptr, *mg_din = nullptr, *mg_hout = nullptr;
    uint8_t *mg_dout = nullptr; /* persistent output buffer (graph-stable) */
    hipGraphExec_t mg_graph = nullptr;
    hipStream_t mg_capture_stream = nullptr; /* idle stream used ONLY for
        capture: capturing e->stream while earlier async work (e.g. a pending
        compaction) is in flight aborts inside the ROCm runtime */
    uint64_t mg_graph_gen = ~0ull; /* runs_gen the graph was captured at */
    uint64_t runs_gen = 0;         /* bumped whenever d_runs is rebuilt */
    bool mg_graph_enabled = true;  /* env "engine.mg_graph" */
    void mg_lane_init()
    {
        if (mg_hin)
            return;
        HIP_OK(hipHostMalloc((void **)&mg_hin, MG_IN_CAP));
        HIP_OK(hipMalloc(&mg_din, MG_IN_CAP));
        HIP_OK(hipHostMalloc((void **)&mg_hout, 32 + MG_OUT_PREFIX));
        HIP_OK(hipMalloc(&mg_dout, 32 + MG_BLOB_BYTES));
    }
    /* (re)capture the serving graph: H2D request slice -> fused kernel ->
     * D2H hdr+prefix.  One hipGraphLaunch replaces 3 submissions. */
    bool mg_graph_ready()
    {
        if (!mg_graph_enabled)
            return false;
        /* refresh the device run table FIRST: it bumps runs_gen when the
         * run set changed, and a stale graph would replay against freed
         * run buffers (GPU aperture fault) */
        DevRun *dr = dev_runs();
        if (mg_graph && mg_graph_gen == runs_gen)
            return true;
        if (mg_graph) {
            (void)hipGraphExecDestroy(mg_graph);
            mg_graph = nullptr;
        }
        int R = (int)runs.size();
        if (!mg_capture_stream && hipStreamCreate(&mg_capture_stream) != hipSuccess) {
            mg_graph_enabled = false;
            return false;
        }
        hipStream_t cs = mg_capture_stream;
        hipGraph_t g = nullptr;
        if (hipStreamBeginCapture(cs, hipStreamCaptureModeThreadLocal) != hipSuccess)
            return false;
        bool ok = hipMemcpyAsync(mg_din, mg_hin, MG_GRAPH_IN, hipMemcpyHostToDevice,
                                 cs) == hipSuccess;
        launch_multi_get_graph(dr, R, mg_din, mg_dout, cs);
        ok = ok && hipMemcpyAsync(mg_hout, mg_dout, 32 + MG_OUT_PREFIX,
                                  hipMemcpyDeviceToHost, cs) == hipSuccess;
        if (hipStreamEndCapture(cs, &g) != hipSuccess || !ok || !g) {
            if (g)
                (void)hipGraphDestroy(g);
            mg_graph_enabled = false; /* capture unsupported: stay on the
                                         plain path */
            return false;
        }
        if (hipGraphInstantiate(&mg_graph, g, nullptr, nullptr, 0) != hipSuccess) {
            (void)hipGraphDestroy(g);
            mg_graph_enabled = false;
            return false;
        }
        (void)hipGraphDestroy(g);
        mg_graph_gen = runs_gen;
        return true;
    }

    /* ---- persistent serving kernel (env engine.mg_persist, default on):
     * a resident 1-workgroup kernel polls the pinned mailbox, removing the
     * per-call dispatch floor.  It self-exits within MG_SRV_IDLE_MS of the
     * last request, so device-wide syncs stall at most that long. ---- */
    MgMailbox *mg_mb = nullptr;
    hipStream_t mg_srv_stream = nullptr;
    uint64_t mg_srv_gen = ~0ull; /* runs_gen the resident kernel was launched at */
    uint64_t mg_srv_seq = 0;
    /* 0 = off, 1 = auto (default: engage only after a sustained burst on
     * THIS handle — a resident kernel helps a hot partition but its poll
     * windows tax handles that are merely visited round-robin), 2 = on */
    int mg_persist_mode = 1;
    bool mg_persist_enabled = true; /* false = runtime failure, stay off */
    std::chrono::steady_clock::time_point mg_last_call{};
    uint32_t mg_hot = 0;
    bool mg_persist_wanted()
    {
        if (!mg_persist_enabled || mg_persist_mode == 0)
            return false;
        if (mg_persist_mode == 2)
            return true;
        auto now = std::chrono::steady_clock::now();
        auto us = std::chrono::duration_cast<std::chrono::microseconds>(now - mg_last_call)
                      .count();
        mg_last_call = now;
        if (us < 1000) {
            if (mg_hot < 1000)
                mg_hot++;
        } else {
            mg_hot = 0;
        }
        return mg_hot >= 32;
    }
    void server_quit_sync()
    {
        if (!mg_mb || mg_srv_gen == ~0ull)
            return;
        __atomic_store_n(&mg_mb->quit, 1u, __ATOMIC_RELEASE);
        auto t0 = std::chrono::steady_clock::now();
        while (__atomic_load_n(&mg_mb->alive, __ATOMIC_ACQUIRE)) {
            if (std::chrono::duration_cast<std::chrono::milliseconds>(
                    std::chrono::steady_clock::now() - t0)
                    .count() > 200) {
                /* should be impossible (bounded loop); stop using it */
                mg_persist_enabled = false;
                break;
            }
        }
        (void)hipStreamSynchronize(mg_srv_stream); /* retire the kernel */
        __atomic_store_n(&mg_mb->quit, 0u, __ATOMIC_RELEASE);
        mg_srv_gen = ~0ull;
    }
    bool server_ready()
    {
        if (!mg_persist_enabled)
            return false;
        DevRun *dr = dev_runs(); /* bumps runs_gen when the run set changed */
        if (mg_srv_gen == runs_gen &&
            __atomic_load_n(&mg_mb->alive, __ATOMIC_ACQUIRE))
            return true;
        server_quit_sync();
        mg_lane_init();
        if (!mg_mb) {
            if (hipHostMalloc((void **)&mg_mb, sizeof(MgMailbox)) != hipSuccess) {
                mg_persist_enabled = false;
                return false;
            }
            memset((void *)mg_mb, 0, sizeof(MgMailbox));
        }
        if (!mg_srv_stream &&
            hipStreamCreateWithFlags(&mg_srv_stream, hipStreamNonBlocking) != hipSuccess) {
            mg_persist_enabled = false;
            return false;
        }
        mg_mb->quit = 0;
        mg_mb->alive = 0;
        mg_mb->req_seq = mg_srv_seq;
        mg_mb->done_seq = mg_srv_seq;
        launch_mg_server(dr, (int)runs.size(), mg_mb, mg_dout, mg_srv_stream);
        auto t0 = std::chrono::steady_clock::now();
        while (!__atomic_load_n(&mg_mb->alive, __ATOMIC_ACQUIRE)) {
            if (std::chrono::duration_cast<std::chrono::milliseconds>(
                    std::chrono::steady_clock::now() - t0)
                    .count() > 100) {
                __atomic_store_n(&mg_mb->quit, 1u, __ATOMIC_RELEASE);
                (void)hipStreamSynchronize(mg_srv_stream);
                mg_persist_enabled = false;
                return false;
            }
        }
        mg_srv_gen = runs_gen;
        return true;
    }
    /* pending fused count scan (rrdb_scan_count_begin/finish): buffers are
     * plain hipMallocs so interleaved reads/compactions cannot reclaim them */
    struct PendingScanCount {
        bool active = false, trivial = false;
        int64_t trivial_count = 0;
        CompactStatsDev *d_stats = nullptr;
        uint64_t *d_anch = nullptr, *d_lo = nullptr, *d_hi = nullptr;
        uint8_t *d_start = nullptr; /* pooled request-bytes slab */
        /* pooled capacities: buffers persist across calls, realloc only on
         * growth (8 hipMallocs per begin measurably taxed the 16-partition
         * fan-out on slow-host boxes) */
        uint64_t anch_cap = 0, lo_cap = 0, bytes_cap = 0;
    } pend_scan;
    uint64_t next_seq_floor = 0;
    /* user ops: host + device */
    std::vector<HostOp> host_ops;
    DevOp *d_ops = nullptr;
    DevRule *d_rules = nullptr;
    uint8_t *d_pats = nullptr;
    int n_ops = 0;
    /* write path (§8(f)1): host memtable, newest write per key wins */
    std::map<std::string, std::tuple<std::string /*encoded value*/, uint64_t /*seq*/,
                                     int /*kind*/>> memtable;
    std::unordered_map<int64_t, HipScanCtx *> ctxs;
    int64_t next_ctx_id = 0;
    std::mutex mu;
    std::unordered_map<std::string, double> phase_ms;
    int emit_mode = 2; /* 2 = chunked (default; 2.8 TB/s on the copy probe),
                          0 = rank-major waves, 1 = input-major waves
                          (env "engine.emit_mode": chunked|rank|input) */
    int rank_mode = 4; /* 4 = group-streaming rank (default; anchor-key
                          groups partition the runs disjointly, each staged
                          through LDS exactly once; falls back to mode 0 per
                          compact when runs are not word-probe eligible),
                          3 = r01 windowed LDS-staged tail-word rank,
                          0 = global searches + bound-table narrowing,
                          1 = LDS-staged full-key block rank
                          (env "engine.rank_mode": grp|ldst|global|lds) */
    int bt_shift = 5;  /* bound-table block = 1<<bt_shift records (env
                          "engine.bt_shift") */
    int grp_blocks = 3584; /* group-rank grid cap (env "engine.grp_blocks");
                              grid-stride beats one-block-per-group by ~24%
                              (workgroup setup/LDS churn) */

    void activate() { HIP_OK(hipSetDevice(device)); }

    /* tail-word rank eligibility (shared by the grp and ldst modes): every
     * run in single-word probe mode with one shared stride and one shared
     * cross-run first-(fk-8)-byte prefix, so tail-word compares decide
     * cross-run order exactly */
    bool word_eligible()
    {
        if (runs.size() < 2)
            return false;
        if (ldst_elig_cache >= 0)
            return ldst_elig_cache != 0;
        ldst_elig_cache = 0;
        uint32_t fk = runs[0].fixed_klen;
        if (fk < 8)
            return false;
        for (auto &rr : runs)
            if (!rr.tails || rr.fixed_klen != fk || rr.n == 0)
                return false;
        if (fk > 8) {
            uint8_t p0[32], pi[32];
            uint64_t pl = fk - 8 < 32 ? fk - 8 : 32;
            HIP_OK(hipMemcpy(p0, runs[0].keys, pl, hipMemcpyDeviceToHost));
            for (size_t ri = 1; ri < runs.size(); ri++) {
                HIP_OK(hipMemcpy(pi, runs[ri].keys, pl, hipMemcpyDeviceToHost));
                if (memcmp(p0, pi, pl) != 0)
                    return false;
            }
        }
        ldst_elig_cache = 1;
        return true;
    }
    bool ldst_eligible() { return rank_mode == 3 && word_eligible(); }
    bool grp_eligible()
    {
        return rank_mode == 4 && runs.size() <= LDST_MAXR && word_eligible();
    }

    void free_run(RunBuf &r)
    {
        (void)hipFree(r.keys);
        (void)hipFree(r.koff);
        (void)hipFree(r.vals);
        (void)hipFree(r.voff);
        (void)hipFree(r.sk);
        if (r.bloom)
            (void)hipFree(r.bloom);
        if (r.tails)
            (void)hipFree(r.tails);
        if (r.pfx_bloom)
            (void)hipFree(r.pfx_bloom);
        if (r.meta)
            (void)hipFree(r.meta);
        r = RunBuf();
    }

    /* packed 8B-strided probe words for the single-word search mode */
    void build_tails(RunBuf &r)
    {
        if (r.n == 0 || r.fixed_klen < 8 || r.lcp_exact + 8 < r.fixed_klen)
            return;
        HIP_OK(hipMalloc(&r.tails, r.n * 8));
        launch_build_tails(r.keys, r.fixed_klen, r.n, r.tails, stream);
        HIP_OK(hipStreamSynchronize(stream));
    }

    /* disposition column (expire<<32|kind): one 8B gather replaces the
     * sk + voff + value-header chain in filter/scan-state evaluation */
    void build_meta(RunBuf &r)
    {
        if (r.n == 0)
            return;
        HIP_OK(hipMalloc(&r.meta, r.n * 8));
        launch_build_meta(r.vals, r.voff, r.fixed_vlen, r.sk, r.n, data_version, r.meta,
                          stream);
        HIP_OK(hipStreamSynchronize(stream));
    }

    /* §8(f)3: ~10 bits/key blocked bloom, built once per run on device */
    void build_bloom(RunBuf &r)
    {
        if (!bloom_enabled || r.n == 0)
            return;
        uint64_t n_blocks = (r.n * 10 + 511) / 512;
        if (n_blocks == 0)
            n_blocks = 1;
        HIP_OK(hipMalloc(&r.bloom, n_blocks * 64));
        HIP_OK(hipMemsetAsync(r.bloom, 0, n_blocks * 64, stream));
        DevRun dr{};
        dr.keys = r.keys;
        dr.koff = r.koff;
        dr.vals = r.vals;
        dr.voff = r.voff;
        dr.sk = r.sk;
        dr.n = r.n;
        launch_bloom_build(dr, r.bloom, n_blocks, stream);
        /* §8(f)3 second half: hashkey-prefix bloom (the reference's
         * prefix-extractor filter, _init.cpp:816-841, hashkey_transform.h) */
        HIP_OK(hipMalloc(&r.pfx_bloom, n_blocks * 64));
        HIP_OK(hipMemsetAsync(r.pfx_bloom, 0, n_blocks * 64, stream));
        launch_bloom_pfx_build(dr, r.pfx_bloom, n_blocks, stream);
        HIP_OK(hipStreamSynchronize(stream));
        r.bloom_blocks = n_blocks;
        r.pfx_bloom_blocks = n_blocks;
    }

    DevRun *dev_runs()
    {
        if (d_runs_dirty) {
            server_quit_sync(); /* the resident kernel holds the old table */
            if (d_runs)
                (void)hipFree(d_runs);
            std::vector<DevRun> h(runs.size() ? runs.size() : 1);
            for (size_t i = 0; i < runs.size(); i++) {
                DevRun d{};
                d.keys = runs[i].keys;
                d.koff = runs[i].koff;
                d.vals = runs[i].vals;
                d.voff = runs[i].voff;
                d.sk = runs[i].sk;
                d.n = runs[i].n;
                d.bloom = runs[i].bloom;
                d.bloom_blocks = runs[i].bloom_blocks;
                d.fixed_klen = runs[i].fixed_klen;
                d.pfx_skip = runs[i].pfx_skip;
                d.lcp_exact = runs[i].lcp_exact;
                d.tails = runs[i].tails;
                d.pfx_bloom = runs[i].pfx_bloom;
                d.pfx_bloom_blocks = runs[i].pfx_bloom_blocks;
                d.meta = runs[i].meta;
                d.fixed_vlen = runs[i].fixed_vlen;
                h[i] = d;
            }
            HIP_OK(hipMalloc(&d_runs, h.size() * sizeof(DevRun)));
            HIP_OK(hipMemcpy(d_runs, h.data(), h.size() * sizeof(DevRun), hipMemcpyHostToDevice));
            d_runs_dirty = false;
            runs_gen++;
        }
        return d_runs;
    }

    void upload_ops()
    {
        if (d_ops)
            (void)hipFree(d_ops), d_ops = nullptr;
        if (d_rules)
            (void)hipFree(d_rules), d_rules = nullptr;
        if (d_pats)
            (void)hipFree(d_pats), d_pats = nullptr;
        n_ops = (int)host_ops.size();
        if (n_ops == 0)
            return;
        std::vector<DevOp> ops;
        std::vector<DevRule> rules;
        std::string pats;
        for (auto &op : host_ops) {
            DevOp d;
            d.type = op.type;
            d.ut_type = op.ut_type;
            d.ut_value = op.ut_value;
            d.rule_off = (int32_t)rules.size();
            d.n_rules = (int32_t)op.rules.size();
            for (auto &r : op.rules) {
                DevRule dr;
                dr.type = r.type;
                dr.match_type = r.match_type;
                dr.start_ttl = r.start_ttl;
                dr.stop_ttl = r.stop_ttl;
                dr.pat_off = (uint32_t)pats.size();
                dr.pat_len = (uint32_t)r.pattern.size();
                pats += r.pattern;
                rules.push_back(dr);
            }
            ops.push_back(d);
        }
        HIP_OK(hipMalloc(&d_ops, ops.size() * sizeof(DevOp)));
        HIP_OK(hipMemcpy(d_ops, ops.data(), ops.size() * sizeof(DevOp), hipMemcpyHostToDevice));
        HIP_OK(hipMalloc(&d_rules, rules.size() * sizeof(DevRule)));
        HIP_OK(hipMemcpy(d_rules, rules.data(), rules.size() * sizeof(DevRule),
                         hipMemcpyHostToDevice));
        HIP_OK(hipMalloc(&d_pats, pats.size() ? pats.size() : 1));
        if (!pats.empty())
            HIP_OK(hipMemcpy(d_pats, pats.data(), pats.size(), hipMemcpyHostToDevice));
    }

    uint8_t *upload_bytes(const void *p, uint64_t n)
    {
        uint8_t *d = nullptr;
        HIP_OK(hipMalloc(&d, n ? n : 1));
        if (n)
            HIP_OK(hipMemcpy(d, p, n, hipMemcpyHostToDevice));
        return d;
    }

    /* ingest-path variant: a full device returns null (the caller unwinds
     * and reports kIOError, like rocksdb's background write errors) instead
     * of aborting the process */
    uint8_t *upload_bytes_try(const void *p, uint64_t n)
    {
        uint8_t *d = nullptr;
        if (hipMalloc(&d, n ? n : 1) != hipSuccess)
            return nullptr;
        if (n && hipMemcpy(d, p, n, hipMemcpyHostToDevice) != hipSuccess) {
            (void)hipFree(d);
            return nullptr;
        }
        return d;
    }

    /* engine-owned persistent scratch arena for per-operation temporaries.
     * Bump-allocated; reset at the start of each C-ABI operation (no live
     * scratch crosses operations — persistent data uses plain hipMalloc).
     * Grows to the high-water mark once and then never allocates again —
     * per-op hipMalloc/hipFree (and mempool behavior variance across
     * driver configs) was a large, box-dependent share of step time. */
    std::vector<std::pair<uint8_t *, size_t>> sblocks;
    size_t sblock_i = 0, s_off = 0;
    /* while a split compaction is pending its phase-1 buffers live at the
     * arena head; reads between begin/finish reset only to this watermark
     * (unguarded resets let reads reuse the pending buffers) */
    bool arena_pinned = false;
    size_t pin_block = 0, pin_off = 0;
    void scratch_reset()
    {
        sblock_i = arena_pinned ? pin_block : 0;
        s_off = arena_pinned ? pin_off : 0;
    }
    void arena_pin()
    {
        arena_pinned = true;
        pin_block = sblock_i;
        pin_off = s_off;
    }
    void arena_unpin() { arena_pinned = false; }
    /* reclaim parked scanner contexts older than 5 minutes (the reference
     * drops unused contexts after std::chrono::minutes(5),
     * pegasus_server_impl.cpp:1381-1387; a used context re-parks under a
     * fresh handle with a fresh timer, as ours does).  Driven by the
     * caller-supplied epoch clock so tests are deterministic. */
    void gc_ctxs(uint32_t epoch_now)
    {
        for (auto it = ctxs.begin(); it != ctxs.end();) {
            if (epoch_now > it->second->parked_at &&
                epoch_now - it->second->parked_at > 300) {
                delete it->second;
                it = ctxs.erase(it);
            } else {
                ++it;
            }
        }
    }
    /* compaction rebuilt the run list: every parked view indexes freed runs.
     * Reclaim them all; a later scan_next returns kNotFound, the reference's
     * expired-context behavior (on_scan:1539-1541). */
    void invalidate_ctxs()
    {
        for (auto &p : ctxs)
            delete p.second;
        ctxs.clear();
    }
    template <typename T> T *talloc(uint64_t n_bytes)
    {
        uint64_t n = (n_bytes + 255) & ~255ull;
        if (n == 0)
            n = 256;
        for (;;) {
            if (sblock_i < sblocks.size()) {
                if (s_off + n <= sblocks[sblock_i].second)
                    break;
                sblock_i++; /* tail of this block is wasted until reset */
                s_off = 0;
                continue;
            }
            /* grow: blocks are kept forever (an op-time hipMalloc happens at
             * most a handful of times per engine lifetime; consolidation-
             * style free+realloc cost ~100ms per engine at 7M-record
             * partitions and poisoned the first timed step) */
            size_t tot = 0;
            for (auto &b : sblocks)
                tot += b.second;
            size_t want = std::max<size_t>({(size_t)n, tot, (size_t)(256ull << 20)});
            uint8_t *pb = nullptr;
            HIP_OK(hipMalloc(&pb, want));
            sblocks.push_back({pb, want});
            s_off = 0;
        }
        T *out = (T *)(sblocks[sblock_i].first + s_off);
        s_off += n;
        return out;
    }
    void tfree(void *) {} /* arena memory is reclaimed at scratch_reset */
    /* build the search bound table for windows lo/hi (BT_SHIFT=8 blocks);
     * returns device row-offset + table pointers (arena memory) */
    void build_bound_table_level(DevRun *dr, int R, const std::vector<uint64_t> &lo,
                                 const std::vector<uint64_t> &hi, const uint64_t *d_lo,
                                 const uint64_t *d_hi, int shift, const uint64_t *cbt_off,
                                 const uint64_t *cbt, int cbt_shift, uint64_t **out_bt_off,
                                 uint64_t **out_bt)
    {
        std::vector<uint64_t> bt_off(R + 1);
        uint64_t rows = 0;
        for (int r = 0; r < R; r++) {
            bt_off[r] = rows;
            uint64_t w = hi[r] - lo[r];
            rows += (w >> shift) + 2; /* P_r + sentinel */
        }
        bt_off[R] = rows;
        uint64_t *d_bt_off = (uint64_t *)upload_tmp(bt_off.data(), (R + 1) * 8);
        uint64_t *d_bt = talloc<uint64_t>(rows * (uint64_t)R * 8);
        launch_bound_table(dr, R, d_lo, d_hi, d_bt_off, rows, shift, d_bt, cbt_off, cbt,
                           cbt_shift, stream);
        *out_bt_off = d_bt_off;
        *out_bt = d_bt;
    }

    void build_bound_table(DevRun *dr, int R, const std::vector<uint64_t> &lo,
                           const std::vector<uint64_t> &hi, const uint64_t *d_lo,
                           const uint64_t *d_hi, uint64_t **out_bt_off, uint64_t **out_bt)
    {
        uint64_t wmax = 0;
        for (int r = 0; r < R; r++)
            wmax = std::max(wmax, hi[r] - lo[r]);
        uint64_t *c_off = nullptr, *c_bt = nullptr;
        int cshift = bt_shift + 6;
        if ((wmax >> cshift) > 8) /* coarse level pays only at scale */
            build_bound_table_level(dr, R, lo, hi, d_lo, d_hi, cshift, nullptr, nullptr, 0,
                                    &c_off, &c_bt);
        build_bound_table_level(dr, R, lo, hi, d_lo, d_hi, bt_shift, c_off, c_bt, cshift,
                                out_bt_off, out_bt);
    }

    uint64_t *psum_scratch(uint64_t n)
    {
        return talloc<uint64_t>(psum_scratch_elems(n) * 8);
    }
    uint8_t *upload_tmp(const void *p, uint64_t n)
    {
        uint8_t *d = talloc<uint8_t>(n);
        if (n)
            HIP_OK(hipMemcpyAsync(d, p, n, hipMemcpyHostToDevice, stream));
        return d;
    }

    /* pinned host staging (reused; grown to high-water) — unpinned D2H runs
     * at a fraction of PCIe rate, so big result copies stage through here */
    uint8_t *pinned = nullptr;
    size_t pinned_cap = 0;
    uint8_t *pinned_buf(size_t n)
    {
        if (n > pinned_cap) {
            if (pinned)
                (void)hipHostFree(pinned);
            pinned_cap = std::max<size_t>(n, pinned_cap * 2);
            HIP_OK(hipHostMalloc((void **)&pinned, pinned_cap));
        }
        return pinned;
    }
    /* D2H via pinned staging into an arbitrary host destination */
    void d2h(void *dst, const void *src, size_t n)
    {
        if (n == 0)
            return;
        if (n <= (64u << 10)) { /* small copies: direct */
            HIP_OK(hipMemcpyAsync(dst, src, n, hipMemcpyDeviceToHost, stream));
            HIP_OK(hipStreamSynchronize(stream));
            return;
        }
        uint8_t *pb = pinned_buf(n);
        HIP_OK(hipMemcpyAsync(pb, src, n, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipStreamSynchronize(stream));
        memcpy(dst, pb, n);
    }

    /* build the visible view for [start, stop_excl); returns device array
     * (caller frees) + count */
    /* group-streaming rank: pick the anchor run (largest window) and stride,
     * build the anchor table (arena memory).  Returns n_groups. */
    uint64_t build_anchors(DevRun *dr, int R, const std::vector<uint64_t> &lo,
                           const std::vector<uint64_t> &hi, const uint64_t *d_lo,
                           const uint64_t *d_hi, uint64_t **out_anch)
    {
        int q0 = 0;
        uint64_t wmax = 0;
        for (int r = 0; r < R; r++)
            if (hi[r] - lo[r] > wmax) {
                wmax = hi[r] - lo[r];
                q0 = r;
            }
        int gs = 4;
        while ((1ull << (gs + 1)) * (uint64_t)R <= GRP_TARGET && gs < 8)
            gs++;
        uint64_t n_groups = (wmax + (1ull << gs) - 1) >> gs;
        if (n_groups == 0)
            n_groups = 1;
        uint64_t *d_anch = talloc<uint64_t>((n_groups + 1) * R * 8);
        launch_anchor_rows(dr, R, q0, d_lo, d_hi, gs, n_groups, d_anch, stream);
        *out_anch = d_anch;
        return n_groups;
    }

    void build_view(const std::string *start, const std::string *stop_excl, uint64_t **out_view,
                    uint64_t *out_n)
    {
        int R = (int)runs.size();
        *out_view = nullptr;
        *out_n = 0;
        if (R == 0)
            return;
        DevRun *dr = dev_runs();
        uint64_t *d_lo = talloc<uint64_t>(R * 8);
        uint64_t *d_hi = talloc<uint64_t>(R * 8);
        uint8_t *d_start = start ? upload_tmp(start->data(), start->size()) : nullptr;
        uint8_t *d_stop = stop_excl ? upload_tmp(stop_excl->data(), stop_excl->size()) : nullptr;
        launch_bounds(dr, R, d_start, start ? start->size() : 0, d_lo, 0, stream);
        /* null stop = unbounded: upper flag makes the null-key case yield n */
        launch_bounds(dr, R, d_stop, stop_excl ? stop_excl->size() : 0, d_hi,
                      stop_excl ? 0 : 1, stream);
        std::vector<uint64_t> lo(R), hi(R), wprefix(R + 1);
        HIP_OK(hipMemcpyAsync(lo.data(), d_lo, R * 8, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipMemcpyAsync(hi.data(), d_hi, R * 8, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipStreamSynchronize(stream));
        uint64_t total = 0;
        for (int r = 0; r < R; r++) {
            if (hi[r] < lo[r])
                hi[r] = lo[r];
            wprefix[r] = total;
            total += hi[r] - lo[r];
        }
        wprefix[R] = total;
        tfree(d_start);
        tfree(d_stop);
        if (total == 0) {
            tfree(d_lo);
            tfree(d_hi);
            return;
        }
        /* re-upload corrected hi + wprefix */
        HIP_OK(hipMemcpy(d_hi, hi.data(), R * 8, hipMemcpyHostToDevice));
        uint64_t *d_wp = (uint64_t *)upload_tmp(wprefix.data(), (R + 1) * 8);
        uint64_t *d_order = talloc<uint64_t>(total * 8);
        uint8_t *d_shadow = talloc<uint8_t>(total);
        uint64_t *d_flags = talloc<uint64_t>(total * 8);
        uint64_t *d_pos = talloc<uint64_t>(total * 8);
        if (R > 1 && grp_eligible()) {
            uint64_t *d_anch = nullptr;
            uint64_t n_groups = build_anchors(dr, R, lo, hi, d_lo, d_hi, &d_anch);
            launch_rank_grp_view(dr, R, d_lo, d_anch, n_groups, d_order, d_shadow, stream);
        } else {
            uint64_t *d_bt_off = nullptr, *d_bt = nullptr;
            if (R > 1 && total > 100000)
                build_bound_table(dr, R, lo, hi, d_lo, d_hi, &d_bt_off, &d_bt);
            if (ldst_eligible())
                launch_rank_ldst(dr, R, d_lo, d_hi, d_wp, total, d_order, d_shadow, d_bt_off,
                                 d_bt, bt_shift, stream);
            else
                launch_rank(dr, R, d_lo, d_hi, d_wp, total, d_order, d_shadow, d_bt_off, d_bt,
                            bt_shift, stream);
        }
        launch_visible(dr, d_order, d_shadow, total, d_flags, stream);
        launch_psum(d_flags, d_pos, total, psum_scratch(total), stream);
        uint64_t lastp = 0, lastf = 0;
        HIP_OK(hipMemcpyAsync(&lastp, d_pos + total - 1, 8, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipMemcpyAsync(&lastf, d_flags + total - 1, 8, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipStreamSynchronize(stream));
        uint64_t nv = lastp + lastf;
        uint64_t *d_view = nullptr;
        if (nv) {
            HIP_OK(hipMalloc(&d_view, nv * 8));
            launch_gather(d_order, d_flags, d_pos, total, d_view, stream);
            HIP_OK(hipStreamSynchronize(stream));
        }
        tfree(d_lo);
        tfree(d_hi);
        tfree(d_wp);
        tfree(d_order);
        tfree(d_shadow);
        tfree(d_flags);
        tfree(d_pos);
        HIP_OK(hipStreamSynchronize(stream));
        *out_view = d_view;
        *out_n = nv;
    }
};

std::once_flag g_init_once;

} // namespace

static int32_t engine_flush(HipEngine *e); /* write path, defined below */

/* internal: upload a prepared sorted run (validation already established);
 * kIOError (with the partial run unwound) when the device is full */
static int32_t ingest_prepared(HipEngine *e, const std::string &keys,
                               const std::vector<uint64_t> &koff, const std::string &vals,
                               const std::vector<uint64_t> &voff,
                               const std::vector<uint64_t> &sk)
{
    RunBuf r;
    r.n = sk.size();
    r.keys = e->upload_bytes_try(keys.data(), keys.size());
    r.koff = (uint64_t *)e->upload_bytes_try(koff.data(), koff.size() * 8);
    r.vals = e->upload_bytes_try(vals.data(), vals.size());
    r.voff = (uint64_t *)e->upload_bytes_try(voff.data(), voff.size() * 8);
    r.sk = (uint64_t *)e->upload_bytes_try(sk.data(), sk.size() * 8);
    if (!r.keys || !r.koff || !r.vals || !r.voff || !r.sk) {
        e->free_run(r);
        return RRDB_IO_ERROR;
    }
    r.fixed_klen = detect_fixed_klen(koff.data(), r.n);
    r.fixed_vlen = detect_fixed_vlen(voff.data(), r.n);
    r.fixed_vlen_put = detect_fixed_vlen_put(voff.data(), sk.data(), r.n);
    set_pfx(lcp_exact32((const uint8_t *)keys.data(), koff[1] - koff[0],
                        (const uint8_t *)keys.data() + koff[r.n - 1],
                        koff[r.n] - koff[r.n - 1]),
            &r.pfx_skip, &r.lcp_exact);
    e->build_tails(r);
    e->build_meta(r);
    e->build_bloom(r);
    e->runs.push_back(r);
    e->d_runs_dirty = true;
    e->ldst_elig_cache = -1;
    return RRDB_OK;
}

extern "C" {

const char *rrdb_backend(void) { return "hip-gfx950"; }

double rrdb_phase_ms(void *h, const char *phase)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    e->activate();
    e->resolve_phase_events();
    auto it = e->phase_ms.find(phase);
    return it == e->phase_ms.end() ? -1.0 : it->second;
}

void *rrdb_open(int32_t app_id, int32_t pidx, int32_t gpu_id)
{
    if (gpu_id < 0)
        return nullptr; /* GPU engine requires a device */
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || gpu_id >= ndev) {
        fprintf(stderr, "rrdb-hip: no HIP device %d (count=%d) — the product engine has no CPU "
                        "fallback\n",
                gpu_id, ndev);
        return nullptr;
    }
    auto *e = new HipEngine();
    e->app_id = app_id;
    e->pidx = pidx;
    e->device = gpu_id;
    e->activate();
    /* spin completion: interrupt-driven stream waits cost ~10-20us each on
     * the small-op serving path (ignored if the runtime refuses) */
    (void)hipSetDeviceFlags(hipDeviceScheduleSpin);
    std::call_once(g_init_once, [] {
        build_crc64_table();
        launch_crc64_table_init(host_crc64_table);
    });
    HIP_OK(hipStreamCreate(&e->stream));
    {
        hipMemPool_t pool;
        if (hipDeviceGetDefaultMemPool(&pool, gpu_id) == hipSuccess) {
            uint64_t thr = UINT64_MAX;
            (void)hipMemPoolSetAttribute(pool, hipMemPoolAttrReleaseThreshold, &thr);
        }
    }
    return e;
}

void rrdb_close(void *h)
{
    auto *e = (HipEngine *)h;
    if (!e)
        return;
    e->activate();
    e->server_quit_sync(); /* retire the resident serving kernel first */
    for (auto &kv : e->ctxs)
        delete kv.second;
    e->ctxs.clear();
    {
        auto &ps = e->pend_scan;
        for (void *p : {(void *)ps.d_stats, (void *)ps.d_anch, (void *)ps.d_lo,
                        (void *)ps.d_hi, (void *)ps.d_start})
            if (p)
                (void)hipFree(p);
    }
    for (auto &r : e->runs)
        e->free_run(r);
    if (e->d_runs)
        (void)hipFree(e->d_runs);
    for (auto &b : e->sblocks)
        (void)hipFree(b.first);
    if (e->pinned)
        (void)hipHostFree(e->pinned);
    if (e->pend_sizes)
        (void)hipHostFree(e->pend_sizes);
    if (e->pend_stats_h)
        (void)hipHostFree(e->pend_stats_h);
    for (auto &x : e->tev)
        if (x)
            (void)hipEventDestroy(x);
    e->server_quit_sync();
    if (e->mg_mb)
        (void)hipHostFree((void *)e->mg_mb);
    if (e->mg_srv_stream)
        (void)hipStreamDestroy(e->mg_srv_stream);
    if (e->mg_graph)
        (void)hipGraphExecDestroy(e->mg_graph);
    if (e->mg_capture_stream)
        (void)hipStreamDestroy(e->mg_capture_stream);
    if (e->mg_hin) {
        (void)hipHostFree(e->mg_hin);
        (void)hipFree(e->mg_din);
        (void)hipHostFree(e->mg_hout);
        (void)hipFree(e->mg_dout);
    }
    if (e->d_ops)
        (void)hipFree(e->d_ops);
    if (e->d_rules)
        (void)hipFree(e->d_rules);
    if (e->d_pats)
        (void)hipFree(e->d_pats);
    (void)hipStreamDestroy(e->stream);
    delete e;
}

int32_t rrdb_set_partition_version(void *h, int32_t pv)
{
    ((HipEngine *)h)->partition_version = pv;
    return RRDB_OK;
}

int32_t rrdb_set_envs(void *h, const char *const *keys, const char *const *values, int32_t n)
{
    auto *e = (HipEngine *)h;
    e->activate();
    for (int32_t i = 0; i < n; i++) {
        std::string k = keys[i], v = values[i];
        if (k == "default_ttl") {
            long long t = atoll(v.c_str());
            e->default_ttl = t > 0 ? (uint32_t)t : 0;
        } else if (k == "user_specified_compaction") {
            std::vector<HostOp> ops;
            if (!parse_user_ops(v, ops))
                ops.clear(); /* invalid json -> no ops (cpp:165-169) */
            e->host_ops = std::move(ops);
            e->upload_ops();
        } else if (k == "replica.split.validate_partition_hash") {
            e->validate_hash = (v == "true");
        } else if (k == "manual_compact.disabled") {
            e->manual_compact_disabled = (v == "true");
        } else if (k == "pegasus.data_version") {
            uint32_t nv = (uint32_t)atoi(v.c_str());
            if (nv != e->data_version) {
                e->data_version = nv;
                /* the meta column bakes in the value-header layout */
                e->activate();
                for (auto &r : e->runs) {
                    if (r.meta) {
                        (void)hipFree(r.meta);
                        r.meta = nullptr;
                    }
                    e->build_meta(r);
                }
                e->d_runs_dirty = true;
            }
        } else if (k == "replica.rocksdb_iteration_threshold_time_ms") {
            e->iter_time_ms = (uint64_t)atoll(v.c_str());
        } else if (k == "rocksdb.max_iteration_count") {
            e->max_iter_count = (uint32_t)atoll(v.c_str());
        } else if (k == "rocksdb.multi_get_max_iteration_count") {
            e->mg_max_iter_count = (uint32_t)atoll(v.c_str());
        } else if (k == "rocksdb.multi_get_max_iteration_size") {
            e->mg_max_iter_size = (uint64_t)atoll(v.c_str());
        } else if (k == "rocksdb.filter_type") {
            e->bloom_enabled = (v != "none"); /* common/prefix -> full-key bloom */
        } else if (k == "engine.rank_mode") {
            e->rank_mode =
                (v == "lds") ? 1 : (v == "ldst" ? 3 : (v == "global" ? 0 : 4));
        } else if (k == "engine.grp_blocks") {
            int b = atoi(v.c_str());
            if (b >= 256 && b <= 65535)
                e->grp_blocks = b;
        } else if (k == "engine.bt_shift") {
            int s_ = atoi(v.c_str());
            if (s_ >= 4 && s_ <= 16)
                e->bt_shift = s_;
        } else if (k == "engine.mg_persist") {
            if (v == "off") {
                e->activate();
                e->server_quit_sync();
                e->mg_persist_mode = 0;
            } else if (v == "on") {
                e->mg_persist_mode = 2;
                e->mg_persist_enabled = true;
            } else { /* "auto" */
                e->mg_persist_mode = 1;
                e->mg_persist_enabled = true;
            }
        } else if (k == "engine.mg_graph") {
            e->mg_graph_enabled = (v != "off");
            if (!e->mg_graph_enabled && e->mg_graph) {
                e->activate();
                (void)hipGraphExecDestroy(e->mg_graph);
                e->mg_graph = nullptr;
            }
        } else if (k == "engine.emit_mode") {
            e->emit_mode = (v == "input") ? 1 : (v == "rank" ? 0 : 2);
        }
    }
    return RRDB_OK;
}

int32_t rrdb_ingest_run(void *h, const uint8_t *keys, const uint64_t *key_offs,
                        const uint8_t *values, const uint64_t *val_offs, const uint64_t *seq_kind,
                        uint64_t n)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    engine_flush(e); /* buffered writes are older than this run */
    if (n == 0)
        return RRDB_OK;
    /* validate on host (data arrives host-side anyway) */
    if (key_offs[0] != 0 || key_offs[1] == 0)
        return RRDB_INVALID_ARGUMENT; /* keys must be nonempty, offsets 0-based */
    for (uint64_t i = 0; i < n; i++) {
        if ((seq_kind[i] >> 1) < e->next_seq_floor)
            return RRDB_INVALID_ARGUMENT;
        if (i > 0) {
            uint64_t al = key_offs[i] - key_offs[i - 1], bl = key_offs[i + 1] - key_offs[i];
            const uint8_t *a = keys + key_offs[i - 1];
            const uint8_t *b = keys + key_offs[i];
            int c = memcmp(a, b, std::min(al, bl));
            if (c > 0 || (c == 0 && al >= bl))
                return RRDB_INVALID_ARGUMENT;
        }
    }
    e->activate();
    RunBuf r;
    r.n = n;
    r.keys = e->upload_bytes_try(keys, key_offs[n]);
    r.koff = (uint64_t *)e->upload_bytes_try(key_offs, (n + 1) * 8);
    r.vals = e->upload_bytes_try(values, val_offs[n]);
    r.voff = (uint64_t *)e->upload_bytes_try(val_offs, (n + 1) * 8);
    r.sk = (uint64_t *)e->upload_bytes_try(seq_kind, n * 8);
    if (!r.keys || !r.koff || !r.vals || !r.voff || !r.sk) {
        e->free_run(r);
        return RRDB_IO_ERROR;
    }
    r.fixed_klen = detect_fixed_klen(key_offs, n);
    r.fixed_vlen = detect_fixed_vlen(val_offs, n);
    r.fixed_vlen_put = detect_fixed_vlen_put(val_offs, seq_kind, n);
    set_pfx(lcp_exact32(keys + key_offs[0], key_offs[1] - key_offs[0],
                        keys + key_offs[n - 1], key_offs[n] - key_offs[n - 1]),
            &r.pfx_skip, &r.lcp_exact);
    uint64_t mx = 0;
    for (uint64_t i = 0; i < n; i++)
        mx = std::max(mx, seq_kind[i] >> 1);
    e->next_seq_floor = mx + 1;
    e->build_tails(r);
    e->build_meta(r);
    e->build_bloom(r);
    e->runs.push_back(r);
    e->d_runs_dirty = true;
    e->ldst_elig_cache = -1;
    return RRDB_OK;
}

uint64_t rrdb_num_runs(void *h) { return ((HipEngine *)h)->runs.size(); }
uint64_t rrdb_num_records(void *h)
{
    auto *e = (HipEngine *)h;
    uint64_t t = 0;
    for (auto &r : e->runs)
        t += r.n;
    return t;
}

void rrdb_free_result(rrdb_result *r)
{
    if (r && r->_arena) {
        delete (Arena *)r->_arena;
        r->_arena = nullptr;
    }
}

/* ---- batched point lookup core (on_get / on_batch_get / DB::MultiGet) ---- */
static int32_t get_core(HipEngine *e, uint64_t nq, const uint8_t *keys, const uint64_t *key_offs,
                        uint32_t epoch_now, std::vector<int32_t> &status,
                        std::vector<std::pair<const uint8_t *, uint64_t>> &vals, Arena *a)
{
    e->activate();
    e->scratch_reset();
    int R = (int)e->runs.size();
    status.assign(nq, RRDB_NOT_FOUND);
    vals.assign(nq, {nullptr, 0});
    if (R == 0)
        return RRDB_OK;
    DevRun *dr = e->dev_runs();
    uint8_t *d_keys = e->upload_tmp(keys, key_offs[nq]);
    uint64_t *d_offs = (uint64_t *)e->upload_tmp(key_offs, (nq + 1) * 8);
    int32_t *d_status = e->talloc<int32_t>(nq * 4);
    uint64_t *d_hit = e->talloc<uint64_t>(nq * 8);
    uint64_t *d_ulen = e->talloc<uint64_t>(nq * 8);
    uint64_t *d_voffs = e->talloc<uint64_t>((nq + 1) * 8);
    hipEvent_t gev[2];
    HIP_OK(hipEventCreate(&gev[0]));
    HIP_OK(hipEventCreate(&gev[1]));
    HIP_OK(hipEventRecord(gev[0], e->stream));
    launch_get(dr, R, d_keys, d_offs, nq, epoch_now, e->data_version, d_status, d_hit, d_ulen,
               nullptr, e->stream);
    HIP_OK(hipEventRecord(gev[1], e->stream));
    launch_psum(d_ulen, d_voffs, nq, e->psum_scratch(nq), e->stream);
    uint64_t last_off = 0, last_len = 0;
    HIP_OK(hipMemcpyAsync(&last_off, d_voffs + nq - 1, 8, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(&last_len, d_ulen + nq - 1, 8, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(status.data(), d_status, nq * 4, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipStreamSynchronize(e->stream));
    {
        float ms;
        if (hipEventElapsedTime(&ms, gev[0], gev[1]) == hipSuccess)
            e->phase_ms["get_search"] = ms;
        (void)hipEventDestroy(gev[0]);
        (void)hipEventDestroy(gev[1]);
    }
    uint64_t total = last_off + last_len;
    std::vector<uint64_t> voffs(nq + 1);
    HIP_OK(hipMemcpy(voffs.data(), d_voffs, nq * 8, hipMemcpyDeviceToHost));
    voffs[nq] = total;
    uint8_t *blob = (uint8_t *)a->alloc(total);
    if (total) {
        uint8_t *d_blob = e->talloc<uint8_t>(total);
        launch_emit_values(dr, d_hit, d_status, nq, e->data_version, d_voffs, d_blob, e->stream);
        e->d2h(blob, d_blob, total);
        e->tfree(d_blob);
    }
    for (uint64_t i = 0; i < nq; i++)
        if (status[i] == RRDB_OK)
            vals[i] = {blob + voffs[i], voffs[i + 1] - voffs[i]};
    e->tfree(d_keys);
    e->tfree(d_offs);
    e->tfree(d_status);
    e->tfree(d_hit);
    e->tfree(d_ulen);
    e->tfree(d_voffs);
    return RRDB_OK;
}

int32_t rrdb_get(void *h, const uint8_t *key, uint64_t key_len, uint32_t epoch_now,
                 rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    engine_flush((HipEngine *)h); /* memtable visible to reads */
    Arena *a = result_init(out);
    uint64_t offs[2] = {0, key_len};
    std::vector<int32_t> status;
    std::vector<std::pair<const uint8_t *, uint64_t>> vals;
    get_core(e, 1, key, offs, epoch_now, status, vals, a);
    if (status[0] != RRDB_OK) {
        out->error = status[0];
        return out->error;
    }
    out->count = 1;
    out->keys = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
    out->values = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
    out->keys[0] = {nullptr, 0};
    out->values[0] = {(uint8_t *)vals[0].first, vals[0].second};
    out->error = RRDB_OK;
    return RRDB_OK;
}

int32_t rrdb_batch_get(void *h, uint64_t n_keys, const uint8_t *keys, const uint64_t *key_offs,
                       uint32_t epoch_now, rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    engine_flush((HipEngine *)h); /* memtable visible to reads */
    Arena *a = result_init(out);
    if (n_keys == 0) {
        out->error = RRDB_INVALID_ARGUMENT; /* on_batch_get:922-928 */
        return out->error;
    }
    std::vector<int32_t> status;
    std::vector<std::pair<const uint8_t *, uint64_t>> vals;
    get_core(e, n_keys, keys, key_offs, epoch_now, status, vals, a);
    out->keys = (rrdb_slice *)a->alloc(n_keys * sizeof(rrdb_slice));
    out->values = (rrdb_slice *)a->alloc(n_keys * sizeof(rrdb_slice));
    /* one block for every returned key (a per-key arena alloc was ~30ms at
     * 1M-key batches) */
    uint8_t *kblock = (uint8_t *)a->alloc(key_offs[n_keys]);
    uint64_t m = 0, kb = 0;
    for (uint64_t i = 0; i < n_keys; i++) {
        if (status[i] != RRDB_OK)
            continue; /* NotFound/expired skipped (on_batch_get:952-965) */
        uint64_t kl = key_offs[i + 1] - key_offs[i];
        memcpy(kblock + kb, keys + key_offs[i], kl);
        out->keys[m] = {kblock + kb, kl};
        out->values[m] = {(uint8_t *)vals[i].first, vals[i].second};
        kb += kl;
        m++;
    }
    out->count = m;
    out->error = RRDB_OK;
    return RRDB_OK;
}

int32_t rrdb_ttl(void *h, const uint8_t *key, uint64_t key_len, uint32_t epoch_now,
                 rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    engine_flush((HipEngine *)h); /* memtable visible to reads */
    ((HipEngine *)h)->scratch_reset();
    Arena *a = result_init(out);
    e->activate();
    int R = (int)e->runs.size();
    if (R == 0) {
        out->error = RRDB_NOT_FOUND;
        return out->error;
    }
    DevRun *dr = e->dev_runs();
    uint64_t offs[2] = {0, key_len};
    uint8_t *d_key = e->upload_tmp(key, key_len);
    uint64_t *d_offs = (uint64_t *)e->upload_tmp(offs, 16);
    int32_t *d_status = e->talloc<int32_t>(4);
    uint64_t *d_hit = e->talloc<uint64_t>(8);
    uint64_t *d_ulen = e->talloc<uint64_t>(8);
    uint32_t *d_expire = e->talloc<uint32_t>(4);
    launch_get(dr, R, d_key, d_offs, 1, epoch_now, e->data_version, d_status, d_hit, d_ulen,
               d_expire, e->stream);
    int32_t st;
    uint32_t expire;
    HIP_OK(hipMemcpyAsync(&st, d_status, 4, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(&expire, d_expire, 4, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipStreamSynchronize(e->stream));
    e->tfree(d_key);
    e->tfree(d_offs);
    e->tfree(d_status);
    e->tfree(d_hit);
    e->tfree(d_ulen);
    e->tfree(d_expire);
    (void)a;
    out->error = st;
    if (st == RRDB_OK)
        out->i64 = expire > 0 ? (int64_t)expire - (int64_t)epoch_now : -1; /* on_ttl:1135-1142 */
    return out->error;
}

/* ---- scan machinery shared by scan/sortkey_count/multi_get-range ---- */

struct BatchOut {
    uint64_t consumed = 0;  /* view entries consumed */
    uint64_t n_out = 0;     /* normal rows */
};

/* run one forward batch over ctx->d_view[cursor..]; emits into result */
static void scan_batch_gpu(HipEngine *e, HipScanCtx *c, uint32_t epoch_now, rrdb_result *out,
                           Arena *a)
{
    auto t_start = std::chrono::steady_clock::now();
    uint32_t batch_count = e->max_iter_count;
    if (c->batch_size > 0 && (uint32_t)c->batch_size < batch_count)
        batch_count = (uint32_t)c->batch_size;
    uint64_t remaining = c->view_n - c->cursor;
    uint64_t w = std::min<uint64_t>(remaining, e->max_iter_count);
    if (w == 0) {
        out->error = RRDB_OK;
        out->context_id = RRDB_SCAN_CONTEXT_ID_COMPLETED;
        if (c->only_return_count) {
            out->i64 = 0;
        } else {
            /* a normal zero-row batch returns allocated (empty) arrays —
             * only the empty-RANGE early return leaves them null (matches
             * the oracle's scan_batch, observable via expire_ts) */
            out->keys = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
            out->values = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
            if (c->return_expire_ts)
                out->expire_ts = (int32_t *)a->alloc(4);
        }
        return;
    }
    DevRun *dr = e->dev_runs();
    const uint64_t *d_win = c->d_view + c->cursor;
    ScanParams sp{};
    sp.epoch_now = epoch_now;
    sp.data_version = e->data_version;
    sp.pidx = e->pidx;
    sp.partition_version = e->partition_version;
    sp.validate_hash = (uint8_t)(c->validate_hash_req && e->validate_hash);
    sp.hk_ft = c->hk_ft;
    sp.sk_ft = c->sk_ft;
    sp.hk_pat = c->d_hk_pat;
    sp.hk_pat_len = c->hk_pat_len;
    sp.sk_pat = c->d_sk_pat;
    sp.sk_pat_len = c->sk_pat_len;
    sp.no_value = c->no_value;
    sp.hash_key_skip = 0;

    uint8_t *d_state = e->talloc<uint8_t>(w);
    uint64_t *d_ksz = e->talloc<uint64_t>(w * 8);
    uint64_t *d_vsz = e->talloc<uint64_t>(w * 8);
    uint64_t *d_flags = e->talloc<uint64_t>(w * 8);
    uint64_t *d_npos = e->talloc<uint64_t>(w * 8);
    uint64_t *d_cut = e->talloc<uint64_t>(16);
    hipEvent_t sev[4];
    for (auto &x : sev)
        HIP_OK(hipEventCreate(&x));
    HIP_OK(hipEventRecord(sev[0], e->stream));
    launch_scan_state(dr, d_win, w, sp, d_state, d_ksz, d_vsz, e->stream);
    HIP_OK(hipEventRecord(sev[1], e->stream));
    launch_normal_flags(d_state, w, d_flags, e->stream);
    launch_psum(d_flags, d_npos, w, e->psum_scratch(w), e->stream);
    launch_cutoff(d_npos, d_flags, w, batch_count, d_cut, e->stream);
    uint64_t cut[2];
    HIP_OK(hipMemcpyAsync(cut, d_cut, 16, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipStreamSynchronize(e->stream));
    uint64_t consumed = cut[0], n_out = cut[1];

    if (c->only_return_count) {
        out->i64 = (int64_t)n_out;
        out->count = 0;
    } else if (n_out == 0) {
        out->count = 0;
        out->keys = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
        out->values = (rrdb_slice *)a->alloc(sizeof(rrdb_slice));
        if (c->return_expire_ts)
            out->expire_ts = (int32_t *)a->alloc(4);
    } else {
        launch_cut_sizes(d_state, w, consumed, d_ksz, d_vsz, e->stream);
        uint64_t *d_koffs = e->talloc<uint64_t>(w * 8);
        uint64_t *d_voffs = e->talloc<uint64_t>(w * 8);
        launch_psum(d_ksz, d_koffs, w, e->psum_scratch(w), e->stream);
        launch_psum(d_vsz, d_voffs, w, e->psum_scratch(w), e->stream);
        uint64_t t[4] = {0, 0, 0, 0};
        HIP_OK(hipMemcpyAsync(&t[0], d_koffs + w - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[1], d_ksz + w - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[2], d_voffs + w - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[3], d_vsz + w - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        uint64_t kbytes = t[0] + t[1], vbytes = t[2] + t[3];
        uint8_t *d_kout, *d_vout;
        uint64_t *d_kooffs, *d_vooffs;
        int32_t *d_ets = nullptr;
        if (c->on_device_out) {
            /* handed to the caller: plain allocations, freed by rrdb_free_result */
            HIP_OK(hipMalloc(&d_kout, kbytes ? kbytes : 1));
            HIP_OK(hipMalloc(&d_vout, vbytes ? vbytes : 1));
            HIP_OK(hipMalloc(&d_kooffs, (n_out + 1) * 8));
            HIP_OK(hipMalloc(&d_vooffs, (n_out + 1) * 8));
        } else {
            d_kout = e->talloc<uint8_t>(kbytes);
            d_vout = e->talloc<uint8_t>(vbytes);
            d_kooffs = e->talloc<uint64_t>((n_out + 1) * 8);
            d_vooffs = e->talloc<uint64_t>((n_out + 1) * 8);
        }
        if (c->return_expire_ts)
            d_ets = e->talloc<int32_t>(n_out * 4);
        HIP_OK(hipEventRecord(sev[2], e->stream));
        launch_emit_scan(dr, d_win, w, d_state, consumed, d_npos, d_koffs, d_voffs, sp, d_kout,
                         d_vout, d_kooffs, d_vooffs, d_ets, n_out, e->stream);
        HIP_OK(hipEventRecord(sev[3], e->stream));
        if (c->on_device_out) {
            HIP_OK(hipStreamSynchronize(e->stream));
            out->dev_keys = d_kout;
            out->dev_key_offs = d_kooffs;
            out->dev_vals = d_vout;
            out->dev_val_offs = d_vooffs;
            a->dev_ptrs.insert(a->dev_ptrs.end(), {d_kout, d_vout, d_kooffs, d_vooffs});
        } else {
            uint8_t *kb = (uint8_t *)a->alloc(kbytes);
            uint8_t *vb = (uint8_t *)a->alloc(vbytes);
            std::vector<uint64_t> kooffs(n_out + 1), vooffs(n_out + 1);
            e->d2h(kb, d_kout, kbytes);
            e->d2h(vb, d_vout, vbytes);
            HIP_OK(hipMemcpyAsync(kooffs.data(), d_kooffs, (n_out + 1) * 8,
                                  hipMemcpyDeviceToHost, e->stream));
            HIP_OK(hipMemcpyAsync(vooffs.data(), d_vooffs, (n_out + 1) * 8,
                                  hipMemcpyDeviceToHost, e->stream));
            int32_t *ets = nullptr;
            if (d_ets) {
                ets = (int32_t *)a->alloc(n_out * 4);
                HIP_OK(hipMemcpyAsync(ets, d_ets, n_out * 4, hipMemcpyDeviceToHost, e->stream));
            }
            HIP_OK(hipStreamSynchronize(e->stream));
            out->keys = (rrdb_slice *)a->alloc(n_out * sizeof(rrdb_slice));
            out->values = (rrdb_slice *)a->alloc(n_out * sizeof(rrdb_slice));
            for (uint64_t i = 0; i < n_out; i++) {
                out->keys[i] = {kb + kooffs[i], kooffs[i + 1] - kooffs[i]};
                out->values[i] = {vb + vooffs[i], vooffs[i + 1] - vooffs[i]};
            }
            out->expire_ts = ets;
            e->tfree(d_kout);
            e->tfree(d_vout);
            e->tfree(d_kooffs);
            e->tfree(d_vooffs);
        }
        if (d_ets)
            e->tfree(d_ets); /* always pool-allocated; never handed out */
        out->count = n_out;
        e->tfree(d_koffs);
        e->tfree(d_voffs);
    }
    {
        float ms;
        HIP_OK(hipEventSynchronize(sev[1]));
        HIP_OK(hipEventElapsedTime(&ms, sev[0], sev[1]));
        e->phase_ms["scan_state"] = ms;
        if (hipEventQuery(sev[3]) == hipSuccess &&
            hipEventElapsedTime(&ms, sev[2], sev[3]) == hipSuccess)
            e->phase_ms["scan_emit"] = ms;
    }
    for (auto &x : sev)
        (void)hipEventDestroy(x);
    e->tfree(d_state);
    e->tfree(d_ksz);
    e->tfree(d_vsz);
    e->tfree(d_flags);
    e->tfree(d_npos);
    e->tfree(d_cut);
    c->cursor += consumed;
    out->error = RRDB_OK;
    out->context_id =
        (c->cursor >= c->view_n) ? RRDB_SCAN_CONTEXT_ID_COMPLETED : 0 /* caller parks */;
    /* time budget (range_read_limiter.h:56-79), batch granularity: an
     * over-budget incomplete batch returns kIncomplete, no re-park */
    if (out->context_id != RRDB_SCAN_CONTEXT_ID_COMPLETED && e->iter_time_ms > 0) {
        auto ms = std::chrono::duration_cast<std::chrono::milliseconds>(
                      std::chrono::steady_clock::now() - t_start)
                      .count();
        if ((uint64_t)ms > e->iter_time_ms) {
            out->error = RRDB_INCOMPLETE;
            out->context_id = RRDB_SCAN_CONTEXT_ID_COMPLETED;
        }
    }
}

int32_t rrdb_scan_open(void *h, const rrdb_scan_request *q, uint32_t epoch_now, rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    if (engine_flush((HipEngine *)h) != RRDB_OK) {
        result_init(out);
        out->error = RRDB_IO_ERROR; /* device full; writes retained */
        return out->error;
    }
    ((HipEngine *)h)->scratch_reset();
    Arena *a = result_init(out);
    if (q->hash_key_filter_type < 0 || q->hash_key_filter_type > 3 ||
        q->sort_key_filter_type < 0 || q->sort_key_filter_type > 3 ||
        (q->hash_key_filter_type == RRDB_FT_MATCH_PREFIX &&
         !hklen_ok(q->hash_key_filter_pattern.len))) {
        out->error = RRDB_INVALID_ARGUMENT; /* on_get_scanner:1168-1186 */
        return out->error;
    }
    e->activate();
    std::string start((const char *)q->start_key.data, q->start_key.len);
    std::string stop((const char *)q->stop_key.data, q->stop_key.len);
    bool start_inclusive = q->start_inclusive, stop_inclusive = q->stop_inclusive;
    /* hash-key prefix filter clamps start (on_get_scanner:1206-1224) */
    if (q->hash_key_filter_type == RRDB_FT_MATCH_PREFIX && q->hash_key_filter_pattern.len > 0) {
        std::string ps =
            make_key(q->hash_key_filter_pattern.data, q->hash_key_filter_pattern.len, nullptr, 0);
        if (key_cmp(ps, start) > 0) {
            start = ps;
            start_inclusive = true;
        }
    }
    int c = key_cmp(start, stop);
    if (c > 0 || (c == 0 && (!start_inclusive || !stop_inclusive))) {
        out->error = RRDB_OK; /* empty range (on_get_scanner:1227-1243) */
        out->context_id = RRDB_SCAN_CONTEXT_ID_COMPLETED;
        return RRDB_OK;
    }
    auto *ctx = new HipScanCtx();
    std::string stop_excl = stop;
    if (stop_inclusive)
        stop_excl.push_back('\0'); /* smallest key > stop */
    e->build_view(&start, &stop_excl, &ctx->d_view, &ctx->view_n);
    /* first_exclusive: skip an exact start match (on_get_scanner:1277-1283) */
    if (!start_inclusive && ctx->view_n > 0) {
        uint32_t *d_eq = e->talloc<uint32_t>(4);
        uint8_t *d_sk = e->upload_tmp(start.data(), start.size());
        launch_first_eq(e->dev_runs(), ctx->d_view, ctx->view_n, d_sk, start.size(), d_eq,
                        e->stream);
        uint32_t eq = 0;
        HIP_OK(hipMemcpyAsync(&eq, d_eq, 4, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        e->tfree(d_eq);
        e->tfree(d_sk);
        if (eq)
            ctx->cursor = 1;
    }
    ctx->batch_size = q->batch_size;
    ctx->no_value = q->no_value;
    ctx->validate_hash_req = q->validate_partition_hash;
    ctx->return_expire_ts = q->return_expire_ts;
    ctx->only_return_count = q->only_return_count;
    ctx->on_device_out = q->on_device_out;
    ctx->hk_ft = q->hash_key_filter_type;
    ctx->sk_ft = q->sort_key_filter_type;
    ctx->hk_pat_len = q->hash_key_filter_pattern.len;
    ctx->d_hk_pat = e->upload_bytes(q->hash_key_filter_pattern.data, ctx->hk_pat_len);
    ctx->sk_pat_len = q->sort_key_filter_pattern.len;
    ctx->d_sk_pat = e->upload_bytes(q->sort_key_filter_pattern.data, ctx->sk_pat_len);

    scan_batch_gpu(e, ctx, epoch_now, out, a);
    e->gc_ctxs(epoch_now);
    if (out->context_id == RRDB_SCAN_CONTEXT_ID_COMPLETED) {
        delete ctx;
    } else {
        ctx->id = ++e->next_ctx_id;
        ctx->parked_at = epoch_now;
        e->ctxs[ctx->id] = ctx;
        out->context_id = ctx->id;
    }
    return out->error;
}

int32_t rrdb_scan_next(void *h, int64_t context_id, uint32_t epoch_now, rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    Arena *a = result_init(out);
    e->activate();
    e->scratch_reset();
    e->gc_ctxs(epoch_now);
    HipScanCtx *c = nullptr;
    {
        auto it = e->ctxs.find(context_id);
        if (it != e->ctxs.end()) {
            c = it->second;
            e->ctxs.erase(it);
        }
    }
    if (!c) {
        out->error = RRDB_NOT_FOUND; /* on_scan:1539-1541; also the 5-min
                                        expired-context path (:1381-1387) */
        return out->error;
    }
    scan_batch_gpu(e, c, epoch_now, out, a);
    if (out->context_id == RRDB_SCAN_CONTEXT_ID_COMPLETED) {
        delete c;
    } else {
        c->id = ++e->next_ctx_id; /* re-park under a fresh handle (on_scan:1516-1526) */
        c->parked_at = epoch_now;
        e->ctxs[c->id] = c;
        out->context_id = c->id;
    }
    return out->error;
}

/* fused pipelined count scan (the count_data path,
 * src/shell/commands/data_operations.cpp:2305 fan-out): begin submits
 * bounds + anchor rows + the MODE=2 group-rank kernel with NO host sync, so
 * the caller can begin every partition and then finish each — partitions'
 * count kernels co-run on their per-engine streams exactly like the
 * split compaction.  Supports the full-count shape (only_return_count,
 * forward, start-inclusive, caps >= table size); other shapes return
 * kInvalidArgument and the caller falls back to rrdb_scan_open.  The 30s
 * time budget does not apply (single fused operation; documented
 * deviation like the batch-granular budget). */
int32_t rrdb_scan_count_begin(void *h, const rrdb_scan_request *q, uint32_t epoch_now)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    if (e->pend_scan.active)
        return RRDB_INVALID_ARGUMENT;
    if (engine_flush(e) != RRDB_OK)
        return RRDB_IO_ERROR;
    e->activate();
    if (!q->only_return_count || !q->start_inclusive || q->hash_key_filter_type < 0 ||
        q->hash_key_filter_type > 3 || q->sort_key_filter_type < 0 ||
        q->sort_key_filter_type > 3)
        return RRDB_INVALID_ARGUMENT;
    uint64_t total = 0;
    for (auto &r : e->runs)
        total += r.n;
    uint64_t batch_cap = q->batch_size > 0 ? (uint64_t)q->batch_size : (uint64_t)INT32_MAX;
    int R0 = (int)e->runs.size();
    bool single = R0 == 1; /* no merge/shadow needed: plain count kernel */
    if ((R0 > 1 && !e->grp_eligible()) || batch_cap < total || e->max_iter_count < total)
        return RRDB_INVALID_ARGUMENT;
    if (R0 == 0) {
        e->pend_scan.active = true;
        e->pend_scan.trivial = true;
        e->pend_scan.trivial_count = 0;
        return RRDB_OK;
    }

    std::string start((const char *)q->start_key.data, q->start_key.len);
    std::string stop((const char *)q->stop_key.data, q->stop_key.len);
    bool start_inclusive = true, stop_inclusive = q->stop_inclusive;
    if (q->hash_key_filter_type == RRDB_FT_MATCH_PREFIX && q->hash_key_filter_pattern.len > 0) {
        if (!hklen_ok(q->hash_key_filter_pattern.len))
            return RRDB_INVALID_ARGUMENT;
        std::string ps =
            make_key(q->hash_key_filter_pattern.data, q->hash_key_filter_pattern.len, nullptr, 0);
        if (key_cmp(ps, start) > 0)
            start = ps;
    }
    int c = key_cmp(start, stop);
    if (c > 0 || (c == 0 && (!start_inclusive || !stop_inclusive))) {
        e->pend_scan.active = true;
        e->pend_scan.trivial = true;
        e->pend_scan.trivial_count = 0;
        return RRDB_OK; /* empty range (on_get_scanner:1227-1243) */
    }
    e->pend_scan.trivial = false;
    std::string stop_excl = stop;
    if (stop_inclusive)
        stop_excl.push_back('\0');

    int R = (int)e->runs.size();
    DevRun *dr = e->dev_runs();
    auto &ps = e->pend_scan;
    /* pooled allocations: grow-only, freed at close */
    if ((uint64_t)R * 8 > ps.lo_cap) {
        if (ps.d_lo)
            (void)hipFree(ps.d_lo);
        if (ps.d_hi)
            (void)hipFree(ps.d_hi);
        HIP_OK(hipMalloc(&ps.d_lo, R * 8));
        HIP_OK(hipMalloc(&ps.d_hi, R * 8));
        ps.lo_cap = (uint64_t)R * 8;
    }
    uint64_t need_bytes = start.size() + stop_excl.size() + q->hash_key_filter_pattern.len +
                          q->sort_key_filter_pattern.len + 4;
    if (need_bytes > ps.bytes_cap) {
        if (ps.d_start)
            (void)hipFree(ps.d_start);
        HIP_OK(hipMalloc(&ps.d_start, need_bytes));
        ps.bytes_cap = need_bytes;
    }
    uint64_t boff = 0;
    auto upl = [&](const uint8_t *p, uint64_t n) {
        uint8_t *d = ps.d_start + boff;
        if (n)
            HIP_OK(hipMemcpyAsync(d, p, n, hipMemcpyHostToDevice, e->stream));
        boff += n + 1;
        return d;
    };
    uint8_t *d_start_b = upl((const uint8_t *)start.data(), start.size());
    uint8_t *d_stop_b = upl((const uint8_t *)stop_excl.data(), stop_excl.size());
    launch_bounds(dr, R, d_start_b, start.size(), ps.d_lo, 0, e->stream);
    launch_bounds(dr, R, d_stop_b, stop_excl.size(), ps.d_hi, 0, e->stream);
    uint64_t n_groups = 0;
    if (!single) {
        /* anchors from the largest run's FULL size (the window is device-
         * side only); rows past the window collapse to sentinels in-kernel */
        int q0 = 0;
        uint64_t nmax = 0;
        for (int r = 0; r < R; r++)
            if (e->runs[r].n > nmax) {
                nmax = e->runs[r].n;
                q0 = r;
            }
        int gs = 4;
        while ((1ull << (gs + 1)) * (uint64_t)R <= GRP_TARGET && gs < 8)
            gs++;
        n_groups = (nmax + (1ull << gs) - 1) >> gs;
        if (n_groups == 0)
            n_groups = 1;
        if ((n_groups + 1) * R * 8 > ps.anch_cap) {
            if (ps.d_anch)
                (void)hipFree(ps.d_anch);
            HIP_OK(hipMalloc(&ps.d_anch, (n_groups + 1) * R * 8));
            ps.anch_cap = (n_groups + 1) * R * 8;
        }
        launch_anchor_rows(dr, R, q0, ps.d_lo, ps.d_hi, gs, n_groups, ps.d_anch, e->stream);
    }

    ScanParams sp{};
    sp.epoch_now = epoch_now;
    sp.data_version = e->data_version;
    sp.pidx = e->pidx;
    sp.partition_version = e->partition_version;
    sp.validate_hash = (uint8_t)(q->validate_partition_hash && e->validate_hash);
    sp.hk_ft = q->hash_key_filter_type;
    sp.sk_ft = q->sort_key_filter_type;
    sp.hk_pat_len = q->hash_key_filter_pattern.len;
    sp.hk_pat = upl(q->hash_key_filter_pattern.data, q->hash_key_filter_pattern.len);
    sp.sk_pat_len = q->sort_key_filter_pattern.len;
    sp.sk_pat = upl(q->sort_key_filter_pattern.data, q->sort_key_filter_pattern.len);
    sp.no_value = 1;
    sp.hash_key_skip = 0;

    if (!ps.d_stats)
        HIP_OK(hipMalloc(&ps.d_stats, 8 * sizeof(CompactStatsDev)));
    HIP_OK(hipMemsetAsync(ps.d_stats, 0, 8 * sizeof(CompactStatsDev), e->stream));
    if (single)
        launch_count_single(dr, ps.d_lo, ps.d_hi, sp, ps.d_stats, e->runs[0].n, e->stream);
    else
        launch_rank_grp_count(dr, R, ps.d_lo, ps.d_anch, n_groups, sp, ps.d_stats, e->stream);
    ps.active = true;
    return RRDB_OK;
}

int32_t rrdb_scan_count_finish(void *h, rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    result_init(out);
    auto &ps = e->pend_scan;
    if (!ps.active) {
        out->error = RRDB_INVALID_ARGUMENT;
        return out->error;
    }
    ps.active = false;
    e->activate();
    if (ps.trivial) {
        out->i64 = ps.trivial_count;
        out->count = 0;
        out->error = RRDB_OK;
        out->context_id = RRDB_SCAN_CONTEXT_ID_COMPLETED;
        return RRDB_OK;
    }
    HIP_OK(hipStreamSynchronize(e->stream));
    CompactStatsDev hsb[8];
    HIP_OK(hipMemcpy(hsb, ps.d_stats, sizeof(hsb), hipMemcpyDeviceToHost));
    uint64_t count = 0;
    for (auto &b : hsb)
        count += b.output_records;
    /* pooled device buffers persist for the next count (freed at close) */
    ps.trivial = false;
    out->i64 = (int64_t)count;
    out->count = 0;
    out->error = RRDB_OK;
    out->context_id = RRDB_SCAN_CONTEXT_ID_COMPLETED;
    return RRDB_OK;
}

void rrdb_clear_scanner(void *h, int64_t context_id)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    auto it = e->ctxs.find(context_id);
    if (it != e->ctxs.end()) {
        e->activate();
        delete it->second;
        e->ctxs.erase(it);
    }
}

int32_t rrdb_sortkey_count(void *h, const uint8_t *hash_key, uint64_t hklen, uint32_t epoch_now,
                           rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    if (engine_flush((HipEngine *)h) != RRDB_OK) {
        result_init(out);
        out->error = RRDB_IO_ERROR;
        return out->error;
    }
    ((HipEngine *)h)->scratch_reset();
    result_init(out);
    e->activate();
    if (!hklen_ok(hklen)) {
        out->error = RRDB_INVALID_ARGUMENT;
        return out->error;
    }
    /* start=(hk,""), stop=next(hk) (on_sortkey_count:1030-1036); count all
     * visible non-expired rows — no count cap in the reference's loop */
    std::string start = make_key(hash_key, hklen, nullptr, 0);
    std::string stop = next_blob(start);
    uint64_t *d_view = nullptr, n = 0;
    e->build_view(&start, &stop, &d_view, &n);
    int64_t count = 0;
    if (n) {
        DevRun *dr = e->dev_runs();
        ScanParams sp{};
        sp.epoch_now = epoch_now;
        sp.data_version = e->data_version;
        sp.pidx = e->pidx;
        sp.partition_version = e->partition_version;
        sp.validate_hash = 0;
        uint8_t *d_state = e->talloc<uint8_t>(n);
        uint64_t *d_ksz = e->talloc<uint64_t>(n * 8);
        uint64_t *d_vsz = e->talloc<uint64_t>(n * 8);
        uint64_t *d_flags = e->talloc<uint64_t>(n * 8);
        uint64_t *d_npos = e->talloc<uint64_t>(n * 8);
        launch_scan_state(dr, d_view, n, sp, d_state, d_ksz, d_vsz, e->stream);
        launch_normal_flags(d_state, n, d_flags, e->stream);
        launch_psum(d_flags, d_npos, n, e->psum_scratch(n), e->stream);
        uint64_t lastp = 0, lastf = 0;
        HIP_OK(hipMemcpyAsync(&lastp, d_npos + n - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&lastf, d_flags + n - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        count = (int64_t)(lastp + lastf);
        e->tfree(d_state);
        e->tfree(d_ksz);
        e->tfree(d_vsz);
        e->tfree(d_flags);
        e->tfree(d_npos);
    }
    if (d_view)
        (void)hipFree(d_view);
    out->i64 = count;
    out->error = RRDB_OK;
    return RRDB_OK;
}

/* body of on_multi_get; caller holds the engine lock */
/* bounded busy-wait completion for the small-op serving path: the blocking
 * stream sync costs ~10-20us of wakeup latency per call */
static inline void spin_sync(hipStream_t st)
{
    hipError_t e;
    int spins = 0;
    while ((e = hipStreamQuery(st)) == hipErrorNotReady) {
        if (++spins > (1 << 22)) { /* ~seconds: fall back to blocking */
            HIP_OK(hipStreamSynchronize(st));
            return;
        }
    }
    HIP_OK(e);
}

/* it->Valid() after a limit exit (on_multi_get:777-788): does any
 * rocksdb-iterator-visible record exist beyond the range boundary?
 * forward: key >= bound; reverse: key < bound. */
static bool valid_beyond(HipEngine *e, const std::string &bound, bool reverse)
{
    if (e->runs.empty())
        return false;
    if ((int)e->runs.size() <= RRDB_MAX_RUNS) {
        uint32_t *d_out = e->talloc<uint32_t>(4);
        uint8_t *d_b = e->upload_tmp(bound.data(), bound.size());
        launch_valid_beyond(e->dev_runs(), (int)e->runs.size(), d_b, bound.size(),
                            reverse ? 1 : 0, d_out, e->stream);
        uint32_t v = 0;
        HIP_OK(hipMemcpyAsync(&v, d_out, 4, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        e->tfree(d_out);
        e->tfree(d_b);
        return v != 0;
    }
    /* > RRDB_MAX_RUNS runs: fall back to a visible-view existence check */
    uint64_t *d_view = nullptr, n = 0;
    std::string b = bound;
    if (!reverse)
        e->build_view(&b, nullptr, &d_view, &n);
    else
        e->build_view(nullptr, &b, &d_view, &n);
    if (d_view)
        (void)hipFree(d_view);
    return n != 0;
}

static int32_t multi_get_locked(void *h, const rrdb_multi_get_request *q, uint32_t epoch_now,
                                rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    if (engine_flush(e) != RRDB_OK) {
        result_init(out);
        out->error = RRDB_IO_ERROR;
        return out->error;
    }
    e->scratch_reset();
    Arena *a = result_init(out);
    e->activate();
    if (q->sort_key_filter_type < 0 || q->sort_key_filter_type > 3 ||
        !hklen_ok(q->hash_key.len)) {
        out->error = RRDB_INVALID_ARGUMENT; /* on_multi_get:508-517 */
        return out->error;
    }
    uint32_t max_kv_count = e->mg_max_iter_count;
    if (q->max_kv_count > 0 && (uint32_t)q->max_kv_count < max_kv_count)
        max_kv_count = (uint32_t)q->max_kv_count;
    int64_t max_kv_size = q->max_kv_size > 0 ? q->max_kv_size : INT32_MAX;
    int64_t max_iter_size = std::min<int64_t>(
        max_kv_size, e->mg_max_iter_size > 0 ? (int64_t)e->mg_max_iter_size : INT32_MAX);
    uint32_t max_iteration_count = e->mg_max_iter_count;

    if (q->n_sort_keys > 0) {
        /* point-list variant (on_multi_get:779-860) over the batched-get core */
        std::vector<uint64_t> offs(q->n_sort_keys + 1);
        std::string keys;
        for (uint64_t i = 0; i < q->n_sort_keys; i++) {
            offs[i] = keys.size();
            keys += make_key(q->hash_key.data, q->hash_key.len,
                             q->sort_keys + q->sort_key_offs[i],
                             q->sort_key_offs[i + 1] - q->sort_key_offs[i]);
        }
        offs[q->n_sort_keys] = keys.size();
        std::vector<int32_t> status;
        std::vector<std::pair<const uint8_t *, uint64_t>> vals;
        get_core(e, q->n_sort_keys, (const uint8_t *)keys.data(), offs.data(), epoch_now, status,
                 vals, a);
        out->keys = (rrdb_slice *)a->alloc(q->n_sort_keys * sizeof(rrdb_slice));
        out->values = (rrdb_slice *)a->alloc(q->n_sort_keys * sizeof(rrdb_slice));
        uint64_t m = 0;
        int64_t count = 0, size = 0;
        bool exceed = false;
        for (uint64_t i = 0; i < q->n_sort_keys; i++) {
            if (status[i] != RRDB_OK)
                continue;
            if (count >= (int64_t)max_kv_count || size >= max_kv_size) {
                exceed = true; /* :837-841 */
                break;
            }
            uint64_t sklen = q->sort_key_offs[i + 1] - q->sort_key_offs[i];
            uint8_t *kc = (uint8_t *)a->alloc(sklen);
            memcpy(kc, q->sort_keys + q->sort_key_offs[i], sklen);
            out->keys[m] = {kc, sklen};
            if (!q->no_value)
                out->values[m] = {(uint8_t *)vals[i].first, vals[i].second};
            else
                out->values[m] = {nullptr, 0};
            count++;
            size += (int64_t)out->keys[m].len + (int64_t)out->values[m].len;
            m++;
        }
        out->count = m;
        out->error = exceed ? RRDB_INCOMPLETE : RRDB_OK;
        return out->error;
    }

    /* range variant (on_multi_get:540-778) */
    std::string start =
        make_key(q->hash_key.data, q->hash_key.len, q->start_sortkey.data, q->start_sortkey.len);
    std::string stop;
    bool start_inclusive = q->start_inclusive, stop_inclusive;
    if (q->stop_sortkey.len == 0) {
        stop = next_blob(make_key(q->hash_key.data, q->hash_key.len, nullptr, 0));
        stop_inclusive = false;
    } else {
        stop = make_key(q->hash_key.data, q->hash_key.len, q->stop_sortkey.data,
                        q->stop_sortkey.len);
        stop_inclusive = q->stop_inclusive;
    }
    if (q->sort_key_filter_type == RRDB_FT_MATCH_PREFIX && q->sort_key_filter_pattern.len > 0) {
        std::string ps = make_key(q->hash_key.data, q->hash_key.len,
                                  q->sort_key_filter_pattern.data, q->sort_key_filter_pattern.len);
        std::string pe = next_blob(ps);
        if (key_cmp(ps, start) > 0) {
            start = ps;
            start_inclusive = true;
        }
        if (key_cmp(pe, stop) <= 0) {
            stop = pe;
            stop_inclusive = false;
        }
    }
    int c = key_cmp(start, stop);
    if (c > 0 || (c == 0 && (!start_inclusive || !stop_inclusive))) {
        out->error = RRDB_OK; /* empty range (:580-607) */
        return RRDB_OK;
    }
    std::string stop_excl = stop;
    if (stop_inclusive)
        stop_excl.push_back('\0');

    /* ---- fused single-launch fast path (small ranges, the YCSB-E shape):
     * one pinned H2D of the request bytes, one launch, one pinned D2H of
     * hdr + blob prefix, one sync ---- */
    if (!e->runs.empty() && (int)e->runs.size() <= RRDB_MAX_RUNS) {
        e->mg_lane_init();
        MgFusedArgs fa{};
        uint64_t in_n = start.size() + stop_excl.size() + q->sort_key_filter_pattern.len;
        const uint8_t *d_blob = nullptr;
        const uint8_t *h_resp = e->mg_hout;
        bool served = false;
        if (sizeof(MgGraphHdr) + in_n <= MG_GRAPH_IN && e->mg_persist_wanted() &&
            e->server_ready()) {
            /* resident-kernel lane: write the request slice, bump the
             * doorbell, spin for completion */
            MgGraphHdr *hh = (MgGraphHdr *)e->mg_mb->req;
            hh->start_len = (uint32_t)start.size();
            hh->stop_len = (uint32_t)stop_excl.size();
            hh->sk_pat_len = (uint32_t)q->sort_key_filter_pattern.len;
            hh->start_inclusive = start_inclusive;
            hh->stop_inclusive = stop_inclusive;
            hh->reverse = q->reverse;
            hh->no_value = q->no_value;
            hh->max_kv_count = max_kv_count;
            hh->max_iteration_count = max_iteration_count;
            hh->max_iteration_size = max_iter_size;
            hh->sk_ft = q->sort_key_filter_type;
            hh->epoch_now = epoch_now;
            hh->data_version = e->data_version;
            hh->hash_key_skip = 2 + q->hash_key.len;
            uint8_t *pp = e->mg_mb->req + sizeof(MgGraphHdr);
            memcpy(pp, start.data(), start.size());
            pp += start.size();
            memcpy(pp, stop_excl.data(), stop_excl.size());
            pp += stop_excl.size();
            if (q->sort_key_filter_pattern.len)
                memcpy(pp, q->sort_key_filter_pattern.data, q->sort_key_filter_pattern.len);
            uint64_t seq = ++e->mg_srv_seq;
            __atomic_store_n(&e->mg_mb->req_seq, seq, __ATOMIC_RELEASE);
            auto t0 = std::chrono::steady_clock::now();
            bool ok = true;
            while (__atomic_load_n(&e->mg_mb->done_seq, __ATOMIC_ACQUIRE) != seq) {
                auto ms = std::chrono::duration_cast<std::chrono::milliseconds>(
                              std::chrono::steady_clock::now() - t0)
                              .count();
                if (ms > 2 && !__atomic_load_n(&e->mg_mb->alive, __ATOMIC_ACQUIRE)) {
                    /* idle-expired between the readiness check and the
                     * doorbell: relaunch (seq rewound so the fresh kernel
                     * sees this request as new) and re-ring */
                    e->mg_srv_gen = ~0ull;
                    e->mg_srv_seq = seq - 1;
                    bool up = e->server_ready();
                    e->mg_srv_seq = seq;
                    if (!up) {
                        ok = false;
                        break;
                    }
                    __atomic_store_n(&e->mg_mb->req_seq, seq, __ATOMIC_RELEASE);
                    t0 = std::chrono::steady_clock::now();
                    continue;
                }
                if (ms > 200) {
                    ok = false; /* server died: fall to the graph lane */
                    break;
                }
            }
            if (ok) {
                h_resp = e->mg_mb->resp;
                d_blob = e->mg_dout + 32;
                served = true;
            }
        }
        if (!served && sizeof(MgGraphHdr) + in_n <= MG_GRAPH_IN && e->mg_graph_ready()) {
            /* captured-graph lane: fill the request slice, one graph launch */
            MgGraphHdr *hh = (MgGraphHdr *)e->mg_hin;
            hh->start_len = (uint32_t)start.size();
            hh->stop_len = (uint32_t)stop_excl.size();
            hh->sk_pat_len = (uint32_t)q->sort_key_filter_pattern.len;
            hh->start_inclusive = start_inclusive;
            hh->stop_inclusive = stop_inclusive;
            hh->reverse = q->reverse;
            hh->no_value = q->no_value;
            hh->max_kv_count = max_kv_count;
            hh->max_iteration_count = max_iteration_count;
            hh->max_iteration_size = max_iter_size;
            hh->sk_ft = q->sort_key_filter_type;
            hh->epoch_now = epoch_now;
            hh->data_version = e->data_version;
            hh->hash_key_skip = 2 + q->hash_key.len;
            uint8_t *pp = e->mg_hin + sizeof(MgGraphHdr);
            memcpy(pp, start.data(), start.size());
            pp += start.size();
            memcpy(pp, stop_excl.data(), stop_excl.size());
            pp += stop_excl.size();
            if (q->sort_key_filter_pattern.len)
                memcpy(pp, q->sort_key_filter_pattern.data, q->sort_key_filter_pattern.len);
            HIP_OK(hipGraphLaunch(e->mg_graph, e->stream));
            spin_sync(e->stream);
            d_blob = e->mg_dout + 32;
            served = true;
        }
        if (!served && in_n <= HipEngine::MG_IN_CAP) {
            uint64_t o = 0;
            memcpy(e->mg_hin + o, start.data(), start.size());
            fa.start = e->mg_din + o;
            o += start.size();
            memcpy(e->mg_hin + o, stop_excl.data(), stop_excl.size());
            fa.stop = e->mg_din + o;
            o += stop_excl.size();
            if (q->sort_key_filter_pattern.len)
                memcpy(e->mg_hin + o, q->sort_key_filter_pattern.data,
                       q->sort_key_filter_pattern.len);
            fa.sk_pat = e->mg_din + o;
            HIP_OK(hipMemcpyAsync(e->mg_din, e->mg_hin, in_n ? in_n : 1,
                                  hipMemcpyHostToDevice, e->stream));
        } else if (!served) { /* giant keys: pageable per-piece uploads */
            fa.start = e->upload_tmp(start.data(), start.size());
            fa.stop = e->upload_tmp(stop_excl.data(), stop_excl.size());
            fa.sk_pat = e->upload_tmp(q->sort_key_filter_pattern.data,
                                      q->sort_key_filter_pattern.len);
        }
        fa.start_len = start.size();
        fa.stop_len = stop_excl.size();
        fa.start_inclusive = start_inclusive;
        fa.stop_inclusive = stop_inclusive;
        fa.reverse = q->reverse;
        fa.no_value = q->no_value;
        fa.max_kv_count = max_kv_count;
        fa.max_iteration_count = max_iteration_count;
        fa.max_iteration_size = max_iter_size;
        fa.sk_ft = q->sort_key_filter_type;
        fa.sk_pat_len = q->sort_key_filter_pattern.len;
        fa.epoch_now = epoch_now;
        fa.data_version = e->data_version;
        fa.hash_key_skip = 2 + q->hash_key.len;
        if (!served) {
            uint8_t *d_out = e->talloc<uint8_t>(32 + MG_BLOB_BYTES);
            fa.out_hdr = (int64_t *)d_out;
            fa.out_blob = d_out + 32;
            launch_multi_get_small(e->dev_runs(), (int)e->runs.size(), fa, e->stream);
            HIP_OK(hipMemcpyAsync(e->mg_hout, d_out, 32 + HipEngine::MG_OUT_PREFIX,
                                  hipMemcpyDeviceToHost, e->stream));
            spin_sync(e->stream);
            d_blob = fa.out_blob;
        }
        const int64_t *hdr4 = (const int64_t *)h_resp;
        if (hdr4[0] >= 0) {
            uint64_t m = (uint64_t)hdr4[0];
            uint64_t kb = (uint64_t)hdr4[2], vb = (uint64_t)hdr4[3];
            int64_t complete_flag = hdr4[1];
            out->keys = (rrdb_slice *)a->alloc(m * sizeof(rrdb_slice));
            out->values = (rrdb_slice *)a->alloc(m * sizeof(rrdb_slice));
            if (m) {
                uint64_t blob_n = 2 * (m + 1) * 8 + kb + vb;
                uint8_t *hb = (uint8_t *)a->alloc(blob_n);
                if (blob_n <= HipEngine::MG_OUT_PREFIX) {
                    memcpy(hb, h_resp + 32, blob_n); /* already on host */
                } else {
                    HIP_OK(hipMemcpyAsync(hb, d_blob, blob_n, hipMemcpyDeviceToHost,
                                          e->stream));
                    HIP_OK(hipStreamSynchronize(e->stream));
                }
                const uint64_t *koffs = (const uint64_t *)hb;
                const uint64_t *voffs = koffs + (m + 1);
                uint8_t *hk = hb + 2 * (m + 1) * 8;
                uint8_t *hv = hk + kb;
                for (uint64_t j = 0; j < m; j++) {
                    out->keys[j] = {hk + koffs[j], koffs[j + 1] - koffs[j]};
                    out->values[j] = {hv + voffs[j], voffs[j + 1] - voffs[j]};
                }
            }
            out->count = m;
            out->error = complete_flag ? RRDB_OK : RRDB_INCOMPLETE;
            return out->error;
        }
        /* fallback: large range or oversize rows — general path below */
    }

    uint64_t *d_view = nullptr, n = 0;
    e->build_view(&start, &stop_excl, &d_view, &n);
    if (n == 0) {
        if (d_view)
            (void)hipFree(d_view);
        out->error = RRDB_OK;
        return RRDB_OK;
    }
    /* window: the limiter consumes at most mg_max_iter_count iterations plus
     * one possible uncounted first/last-exclusive skip */
    uint64_t wcap = (uint64_t)max_iteration_count + 1;
    uint64_t wstart = 0, w = std::min<uint64_t>(n, wcap);
    if (q->reverse)
        wstart = n - w;
    DevRun *dr = e->dev_runs();
    ScanParams sp{};
    sp.epoch_now = epoch_now;
    sp.data_version = e->data_version;
    sp.pidx = e->pidx;
    sp.partition_version = e->partition_version;
    sp.validate_hash = 0; /* multi_get does no hash validation */
    sp.hk_ft = 0;
    sp.sk_ft = q->sort_key_filter_type;
    sp.sk_pat = nullptr;
    sp.sk_pat_len = q->sort_key_filter_pattern.len;
    uint8_t *d_skpat = e->upload_tmp(q->sort_key_filter_pattern.data, sp.sk_pat_len);
    sp.sk_pat = d_skpat;
    sp.no_value = q->no_value;
    sp.hash_key_skip = 2 + q->hash_key.len; /* emit sortkey only (:2496-2499) */

    uint8_t *d_state = e->talloc<uint8_t>(w);
    uint64_t *d_ksz = e->talloc<uint64_t>(w * 8);
    uint64_t *d_vsz = e->talloc<uint64_t>(w * 8);
    /* sortkey filter types: scan_state's sk filter operates on key minus
     * 2+hklen prefix — same as multi_get's sortkey check */
    launch_scan_state(dr, d_view + wstart, w, sp, d_state, d_ksz, d_vsz, e->stream);
    std::vector<uint8_t> state(w);
    std::vector<uint64_t> ksz(w), vsz(w);
    HIP_OK(hipMemcpyAsync(state.data(), d_state, w, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(ksz.data(), d_ksz, w * 8, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(vsz.data(), d_vsz, w * 8, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipStreamSynchronize(e->stream));
    e->tfree(d_state);
    e->tfree(d_ksz);
    e->tfree(d_vsz);

    /* host limiter loop, mirroring on_multi_get:616-778 exactly */
    int64_t count = 0, size = 0;
    uint64_t iteration_count = 0;
    std::vector<uint64_t> sel; /* view positions to emit, ascending */
    bool complete = false;
    bool skipped_first = false;
    /* reverse + start-exclusive: a record == start is OUT OF RANGE for the
     * reverse walk (on_multi_get:697-700) — excluded, and it keeps the
     * iterator Valid() on a limit exit */
    uint64_t lo_skip = 0;
    if (q->reverse && !start_inclusive && wstart == 0 && n > 0) {
        uint32_t *d_eq = e->talloc<uint32_t>(4);
        uint8_t *d_b = e->upload_tmp(start.data(), start.size());
        launch_first_eq(dr, d_view, 1, d_b, start.size(), d_eq, e->stream);
        uint32_t eq = 0;
        HIP_OK(hipMemcpyAsync(&eq, d_eq, 4, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        e->tfree(d_eq);
        e->tfree(d_b);
        lo_skip = eq ? 1 : 0;
    }
    uint64_t steps = w;
    for (uint64_t s = 0; s < steps; s++) {
        uint64_t wi = q->reverse ? (steps - 1 - s) : s;
        if (q->reverse && lo_skip && wi == 0)
            break; /* reached the excluded == start record: out of range */
        if (count >= (int64_t)max_kv_count || iteration_count >= max_iteration_count ||
            size >= max_iter_size)
            break;
        /* first-exclusive skip of the exact boundary key */
        if (s == 0) {
            /* check boundary equality via sizes: need the key — use a cheap
             * device compare */
            bool check = (!q->reverse && !start_inclusive) || (q->reverse && !stop_inclusive);
            if (check) {
                const std::string &bound = q->reverse ? stop : start;
                uint32_t *d_eq = e->talloc<uint32_t>(4);
                uint8_t *d_b = e->upload_tmp(bound.data(), bound.size());
                launch_first_eq(dr, d_view + wstart + wi, 1, d_b, bound.size(), d_eq, e->stream);
                uint32_t eq = 0;
                HIP_OK(hipMemcpyAsync(&eq, d_eq, 4, hipMemcpyDeviceToHost, e->stream));
                HIP_OK(hipStreamSynchronize(e->stream));
                e->tfree(d_eq);
                e->tfree(d_b);
                if (eq) {
                    skipped_first = true;
                    continue;
                }
            }
        }
        iteration_count++;
        if (state[wi] != ST_NORMAL)
            continue; /* kExpired / kFiltered */
        sel.push_back(wstart + wi);
        count++;
        size += (int64_t)ksz[wi] + (int64_t)vsz[wi];
    }
    /* completion (on_multi_get:777-788): kIncomplete iff the iterator is
     * still Valid() after a limit exit — even when the remaining records lie
     * past the range.  Exception: the post-append c==0 break
     * (:668-672/:739-744) completes when the LAST iterated record equals the
     * inclusive far boundary, regardless of limits. */
    {
        bool boundary_hit = false;
        uint64_t countable_chk = w - (skipped_first ? 1 : 0) - lo_skip;
        if (iteration_count >= countable_chk && iteration_count > 0 && w == n && w > 0 &&
            ((!q->reverse && stop_inclusive) || (q->reverse && start_inclusive))) {
            /* last iterated in-range record = the far end of the view */
            uint64_t vidx = q->reverse ? (lo_skip) : (n - 1);
            const std::string &bnd = q->reverse ? start : stop;
            uint32_t *d_eq = e->talloc<uint32_t>(4);
            uint8_t *d_b = e->upload_tmp(bnd.data(), bnd.size());
            launch_first_eq(dr, d_view + vidx, 1, d_b, bnd.size(), d_eq, e->stream);
            uint32_t eq = 0;
            HIP_OK(hipMemcpyAsync(&eq, d_eq, 4, hipMemcpyDeviceToHost, e->stream));
            HIP_OK(hipStreamSynchronize(e->stream));
            e->tfree(d_eq);
            e->tfree(d_b);
            boundary_hit = eq != 0;
        }
        bool limit_exit = !boundary_hit &&
                          (count >= (int64_t)max_kv_count ||
                           iteration_count >= max_iteration_count || size >= max_iter_size);
        uint64_t countable = w - (skipped_first ? 1 : 0) - lo_skip;
        bool consumed_all = (iteration_count >= countable);
        if (w < n || !consumed_all)
            complete = false; /* in-range records remain: iterator valid */
        else if (!limit_exit)
            complete = true; /* walked past the range end */
        else if (q->reverse)
            complete = !(lo_skip || valid_beyond(e, start, true));
        else
            complete = !valid_beyond(e, stop_excl, false);
    }
    uint64_t m = sel.size();
    if (q->reverse)
        std::reverse(sel.begin(), sel.end()); /* ascending output (:758-765) */
    out->keys = (rrdb_slice *)a->alloc(m * sizeof(rrdb_slice));
    out->values = (rrdb_slice *)a->alloc(m * sizeof(rrdb_slice));
    if (m > 0) {
        std::vector<uint64_t> koffs(m + 1), voffs(m + 1);
        uint64_t kb = 0, vb = 0;
        for (uint64_t j = 0; j < m; j++) {
            uint64_t wi = sel[j] - wstart;
            koffs[j] = kb;
            voffs[j] = vb;
            kb += ksz[wi];
            vb += vsz[wi];
        }
        koffs[m] = kb;
        voffs[m] = vb;
        uint64_t *d_rows = (uint64_t *)e->upload_tmp(sel.data(), m * 8);
        uint64_t *d_ko = (uint64_t *)e->upload_tmp(koffs.data(), (m + 1) * 8);
        uint64_t *d_vo = (uint64_t *)e->upload_tmp(voffs.data(), (m + 1) * 8);
        uint8_t *d_kout = e->talloc<uint8_t>(kb);
        uint8_t *d_vout = e->talloc<uint8_t>(vb);
        launch_emit_rows(dr, d_view, d_rows, m, d_ko, d_vo, sp, d_kout, d_vout, e->stream);
        uint8_t *hkb = (uint8_t *)a->alloc(kb);
        uint8_t *hvb = (uint8_t *)a->alloc(vb);
        HIP_OK(hipMemcpyAsync(hkb, d_kout, kb, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(hvb, d_vout, vb, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        for (uint64_t j = 0; j < m; j++) {
            out->keys[j] = {hkb + koffs[j], koffs[j + 1] - koffs[j]};
            out->values[j] = {hvb + voffs[j], voffs[j + 1] - voffs[j]};
        }
        e->tfree(d_rows);
        e->tfree(d_ko);
        e->tfree(d_vo);
        e->tfree(d_kout);
        e->tfree(d_vout);
    }
    e->tfree(d_skpat);
    (void)hipFree(d_view);
    out->count = m;
    out->error = complete ? RRDB_OK : RRDB_INCOMPLETE; /* :789-799 */
    return out->error;
}

int32_t rrdb_multi_get(void *h, const rrdb_multi_get_request *q, uint32_t epoch_now,
                       rrdb_result *out)
{
    std::lock_guard<std::mutex> g(((HipEngine *)h)->mu);
    return multi_get_locked(h, q, epoch_now, out);
}

/* phase 1 of the compaction (the pipelined-partitions seam; the reference
 * runs per-replica compactions concurrently on THREAD_POOL_COMPACT —
 * pegasus_server_impl.cpp:3373): submit rank + prefix sums + async size
 * reads, NO host-blocking sync.  Caller holds the handle lock. */
static int32_t compact_begin(HipEngine *e, const rrdb_compact_options *opts, uint32_t epoch_now)
{
    if (e->pend.active)
        return RRDB_INVALID_ARGUMENT;
    if (engine_flush(e) != RRDB_OK) /* memtable visible to reads */
        return RRDB_IO_ERROR;
    e->scratch_reset();
    rrdb_compact_stats st{};
    if (e->manual_compact_disabled)
        return RRDB_INVALID_ARGUMENT;
    e->activate();
    int R = (int)e->runs.size();
    uint64_t total = 0;
    for (auto &r : e->runs)
        total += r.n;
    st.input_records = total;
    if (total == 0) {
        e->pend = {};
        e->pend.active = true;
        e->pend.trivial = true;
        e->pend.st = st;
        return RRDB_OK;
    }
    DevRun *dr = e->dev_runs();
    /* full-range rank merge */
    std::vector<uint64_t> lo(R, 0), hi(R), wprefix(R + 1);
    uint64_t acc = 0;
    for (int r = 0; r < R; r++) {
        hi[r] = e->runs[r].n;
        wprefix[r] = acc;
        acc += e->runs[r].n;
    }
    wprefix[R] = acc;
    uint64_t *d_lo = (uint64_t *)e->upload_tmp(lo.data(), R * 8);
    uint64_t *d_hi = (uint64_t *)e->upload_tmp(hi.data(), R * 8);
    uint64_t *d_wp = (uint64_t *)e->upload_tmp(wprefix.data(), (R + 1) * 8);
    uint64_t *d_order = e->talloc<uint64_t>(total * 8);
    e->resolve_phase_events(); /* previous pass's timing, before reuse */
    e->tev_have_emit = false;
    if (!e->tev[0])
        for (auto &x : e->tev)
            HIP_OK(hipEventCreate(&x));
    hipEvent_t *ev = e->tev;

    CompactParams cp{};
    cp.epoch_now = epoch_now;
    cp.data_version = e->data_version;
    cp.default_ttl = e->default_ttl;
    cp.pidx = e->pidx;
    cp.partition_version = e->partition_version;
    cp.validate_hash = e->validate_hash ? 1 : 0;
    cp.ops = e->d_ops;
    cp.n_ops = e->n_ops;
    cp.rules = e->d_rules;
    cp.pats = e->d_pats;

    /* all-fixed-stride fast mode: every run shares one key and one value
     * stride (the hashkey-table schema), so output offsets derive from the
     * keep count — the ksz/vsz arrays and two of the three full-length
     * prefix sums are skipped entirely */
    uint64_t ffk = e->runs.empty() ? 0 : e->runs[0].fixed_klen;
    uint64_t ffv = e->runs.empty() ? 0 : e->runs[0].fixed_vlen_put;
    for (auto &rr : e->runs) {
        if (rr.fixed_klen != ffk)
            ffk = 0;
        if (rr.fixed_vlen_put != ffv)
            ffv = 0;
    }
    bool fixed_emit = ffk > 0 && ffv > 0 && e->emit_mode == 2;
    /* no default-TTL and no user ops => the filter can never rewrite a
     * value: skip the changed/new_expire arrays and the patch pass */
    bool no_rewrite = e->default_ttl == 0 && e->n_ops == 0 && e->emit_mode == 2;
    uint8_t *d_changed = no_rewrite ? nullptr : e->talloc<uint8_t>(total);
    uint32_t *d_new_expire = no_rewrite ? nullptr : e->talloc<uint32_t>(total * 4);
    uint64_t *d_ksz = fixed_emit ? nullptr : e->talloc<uint64_t>(total * 8);
    uint64_t *d_vsz = fixed_emit ? nullptr : e->talloc<uint64_t>(total * 8);
    uint64_t *d_keepw = e->talloc<uint64_t>(total * 8);
    uint64_t *d_kpos = e->talloc<uint64_t>(total * 8);
    uint64_t *d_koffs = fixed_emit ? nullptr : e->talloc<uint64_t>(total * 8);
    uint64_t *d_voffs = fixed_emit ? nullptr : e->talloc<uint64_t>(total * 8);
    uint64_t *d_rank_of = e->emit_mode == 1 ? e->talloc<uint64_t>(total * 8) : nullptr;
    /* 8 stat banks: the group rank flushes per-block tallies to bank
     * blockIdx&7; legacy kernels add to bank 0; finish sums all banks */
    CompactStatsDev *d_stats = e->talloc<CompactStatsDev>(8 * sizeof(CompactStatsDev));
    HIP_OK(hipMemsetAsync(d_stats, 0, 8 * sizeof(CompactStatsDev), e->stream));
    if (e->rank_mode == 1 && R > 1 && e->emit_mode == 2) {
        /* LDS-staged block rank: shift-8 bound table + per-run block counts */
        uint64_t *d_bt8_off = nullptr, *d_bt8 = nullptr;
        {
            uint64_t *c_off = nullptr, *c_bt = nullptr;
            uint64_t wmax = 0;
            for (int r = 0; r < R; r++)
                wmax = std::max(wmax, hi[r] - lo[r]);
            if ((wmax >> 14) > 8)
                e->build_bound_table_level(dr, R, lo, hi, d_lo, d_hi, 14, nullptr, nullptr, 0,
                                           &c_off, &c_bt);
            e->build_bound_table_level(dr, R, lo, hi, d_lo, d_hi, 8, c_off, c_bt, 14, &d_bt8_off,
                                       &d_bt8);
        }
        std::vector<uint64_t> blkp(R + 1);
        uint64_t nb = 0;
        for (int r = 0; r < R; r++) {
            blkp[r] = nb;
            nb += (hi[r] - lo[r] + 255) >> 8;
        }
        blkp[R] = nb;
        uint64_t *d_blkp = (uint64_t *)e->upload_tmp(blkp.data(), (R + 1) * 8);
        HIP_OK(hipEventRecord(ev[0], e->stream));
        launch_rank_compact_lds(dr, R, d_lo, d_hi, d_blkp, nb, cp, d_bt8_off, d_bt8, d_order,
                                d_keepw, d_changed, d_new_expire, d_ksz, d_vsz, d_stats,
                                e->stream);
        HIP_OK(hipEventRecord(ev[1], e->stream));
    } else if (R > 1 && e->emit_mode != 1 && e->grp_eligible()) {
        /* group-streaming rank: no bound table; anchors partition the runs */
        uint64_t *d_anch = nullptr;
        uint64_t n_groups = e->build_anchors(dr, R, lo, hi, d_lo, d_hi, &d_anch);
        HIP_OK(hipEventRecord(ev[0], e->stream));
        launch_rank_grp_compact(dr, R, d_lo, d_anch, n_groups, cp, d_order, d_keepw, d_changed,
                                d_new_expire, d_ksz, d_vsz, d_stats, e->grp_blocks, e->stream);
        HIP_OK(hipEventRecord(ev[1], e->stream));
    } else {
    uint64_t *d_bt_off = nullptr, *d_bt = nullptr;
    if (R > 1 && total > 100000)
        e->build_bound_table(dr, R, lo, hi, d_lo, d_hi, &d_bt_off, &d_bt);
    HIP_OK(hipEventRecord(ev[0], e->stream));
    bool ldst_ok = R > 1 && e->ldst_eligible();
    if (ldst_ok)
        launch_rank_compact_ldst(dr, R, d_lo, d_hi, d_wp, total, cp, d_order, d_keepw,
                                 d_changed, d_new_expire, d_ksz, d_vsz, d_rank_of, d_bt_off,
                                 d_bt, e->bt_shift, d_stats, e->stream);
    else
        launch_rank_compact(dr, R, d_lo, d_hi, d_wp, total, cp, d_order, d_keepw, d_changed,
                            d_new_expire, d_ksz, d_vsz, d_rank_of, d_bt_off, d_bt, e->bt_shift,
                            d_stats, e->stream);
    HIP_OK(hipEventRecord(ev[1], e->stream));
    }
    HIP_OK(hipEventRecord(ev[2], e->stream));
    if (!e->pend_stats_h)
        HIP_OK(hipHostMalloc((void **)&e->pend_stats_h, 8 * sizeof(CompactStatsDev)));
    HIP_OK(hipMemcpyAsync(e->pend_stats_h, d_stats, 8 * sizeof(CompactStatsDev),
                          hipMemcpyDeviceToHost, e->stream));
    launch_psum(d_keepw, d_kpos, total, e->psum_scratch(total), e->stream);
    if (!fixed_emit) {
        launch_psum(d_ksz, d_koffs, total, e->psum_scratch(total), e->stream);
        launch_psum(d_vsz, d_voffs, total, e->psum_scratch(total), e->stream);
    }
    if (!e->pend_sizes)
        HIP_OK(hipHostMalloc((void **)&e->pend_sizes, 6 * 8));
    uint64_t *t = e->pend_sizes;
    memset(t, 0, 6 * 8);
    HIP_OK(hipMemcpyAsync(&t[0], d_kpos + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
    HIP_OK(hipMemcpyAsync(&t[1], d_keepw + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
    if (!fixed_emit) {
        HIP_OK(hipMemcpyAsync(&t[2], d_koffs + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[3], d_ksz + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[4], d_voffs + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
        HIP_OK(hipMemcpyAsync(&t[5], d_vsz + total - 1, 8, hipMemcpyDeviceToHost, e->stream));
    }
    e->pend = {};
    e->pend.active = true;
    e->pend.keep_inputs = opts && opts->keep_inputs;
    e->pend.R = R;
    e->pend.total = total;
    e->pend.dr = dr;
    e->pend.d_wp = d_wp;
    e->pend.d_order = d_order;
    e->pend.d_keepw = d_keepw;
    e->pend.d_changed = d_changed;
    e->pend.d_new_expire = d_new_expire;
    e->pend.d_kpos = d_kpos;
    e->pend.d_koffs = d_koffs;
    e->pend.d_voffs = d_voffs;
    e->pend.d_rank_of = d_rank_of;
    e->pend.d_stats = d_stats;
    e->pend.fk = fixed_emit ? ffk : 0;
    e->pend.fv = fixed_emit ? ffv : 0;
    e->pend.no_rewrite = no_rewrite;
    for (int i = 0; i < 6; i++)
        e->pend.ev[i] = ev[i];
    e->pend.st = st;
    e->arena_pin(); /* reads between begin/finish must not reuse pend buffers */
    (void)d_lo;
    (void)d_hi;
    (void)d_ksz;
    (void)d_vsz;
    return RRDB_OK;
}

/* phase 2: wait for the submitted work, allocate + emit the output run,
 * collect stats.  Caller holds the handle lock. */
static int32_t compact_finish(HipEngine *e, rrdb_compact_stats *stats)
{
    if (!e->pend.active)
        return RRDB_INVALID_ARGUMENT;
    e->activate();
    e->pend.active = false;
    e->arena_unpin();
    rrdb_compact_stats st = e->pend.st;
    if (e->pend.trivial) {
        if (stats)
            *stats = st;
        return RRDB_OK;
    }
    int R = e->pend.R;
    (void)R;
    uint64_t total = e->pend.total;
    DevRun *dr = e->pend.dr;
    uint64_t *d_wp = e->pend.d_wp, *d_order = e->pend.d_order, *d_keepw = e->pend.d_keepw;
    uint8_t *d_changed = e->pend.d_changed;
    uint32_t *d_new_expire = e->pend.d_new_expire;
    uint64_t *d_kpos = e->pend.d_kpos, *d_koffs = e->pend.d_koffs, *d_voffs = e->pend.d_voffs;
    uint64_t *d_rank_of = e->pend.d_rank_of;
    CompactStatsDev *d_stats = e->pend.d_stats;
    (void)d_stats;
    hipEvent_t *ev = e->tev;
    /* wait for the submitted merge phase only (rank + prefix sums + the
     * async size/stats reads — all of this stream's pending work) */
    HIP_OK(hipStreamSynchronize(e->stream));
    uint64_t *t = e->pend_sizes;
    uint64_t n_out = t[0] + t[1];
    uint64_t kbytes = e->pend.fk ? n_out * e->pend.fk : t[2] + t[3];
    uint64_t vbytes = e->pend.fk ? n_out * e->pend.fv : t[4] + t[5];

    CompactStatsDev *hsb = e->pend_stats_h;
    CompactStatsDev hs{};
    for (int b = 0; b < 8; b++) {
        hs.shadowed += hsb[b].shadowed;
        hs.tombstones += hsb[b].tombstones;
        hs.expired += hsb[b].expired;
        hs.filtered += hsb[b].filtered;
        hs.output_records += hsb[b].output_records;
    }
    st.output_records = n_out;
    st.expired = hs.expired;
    st.filtered = hs.filtered;
    st.tombstones = hs.tombstones;
    st.shadowed = hs.shadowed;
    st.output_bytes = kbytes + vbytes;

    RunBuf nr;
    bool keep_inputs = e->pend.keep_inputs;
    if (n_out > 0) {
        nr.n = n_out;
        /* output keys are verbatim copies of input keys, so a stride shared
         * by every input run is preserved */
        nr.fixed_klen = e->runs.empty() ? 0 : e->runs[0].fixed_klen;
        for (auto &ir : e->runs)
            if (ir.fixed_klen != nr.fixed_klen)
                nr.fixed_klen = 0;
        /* output values are verbatim-length copies of input values, so a
         * stride shared by every input run is preserved too */
        nr.fixed_vlen = e->runs.empty() ? 0 : e->runs[0].fixed_vlen;
        for (auto &ir : e->runs)
            if (ir.fixed_vlen != nr.fixed_vlen)
                nr.fixed_vlen = 0;
        /* the merged output holds only PUTs: its plain stride equals the
         * shared put-stride of the inputs (when one exists) */
        nr.fixed_vlen_put = e->runs.empty() ? 0 : e->runs[0].fixed_vlen_put;
        for (auto &ir : e->runs)
            if (ir.fixed_vlen_put != nr.fixed_vlen_put)
                nr.fixed_vlen_put = 0;
        if (!nr.fixed_vlen)
            nr.fixed_vlen = nr.fixed_vlen_put;
        if (keep_inputs) {
            /* output is dropped at the end of the pass: pooled temporaries */
            nr.keys = e->talloc<uint8_t>(kbytes);
            nr.vals = e->talloc<uint8_t>(vbytes);
            nr.koff = e->talloc<uint64_t>((n_out + 1) * 8);
            nr.voff = e->talloc<uint64_t>((n_out + 1) * 8);
            nr.sk = e->talloc<uint64_t>(n_out * 8);
        } else {
            HIP_OK(hipMalloc(&nr.keys, kbytes ? kbytes : 1));
            HIP_OK(hipMalloc(&nr.vals, vbytes ? vbytes : 1));
            HIP_OK(hipMalloc(&nr.koff, (n_out + 1) * 8));
            HIP_OK(hipMalloc(&nr.voff, (n_out + 1) * 8));
            HIP_OK(hipMalloc(&nr.sk, n_out * 8));
        }
        HIP_OK(hipEventRecord(ev[3], e->stream));
        if (e->emit_mode == 2 || e->pend.fk || e->pend.no_rewrite) {
            uint64_t *d_row_ksrc = e->talloc<uint64_t>(n_out * 8);
            uint64_t *d_row_vsrc = e->talloc<uint64_t>(n_out * 8);
            uint32_t *d_row_patch = e->talloc<uint32_t>(n_out * 4);
            uint32_t *d_row_expire = e->talloc<uint32_t>(n_out * 4);
            uint64_t kanch = (((kbytes + 15) >> 4) >> 6) + 1;
            uint64_t vanch = (((vbytes + 15) >> 4) >> 6) + 1;
            uint64_t *d_kanchor = e->talloc<uint64_t>(kanch * 8);
            uint64_t *d_vanchor = e->talloc<uint64_t>(vanch * 8);
            launch_emit_compact_chunked(dr, d_order, total, d_keepw, d_changed, d_new_expire,
                                        d_kpos, d_koffs, d_voffs, e->data_version, n_out, kbytes,
                                        vbytes, d_row_ksrc, d_row_vsrc, d_row_patch,
                                        d_row_expire, nr.keys, nr.vals, nr.koff, nr.voff, nr.sk,
                                        d_kanchor, d_vanchor, e->pend.fk, e->pend.fv,
                                        e->stream);
        } else if (e->emit_mode == 1)
            launch_emit_compact_inmajor(dr, R, d_wp, total, d_rank_of, d_keepw, d_changed,
                                        d_new_expire, d_kpos, d_koffs, d_voffs,
                                        e->data_version, nr.keys, nr.vals, nr.koff, nr.voff,
                                        nr.sk, n_out, e->stream);
        else
            launch_emit_compact(dr, d_order, total, d_keepw, d_changed, d_new_expire, d_kpos,
                                d_koffs, d_voffs, e->data_version, nr.keys, nr.vals, nr.koff,
                                nr.voff, nr.sk, n_out, e->stream);
        HIP_OK(hipEventRecord(ev[4], e->stream));
        e->tev_have_emit = true;
        if (!keep_inputs) {
            /* the swap below frees the input runs the emit reads: drain */
            HIP_OK(hipStreamSynchronize(e->stream));
        }
        /* keep_inputs (repeatable-pass) mode returns with the emit still
         * in flight on this engine's stream — later partitions' finishes
         * overlap it; stream order protects this engine's arena reuse.
         * Timing is resolved lazily by phase_ms(). */
    }
    e->tev_live = true;
    /* arena temporaries (d_order, flags, sums, keep_inputs outputs) are
     * reclaimed at the next scratch_reset */
    if (keep_inputs) {
        /* benchmarking: the pass ran in full; drop the output, keep inputs */
    } else {
        e->server_quit_sync(); /* the resident kernel holds the freed runs */
        e->invalidate_ctxs(); /* parked views reference the freed runs */
        for (auto &r : e->runs)
            e->free_run(r);
        e->runs.clear();
        if (n_out > 0) {
            uint64_t fo2[2], lo2[2];
            uint8_t h32[32], t32[32];
            HIP_OK(hipMemcpy(fo2, nr.koff, 16, hipMemcpyDeviceToHost));
            HIP_OK(hipMemcpy(lo2, nr.koff + (n_out - 1), 16, hipMemcpyDeviceToHost));
            uint64_t fl = std::min<uint64_t>(fo2[1] - fo2[0], 32);
            uint64_t ll = std::min<uint64_t>(lo2[1] - lo2[0], 32);
            HIP_OK(hipMemcpy(h32, nr.keys + fo2[0], fl ? fl : 1, hipMemcpyDeviceToHost));
            HIP_OK(hipMemcpy(t32, nr.keys + lo2[0], ll ? ll : 1, hipMemcpyDeviceToHost));
            set_pfx(lcp_exact32(h32, fl, t32, ll), &nr.pfx_skip, &nr.lcp_exact);
            e->build_tails(nr);
            e->build_meta(nr);
            e->build_bloom(nr);
            e->runs.push_back(nr);
        }
        e->d_runs_dirty = true;
        e->ldst_elig_cache = -1;
    }
    if (stats)
        *stats = st;
    return RRDB_OK;
}

int32_t rrdb_manual_compact(void *h, const rrdb_compact_options *opts, uint32_t epoch_now,
                            rrdb_compact_stats *stats)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    int32_t rc = compact_begin(e, opts, epoch_now);
    if (rc != RRDB_OK) {
        if (stats)
            *stats = rrdb_compact_stats{};
        return rc;
    }
    return compact_finish(e, stats);
}

int32_t rrdb_manual_compact_begin(void *h, const rrdb_compact_options *opts, uint32_t epoch_now)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    return compact_begin(e, opts, epoch_now);
}

int32_t rrdb_manual_compact_finish(void *h, rrdb_compact_stats *stats)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    return compact_finish(e, stats);
}

} /* extern "C" */

/* ================= write path (§8(f)1) =================
 * Host memtable + flush -> sorted run; mirrors pegasus_write_service
 * put/remove + rocksdb_wrapper::write_batch_put/_delete (reference
 * pegasus_write_service.h:119-207, rocksdb_wrapper.cpp:121-247).  Reads
 * flush lazily so committed writes are immediately visible (rocksdb
 * memtable read-path equivalent). */
extern "C" {

int32_t rrdb_put(void *h, const uint8_t *hash_key, uint64_t hklen, const uint8_t *sort_key,
                 uint64_t sklen, const uint8_t *value, uint64_t vlen, uint32_t expire_ts,
                 uint32_t epoch_now)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    if (!hklen_ok(hklen))
        return RRDB_INVALID_ARGUMENT;
    /* write-time default_ttl (rocksdb_wrapper::db_expire_ts, :280-286) */
    if (expire_ts == 0 && e->default_ttl != 0)
        expire_ts = epoch_now + e->default_ttl;
    std::string key = make_key(hash_key, hklen, sort_key, sklen);
    uint32_t hdr = hdr_len(e->data_version);
    std::string val(hdr + vlen, '\0');
    uint32_t off = (e->data_version == 2) ? 1 : 0;
    if (e->data_version == 2)
        val[0] = (char)0x82;
    val[off] = (char)(expire_ts >> 24);
    val[off + 1] = (char)(expire_ts >> 16);
    val[off + 2] = (char)(expire_ts >> 8);
    val[off + 3] = (char)expire_ts;
    if (vlen)
        memcpy(&val[hdr], value, vlen);
    e->memtable[key] = {std::move(val), e->next_seq_floor++, RRDB_KIND_PUT};
    return RRDB_OK;
}

int32_t rrdb_remove(void *h, const uint8_t *hash_key, uint64_t hklen, const uint8_t *sort_key,
                    uint64_t sklen)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    if (!hklen_ok(hklen))
        return RRDB_INVALID_ARGUMENT;
    std::string key = make_key(hash_key, hklen, sort_key, sklen);
    e->memtable[key] = {std::string(), e->next_seq_floor++, RRDB_KIND_DELETE};
    return RRDB_OK;
}

uint64_t rrdb_memtable_entries(void *h) { return ((HipEngine *)h)->memtable.size(); }

static int32_t engine_flush(HipEngine *e)
{
    if (e->memtable.empty())
        return RRDB_OK;
    e->activate();
    std::string keys, vals;
    std::vector<uint64_t> koff{0}, voff{0}, sk;
    for (auto &kv : e->memtable) {
        keys += kv.first;
        koff.push_back(keys.size());
        vals += std::get<0>(kv.second);
        voff.push_back(vals.size());
        sk.push_back((std::get<1>(kv.second) << 1) | (uint64_t)std::get<2>(kv.second));
    }
    int32_t rc = ingest_prepared(e, keys, koff, vals, voff, sk);
    if (rc != RRDB_OK)
        return rc; /* memtable retained: the write is not lost */
    e->memtable.clear();
    return RRDB_OK;
}

int32_t rrdb_flush(void *h)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    return engine_flush(e);
}

} /* extern "C" */

/* ================= checkpoint (§8(f)2) ================= */
#include <sys/stat.h>
#include <cstdio>

extern "C" {

static bool write_file(const std::string &path, const void *data, uint64_t n)
{
    FILE *f = fopen(path.c_str(), "wb");
    if (!f)
        return false;
    bool ok = (n == 0) || fwrite(data, 1, n, f) == n;
    fclose(f);
    return ok;
}

static bool read_file(const std::string &path, std::vector<uint8_t> &out)
{
    FILE *f = fopen(path.c_str(), "rb");
    if (!f)
        return false;
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    out.resize((size_t)n);
    bool ok = (n == 0) || fread(out.data(), 1, (size_t)n, f) == (size_t)n;
    fclose(f);
    return ok;
}

int32_t rrdb_checkpoint(void *h, const char *dir, uint64_t decree)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    engine_flush(e);
    e->activate();
    std::string path = std::string(dir) + "/checkpoint." + std::to_string(decree);
    mkdir(dir, 0755);
    if (mkdir(path.c_str(), 0755) != 0)
        return RRDB_IO_ERROR;
    /* run files first (crc64 of each goes into the MANIFEST; a restorer
     * verifies — the reference's checkpoints carry rocksdb's per-block
     * checksums, restated here at file granularity) */
    std::string crc_lines;
    char cl[128];
    std::vector<uint8_t> buf;
    for (size_t i = 0; i < e->runs.size(); i++) {
        const RunBuf &r = e->runs[i];
        /* offsets first (host copies give the blob sizes) */
        std::vector<uint64_t> koff(r.n + 1), voff(r.n + 1), sk(r.n);
        HIP_OK(hipMemcpy(koff.data(), r.koff, (r.n + 1) * 8, hipMemcpyDeviceToHost));
        HIP_OK(hipMemcpy(voff.data(), r.voff, (r.n + 1) * 8, hipMemcpyDeviceToHost));
        HIP_OK(hipMemcpy(sk.data(), r.sk, r.n * 8, hipMemcpyDeviceToHost));
        std::string base = path + "/run_" + std::to_string(i);
        auto emit = [&](const char *ext, const void *p, uint64_t n) {
            if (!write_file(base + "." + ext, p, n))
                return false;
            snprintf(cl, sizeof(cl), "crc run_%zu.%s %016llx\n", i, ext,
                     (unsigned long long)host_crc64(p, n));
            crc_lines += cl;
            return true;
        };
        buf.resize(koff[r.n]);
        HIP_OK(hipMemcpy(buf.data(), r.keys, koff[r.n], hipMemcpyDeviceToHost));
        if (!emit("keys", buf.data(), koff[r.n]))
            return RRDB_IO_ERROR;
        if (!emit("koff", koff.data(), (r.n + 1) * 8))
            return RRDB_IO_ERROR;
        buf.resize(voff[r.n]);
        HIP_OK(hipMemcpy(buf.data(), r.vals, voff[r.n], hipMemcpyDeviceToHost));
        if (!emit("vals", buf.data(), voff[r.n]))
            return RRDB_IO_ERROR;
        if (!emit("voff", voff.data(), (r.n + 1) * 8))
            return RRDB_IO_ERROR;
        if (!emit("sk", sk.data(), r.n * 8))
            return RRDB_IO_ERROR;
    }
    FILE *mf = fopen((path + "/MANIFEST").c_str(), "w");
    if (!mf)
        return RRDB_IO_ERROR;
    fprintf(mf, "rrdb-checkpoint 1\ndata_version %u\nnext_seq_floor %llu\nn_runs %d\n",
            e->data_version, (unsigned long long)e->next_seq_floor, (int)e->runs.size());
    for (size_t i = 0; i < e->runs.size(); i++)
        fprintf(mf, "run %zu %llu\n", i, (unsigned long long)e->runs[i].n);
    fputs(crc_lines.c_str(), mf); /* trailing lines; older restorers ignore */
    fclose(mf);
    return RRDB_OK;
}

int32_t rrdb_restore(void *h, const char *dir, uint64_t decree)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    if (!e->runs.empty() || !e->memtable.empty())
        return RRDB_INVALID_ARGUMENT;
    e->activate();
    std::string path = std::string(dir) + "/checkpoint." + std::to_string(decree);
    FILE *mf = fopen((path + "/MANIFEST").c_str(), "r");
    if (!mf)
        return RRDB_IO_ERROR;
    unsigned dv = 1;
    unsigned long long floor_ = 0;
    int n_runs = 0, ver = 0;
    if (fscanf(mf, "rrdb-checkpoint %d\ndata_version %u\nnext_seq_floor %llu\nn_runs %d\n", &ver,
               &dv, &floor_, &n_runs) != 4 ||
        ver != 1 || n_runs < 0) {
        fclose(mf);
        return RRDB_CORRUPTION;
    }
    /* run record counts + optional per-file crc lines (older checkpoints
     * carry none; then crc validation is skipped) */
    std::vector<unsigned long long> mrows((size_t)n_runs, ~0ull);
    std::unordered_map<std::string, uint64_t> mcrc;
    {
        char line[256];
        while (fgets(line, sizeof(line), mf)) {
            unsigned long long a, b;
            char name[128];
            if (sscanf(line, "run %llu %llu", &a, &b) == 2) {
                if (a < (unsigned long long)n_runs)
                    mrows[(size_t)a] = b;
            } else if (sscanf(line, "crc %127s %llx", name, &b) == 2) {
                mcrc[name] = b;
            }
        }
    }
    fclose(mf);
    for (int i = 0; i < n_runs; i++)
        if (mrows[(size_t)i] == ~0ull)
            return RRDB_CORRUPTION; /* MANIFEST missing a run line */
    e->data_version = dv;
    for (int i = 0; i < n_runs; i++) {
        std::string base = path + "/run_" + std::to_string(i);
        std::vector<uint8_t> keys, koff, vals, voff, sk;
        if (!read_file(base + ".keys", keys) || !read_file(base + ".koff", koff) ||
            !read_file(base + ".vals", vals) || !read_file(base + ".voff", voff) ||
            !read_file(base + ".sk", sk))
            return RRDB_IO_ERROR;
        /* integrity validation: crc when present, sizes vs the
         * MANIFEST count, offset monotonicity, key sortedness — a truncated
         * or bit-flipped file returns kCorruption instead of corrupting the
         * engine (the reference restores from checksummed rocksdb
         * checkpoints; same guarantee at our file granularity) */
        auto crc_ok = [&](const char *ext, const std::vector<uint8_t> &data) {
            auto it = mcrc.find("run_" + std::to_string(i) + "." + ext);
            return it == mcrc.end() || it->second == host_crc64(data.data(), data.size());
        };
        if (!crc_ok("keys", keys) || !crc_ok("koff", koff) || !crc_ok("vals", vals) ||
            !crc_ok("voff", voff) || !crc_ok("sk", sk))
            return RRDB_CORRUPTION;
        uint64_t nrec = mrows[(size_t)i];
        if (koff.size() != (nrec + 1) * 8 || voff.size() != (nrec + 1) * 8 ||
            sk.size() != nrec * 8)
            return RRDB_CORRUPTION;
        const uint64_t *ko = (const uint64_t *)koff.data();
        const uint64_t *vo = (const uint64_t *)voff.data();
        if (ko[0] != 0 || vo[0] != 0 || ko[nrec] != keys.size() || vo[nrec] != vals.size())
            return RRDB_CORRUPTION;
        for (uint64_t j = 0; j < nrec; j++)
            if (ko[j + 1] <= ko[j] || vo[j + 1] < vo[j])
                return RRDB_CORRUPTION; /* keys nonempty+increasing offsets */
        for (uint64_t j = 0; j + 1 < nrec; j++) {
            uint64_t la = ko[j + 1] - ko[j], lb = ko[j + 2] - ko[j + 1];
            uint64_t m = la < lb ? la : lb;
            int c = memcmp(keys.data() + ko[j], keys.data() + ko[j + 1], m);
            if (c > 0 || (c == 0 && la >= lb))
                return RRDB_CORRUPTION; /* keys must be strictly increasing */
        }
        RunBuf r;
        r.n = nrec;
        r.keys = e->upload_bytes(keys.data(), keys.size());
        r.koff = (uint64_t *)e->upload_bytes(koff.data(), koff.size());
        r.vals = e->upload_bytes(vals.data(), vals.size());
        r.voff = (uint64_t *)e->upload_bytes(voff.data(), voff.size());
        r.sk = (uint64_t *)e->upload_bytes(sk.data(), sk.size());
        r.fixed_klen = detect_fixed_klen((const uint64_t *)koff.data(), r.n);
        r.fixed_vlen = detect_fixed_vlen((const uint64_t *)voff.data(), r.n);
        r.fixed_vlen_put = detect_fixed_vlen_put((const uint64_t *)voff.data(),
                                                 (const uint64_t *)sk.data(), r.n);
        if (r.n) {
            const uint64_t *ko = (const uint64_t *)koff.data();
            set_pfx(lcp_exact32(keys.data() + ko[0], ko[1] - ko[0],
                                keys.data() + ko[r.n - 1], ko[r.n] - ko[r.n - 1]),
                    &r.pfx_skip, &r.lcp_exact);
        }
        e->build_tails(r); /* rebuilt, not serialized (like blooms) */
        e->build_meta(r);
        e->build_bloom(r);
        e->runs.push_back(r);
    }
    e->next_seq_floor = floor_;
    e->d_runs_dirty = true;
    e->ldst_elig_cache = -1;
    return RRDB_OK;
}

} /* extern "C" */

/* ================= batched multi_get =================
 * One workgroup per request (full-range multi_get per hashkey) — models the
 * reference's concurrent THREAD_POOL_SCAN handlers in a single launch.
 * Requests whose ranges exceed the fused budget (or exotic shared params)
 * fall back to the general per-request path. */
void launch_multi_get_batch(const DevRun *, int, const MgFusedArgs &, const uint8_t *,
                            const uint64_t *, uint64_t, int64_t *, uint8_t *, uint64_t,
                            hipStream_t);
void launch_pack_blobs(const uint8_t *, uint64_t, uint64_t, const uint64_t *, const uint64_t *,
                       uint8_t *, hipStream_t);

extern "C" int32_t rrdb_multi_get_batch(void *h, uint64_t n_req, const uint8_t *hash_keys,
                                        const uint64_t *hk_offs,
                                        const rrdb_multi_get_request *shared,
                                        uint32_t epoch_now, rrdb_result *out)
{
    auto *e = (HipEngine *)h;
    std::lock_guard<std::mutex> g(e->mu);
    if (engine_flush(e) != RRDB_OK) {
        result_init(out);
        out->error = RRDB_IO_ERROR;
        return out->error;
    }
    e->activate();
    /* bulk path: retire the resident serving kernel so its poll windows do
     * not share hardware queues with the 4096-workgroup batch launches */
    e->server_quit_sync();
    e->scratch_reset();
    Arena *a = result_init(out);
    if (shared->n_sort_keys != 0 || shared->sort_key_filter_type < 0 ||
        shared->sort_key_filter_type > 3) {
        out->error = RRDB_INVALID_ARGUMENT;
        return out->error;
    }
    out->group_counts = (uint64_t *)a->alloc(n_req * 8);
    out->group_errors = (int32_t *)a->alloc(n_req * 4);
    /* fused batch only covers the full-range shape (no sortkey bounds, no
     * range-clamping prefix filter); others run per-request */
    bool fused_ok = shared->start_sortkey.len == 0 && shared->stop_sortkey.len == 0 &&
                    shared->start_inclusive &&
                    !(shared->sort_key_filter_type == RRDB_FT_MATCH_PREFIX &&
                      shared->sort_key_filter_pattern.len > 0) &&
                    !e->runs.empty();
    const uint64_t BLOB_STRIDE = 64 << 10;
    std::vector<int64_t> hdrs(n_req * 4, -1);
    uint8_t *d_blobs = nullptr;
    std::vector<uint64_t> used(n_req, 0), pack_off(n_req + 1, 0);
    std::vector<uint8_t> packed_host;
    if (fused_ok && n_req > 0) {
        uint32_t max_kv_count = e->mg_max_iter_count;
        if (shared->max_kv_count > 0 && (uint32_t)shared->max_kv_count < max_kv_count)
            max_kv_count = (uint32_t)shared->max_kv_count;
        int64_t max_kv_size = shared->max_kv_size > 0 ? shared->max_kv_size : INT32_MAX;
        MgFusedArgs fa{};
        fa.start_inclusive = 1;
        fa.stop_inclusive = 0;
        fa.reverse = shared->reverse;
        fa.no_value = shared->no_value;
        fa.max_kv_count = max_kv_count;
        fa.max_iteration_count = e->mg_max_iter_count;
        fa.max_iteration_size = std::min<int64_t>(
            max_kv_size, e->mg_max_iter_size > 0 ? (int64_t)e->mg_max_iter_size : INT32_MAX);
        fa.sk_ft = shared->sort_key_filter_type;
        fa.sk_pat_len = shared->sort_key_filter_pattern.len;
        fa.sk_pat = e->upload_tmp(shared->sort_key_filter_pattern.data, fa.sk_pat_len);
        fa.epoch_now = epoch_now;
        fa.data_version = e->data_version;
        /* per-request hash_key_skip (= start key length) is computed
         * in-kernel so size caps count sortkey+value, as the reference does */
        uint8_t *d_hks = e->upload_tmp(hash_keys, hk_offs[n_req]);
        uint64_t *d_offs = (uint64_t *)e->upload_tmp(hk_offs, (n_req + 1) * 8);
        int64_t *d_hdrs = e->talloc<int64_t>(n_req * 4 * 8);
        HIP_OK(hipMemsetAsync(d_hdrs, 0xFF, n_req * 4 * 8, e->stream)); /* -1 fill */
        d_blobs = e->talloc<uint8_t>(n_req * BLOB_STRIDE);
        launch_multi_get_batch(e->dev_runs(), (int)e->runs.size(), fa, d_hks, d_offs, n_req,
                               d_hdrs, d_blobs, BLOB_STRIDE, e->stream);
        HIP_OK(hipMemcpyAsync(hdrs.data(), d_hdrs, n_req * 4 * 8, hipMemcpyDeviceToHost,
                              e->stream));
        HIP_OK(hipStreamSynchronize(e->stream));
        /* pack used regions, one D2H */
        uint64_t total = 0;
        for (uint64_t i = 0; i < n_req; i++) {
            if (hdrs[i * 4] >= 0) {
                uint64_t m = (uint64_t)hdrs[i * 4];
                used[i] = 2 * (m + 1) * 8 + (uint64_t)hdrs[i * 4 + 2] + (uint64_t)hdrs[i * 4 + 3];
            }
            pack_off[i] = total;
            total += used[i];
        }
        pack_off[n_req] = total;
        bool all_fused = true;
        for (uint64_t i = 0; i < n_req; i++)
            if (hdrs[i * 4] < 0)
                all_fused = false;
        if (total && shared->on_device_out && all_fused) {
            /* device-out: pack into caller-owned device memory, no host
             * marshal (the batched e2e limiter on slow-host boxes) */
            uint8_t *d_packed = nullptr;
            uint64_t *d_poff_out = nullptr;
            HIP_OK(hipMalloc(&d_packed, total));
            HIP_OK(hipMalloc(&d_poff_out, (n_req + 1) * 8));
            HIP_OK(hipMemcpyAsync(d_poff_out, pack_off.data(), (n_req + 1) * 8,
                                  hipMemcpyHostToDevice, e->stream));
            uint64_t *d_used = (uint64_t *)e->upload_tmp(used.data(), n_req * 8);
            uint64_t *d_poff = (uint64_t *)e->upload_tmp(pack_off.data(), (n_req + 1) * 8);
            launch_pack_blobs(d_blobs, BLOB_STRIDE, n_req, d_used, d_poff, d_packed, e->stream);
            HIP_OK(hipStreamSynchronize(e->stream));
            uint64_t rows = 0;
            for (uint64_t i = 0; i < n_req; i++) {
                out->group_counts[i] = (uint64_t)hdrs[i * 4];
                out->group_errors[i] = hdrs[i * 4 + 1] ? RRDB_OK : RRDB_INCOMPLETE;
                rows += (uint64_t)hdrs[i * 4];
            }
            out->dev_vals = d_packed;
            out->dev_val_offs = d_poff_out;
            a->dev_ptrs.insert(a->dev_ptrs.end(), {(void *)d_packed, (void *)d_poff_out});
            out->count = rows;
            out->error = RRDB_OK;
            return RRDB_OK;
        }
        if (total) {
            uint64_t *d_used = (uint64_t *)e->upload_tmp(used.data(), n_req * 8);
            uint64_t *d_poff = (uint64_t *)e->upload_tmp(pack_off.data(), (n_req + 1) * 8);
            uint8_t *d_packed = e->talloc<uint8_t>(total);
            launch_pack_blobs(d_blobs, BLOB_STRIDE, n_req, d_used, d_poff, d_packed, e->stream);
            packed_host.resize(total);
            e->d2h(packed_host.data(), d_packed, total);
        }
    }
    /* count rows (fused + fallback) */
    uint64_t total_rows = 0;
    std::vector<rrdb_result> fb(n_req);
    std::vector<uint8_t> is_fb(n_req, 0);
    for (uint64_t i = 0; i < n_req; i++) {
        if (hdrs[i * 4] >= 0) {
            total_rows += (uint64_t)hdrs[i * 4];
        } else {
            /* general path per request (rare) */
            is_fb[i] = 1;
            rrdb_multi_get_request req = *shared;
            req.hash_key.data = hash_keys + hk_offs[i];
            req.hash_key.len = hk_offs[i + 1] - hk_offs[i];
            multi_get_locked(h, &req, epoch_now, &fb[i]);
            total_rows += fb[i].count;
        }
    }
    out->keys = (rrdb_slice *)a->alloc((total_rows ? total_rows : 1) * sizeof(rrdb_slice));
    out->values = (rrdb_slice *)a->alloc((total_rows ? total_rows : 1) * sizeof(rrdb_slice));
    uint8_t *pb = nullptr;
    if (!packed_host.empty()) {
        pb = (uint8_t *)a->alloc(packed_host.size());
        memcpy(pb, packed_host.data(), packed_host.size());
    }
    uint64_t m = 0;
    for (uint64_t i = 0; i < n_req; i++) {
        if (!is_fb[i]) {
            uint64_t n = (uint64_t)hdrs[i * 4];
            uint64_t kb = (uint64_t)hdrs[i * 4 + 2];
            out->group_counts[i] = n;
            out->group_errors[i] = hdrs[i * 4 + 1] ? RRDB_OK : RRDB_INCOMPLETE;
            if (n == 0)
                continue;
            uint8_t *base = pb + pack_off[i];
            const uint64_t *koffs = (const uint64_t *)base;
            const uint64_t *voffs = koffs + (n + 1);
            uint8_t *keys_b = base + 2 * (n + 1) * 8;
            uint8_t *vals_b = keys_b + kb;
            for (uint64_t j = 0; j < n; j++) {
                out->keys[m] = {keys_b + koffs[j], koffs[j + 1] - koffs[j]};
                out->values[m] = {vals_b + voffs[j], voffs[j + 1] - voffs[j]};
                m++;
            }
        } else {
            out->group_counts[i] = fb[i].count;
            out->group_errors[i] = fb[i].error;
            for (uint64_t j = 0; j < fb[i].count; j++) {
                uint64_t kl = fb[i].keys[j].len, vl = fb[i].values[j].len;
                uint8_t *kc = (uint8_t *)a->alloc(kl ? kl : 1);
                memcpy(kc, fb[i].keys[j].data, kl);
                uint8_t *vc = (uint8_t *)a->alloc(vl ? vl : 1);
                memcpy(vc, fb[i].values[j].data, vl);
                out->keys[m] = {kc, kl};
                out->values[m] = {vc, vl};
                m++;
            }
            rrdb_free_result(&fb[i]);
        }
    }
    out->count = m;
    out->error = RRDB_OK;
    return RRDB_OK;
}
