/* bench_mg.cpp — C++ driver for the multi_get latency path (the reference's
 * host is C++; the Python ctypes loop adds ~0.1ms/call of interpreter
 * overhead, so this measures the real C-ABI per-call cost).
 *
 * Build: hipcc -O3 tools/bench_mg.cpp -o bin/bench_mg -ldl
 * Run (GPU box): ./bin/bench_mg incubator_pegasus_amd/csrc/librrdb_hip.so [n_hashkeys] [calls]
 */
#include <dlfcn.h>
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <chrono>
#include <string>
#include <vector>

#include "../include/rrdb_engine.h"

typedef void *(*open_fn)(int32_t, int32_t, int32_t);
typedef int32_t (*ingest_fn)(void *, const uint8_t *, const uint64_t *, const uint8_t *,
                             const uint64_t *, const uint64_t *, uint64_t);
typedef int32_t (*mg_fn)(void *, const rrdb_multi_get_request *, uint32_t, rrdb_result *);
typedef int32_t (*mgb_fn)(void *, uint64_t, const uint8_t *, const uint64_t *,
                          const rrdb_multi_get_request *, uint32_t, rrdb_result *);
typedef void (*free_fn)(rrdb_result *);
typedef int32_t (*envs_fn)(void *, const char *const *, const char *const *, int32_t);

static uint64_t splitmix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

int main(int argc, char **argv)
{
    const char *so = argc > 1 ? argv[1] : "incubator_pegasus_amd/csrc/librrdb_hip.so";
    uint64_t n_hash = argc > 2 ? strtoull(argv[2], nullptr, 10) : 200000;
    uint64_t calls = argc > 3 ? strtoull(argv[3], nullptr, 10) : 20000;
    const int SKS = 10; /* sortkeys per hashkey */
    void *lib = dlopen(so, RTLD_NOW);
    if (!lib) {
        printf("dlopen failed: %s\n", dlerror());
        return 1;
    }
    auto rrdb_open_ = (open_fn)dlsym(lib, "rrdb_open");
    auto rrdb_ingest = (ingest_fn)dlsym(lib, "rrdb_ingest_run");
    auto rrdb_mg = (mg_fn)dlsym(lib, "rrdb_multi_get");
    auto rrdb_mgb = (mgb_fn)dlsym(lib, "rrdb_multi_get_batch");
    auto rrdb_free = (free_fn)dlsym(lib, "rrdb_free_result");
    auto rrdb_envs = (envs_fn)dlsym(lib, "rrdb_set_envs");
    void *h = rrdb_open_(9, 0, 0);
    if (!h) {
        printf("open failed (no GPU?)\n");
        return 1;
    }
    if (argc > 4 && rrdb_envs) { /* e.g. "engine.mg_persist=off" */
        char kv[128];
        snprintf(kv, sizeof(kv), "%s", argv[4]);
        char *eq = strchr(kv, '=');
        if (eq) {
            *eq = 0;
            const char *ks[1] = {kv};
            const char *vs[1] = {eq + 1};
            rrdb_envs(h, ks, vs, 1);
        }
    }
    /* build one sorted run: n_hash hashkeys x SKS sortkeys, v1 values */
    uint64_t n = n_hash * SKS;
    std::string keys, vals;
    std::vector<uint64_t> koff{0}, voff{0}, sk;
    keys.reserve(n * 26);
    vals.reserve(n * 112);
    char hk[17], skb[9];
    for (uint64_t i = 0; i < n_hash; i++) {
        snprintf(hk, sizeof(hk), "u:%014llu", (unsigned long long)i);
        for (int j = 0; j < SKS; j++) {
            snprintf(skb, sizeof(skb), "%08d", j);
            keys.push_back(0);
            keys.push_back(16);
            keys.append(hk, 16);
            keys.append(skb, 8);
            koff.push_back(keys.size());
            char v[112];
            memset(v, 0, 12);
            for (int b = 12; b < 112; b++)
                v[b] = (char)(splitmix64(i * 112 + b) & 0xFF);
            vals.append(v, 112);
            voff.push_back(vals.size());
            sk.push_back(((i * SKS + j + 1) << 1));
        }
    }
    int32_t st = rrdb_ingest(h, (const uint8_t *)keys.data(), koff.data(),
                             (const uint8_t *)vals.data(), voff.data(), sk.data(), n);
    if (st != 0) {
        printf("ingest failed %d\n", st);
        return 1;
    }
    /* zipfian-ish hashkey picks via splitmix (uniform is fine for latency) */
    uint64_t rows = 0;
    auto t0 = std::chrono::steady_clock::now();
    for (uint64_t c = 0; c < calls; c++) {
        uint64_t id = splitmix64(c) % n_hash;
        snprintf(hk, sizeof(hk), "u:%014llu", (unsigned long long)id);
        rrdb_multi_get_request req{};
        req.hash_key = {(const uint8_t *)hk, 16};
        req.start_inclusive = 1;
        req.stop_inclusive = 0;
        req.max_kv_count = -1;
        req.max_kv_size = -1;
        rrdb_result res;
        st = rrdb_mg(h, &req, 1000000, &res);
        if (st != 0) {
            printf("mg failed %d\n", st);
            return 1;
        }
        rows += res.count;
        rrdb_free(&res);
    }
    auto el = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    /* batched: one rrdb_multi_get_batch call per 4096 hashkeys (the
     * THREAD_POOL_SCAN concurrency model as one launch) */
    const uint64_t B = 4096, NB = 8;
    std::string bhks;
    std::vector<uint64_t> boffs{0};
    bhks.reserve(B * 16);
    uint64_t brows = 0, bcalls = 0;
    auto bt0 = std::chrono::steady_clock::now();
    for (uint64_t pass = 0; pass < NB; pass++) {
        bhks.clear();
        boffs.resize(1);
        for (uint64_t i = 0; i < B; i++) {
            uint64_t id = splitmix64(pass * B + i + 7777) % n_hash;
            snprintf(hk, sizeof(hk), "u:%014llu", (unsigned long long)id);
            bhks.append(hk, 16);
            boffs.push_back(bhks.size());
        }
        rrdb_multi_get_request shared{};
        shared.start_inclusive = 1;
        shared.max_kv_count = -1;
        shared.max_kv_size = -1;
        rrdb_result res;
        st = rrdb_mgb(h, B, (const uint8_t *)bhks.data(), boffs.data(), &shared, 1000000,
                      &res);
        if (st != 0) {
            printf("mg_batch failed %d\n", st);
            return 1;
        }
        brows += res.count;
        bcalls += B;
        rrdb_free(&res);
    }
    auto bel = std::chrono::duration<double>(std::chrono::steady_clock::now() - bt0).count();
    /* device-out batched: results stay in HBM (the serving shim would
     * serialize from a pinned mirror); measures the ABI rate without the
     * host slice marshal */
    uint64_t drows = 0, dcalls = 0;
    auto dt0 = std::chrono::steady_clock::now();
    for (uint64_t pass = 0; pass < NB; pass++) {
        bhks.clear();
        boffs.resize(1);
        for (uint64_t i = 0; i < B; i++) {
            uint64_t id = splitmix64(pass * B + i + 555) % n_hash;
            snprintf(hk, sizeof(hk), "u:%014llu", (unsigned long long)id);
            bhks.append(hk, 16);
            boffs.push_back(bhks.size());
        }
        rrdb_multi_get_request shared{};
        shared.start_inclusive = 1;
        shared.max_kv_count = -1;
        shared.max_kv_size = -1;
        shared.on_device_out = 1;
        rrdb_result res;
        st = rrdb_mgb(h, B, (const uint8_t *)bhks.data(), boffs.data(), &shared, 1000000,
                      &res);
        if (st != 0) {
            printf("mg_batch devout failed %d\n", st);
            return 1;
        }
        drows += res.count;
        dcalls += B;
        rrdb_free(&res);
    }
    auto del = std::chrono::duration<double>(std::chrono::steady_clock::now() - dt0).count();
    printf("{\"multi_get_ops_per_s\": %.1f, \"rows_per_s\": %.1f, \"rows\": %llu, "
           "\"us_per_call\": %.1f, \"calls\": %llu, "
           "\"batched_ops_per_s\": %.1f, \"batched_rows_per_s\": %.1f, "
           "\"batched_rows\": %llu, "
           "\"batched_devout_ops_per_s\": %.1f, \"batched_devout_rows_per_s\": %.1f, "
           "\"host\": \"c++\"}\n",
           calls / el, rows / el, (unsigned long long)rows, el * 1e6 / calls,
           (unsigned long long)calls, bcalls / bel, brows / bel,
           (unsigned long long)brows, dcalls / del, drows / del);
    return 0;
}
