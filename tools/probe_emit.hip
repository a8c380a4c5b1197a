/* probe_emit.hip — isolated A/B of emit-copy strategies for the compaction
 * emit kernel shape: gather ~130B records (18B key + 112B value) from 8
 * interleaved source runs in rank order, write packed output.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 tools/probe_emit.hip -o gpurun_out/probe_emit
 * Run (GPU box): ./gpurun_out/probe_emit
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <vector>
#include <algorithm>
#include <random>

#define WAVE 64
#define BLOCK 256
#define HIP_OK(x)                                                                                  \
    do {                                                                                           \
        hipError_t e_ = (x);                                                                       \
        if (e_ != hipSuccess) {                                                                    \
            printf("ERR %s @%d\n", hipGetErrorString(e_), __LINE__);                               \
            abort();                                                                               \
        }                                                                                          \
    } while (0)

struct Rec {
    const uint8_t *src; /* device ptr to record bytes */
};

/* v0: wave-per-record, byte lanes */
__global__ void k_v0(const uint64_t *src_off, const uint8_t *src, const uint64_t *dst_off,
                     uint8_t *dst, uint64_t n, uint64_t reclen)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    for (uint64_t p = wave; p < n; p += nwaves) {
        const uint8_t *s = src + src_off[p];
        uint8_t *d = dst + dst_off[p];
        for (uint64_t b = lane; b < reclen; b += WAVE)
            d[b] = s[b];
    }
}

/* v1: wave-per-record, 4B lanes */
__global__ void k_v1(const uint64_t *src_off, const uint8_t *src, const uint64_t *dst_off,
                     uint8_t *dst, uint64_t n, uint64_t reclen)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    for (uint64_t p = wave; p < n; p += nwaves) {
        const uint8_t *s = src + src_off[p];
        uint8_t *d = dst + dst_off[p];
        uint64_t n4 = reclen >> 2;
        for (uint64_t c = lane; c < n4; c += WAVE) {
            uint32_t w;
            __builtin_memcpy(&w, s + 4 * c, 4);
            __builtin_memcpy(d + 4 * c, &w, 4);
        }
        for (uint64_t b = (n4 << 2) + lane; b < reclen; b += WAVE)
            d[b] = s[b];
    }
}

/* v2: 2 records per wave in flight (lanes 0-31 record A, 32-63 record B), 4B */
__global__ void k_v2(const uint64_t *src_off, const uint8_t *src, const uint64_t *dst_off,
                     uint8_t *dst, uint64_t n, uint64_t reclen)
{
    uint64_t wave = (blockIdx.x * (uint64_t)blockDim.x + threadIdx.x) / WAVE;
    uint64_t nwaves = (gridDim.x * (uint64_t)blockDim.x) / WAVE;
    int lane = threadIdx.x % WAVE;
    int half = lane >> 5, hl = lane & 31;
    for (uint64_t p = wave * 2 + half; p < n; p += nwaves * 2) {
        const uint8_t *s = src + src_off[p];
        uint8_t *d = dst + dst_off[p];
        uint64_t n4 = reclen >> 2;
        for (uint64_t c = hl; c < n4; c += 32) {
            uint32_t w;
            __builtin_memcpy(&w, s + 4 * c, 4);
            __builtin_memcpy(d + 4 * c, &w, 4);
        }
        for (uint64_t b = (n4 << 2) + hl; b < reclen; b += 32)
            d[b] = s[b];
    }
}

/* v3: thread-per-16B-chunk over the whole output (requires fixed reclen):
 * perfectly coalesced global index -> (record, offset) by division */
__global__ void k_v3(const uint64_t *src_off, const uint8_t *src, const uint64_t *dst_off,
                     uint8_t *dst, uint64_t n, uint64_t reclen)
{
    uint64_t chunks_per_rec = (reclen + 15) / 16;
    uint64_t total = n * chunks_per_rec;
    for (uint64_t t = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; t < total;
         t += gridDim.x * (uint64_t)blockDim.x) {
        uint64_t p = t / chunks_per_rec;
        uint64_t c = t % chunks_per_rec;
        const uint8_t *s = src + src_off[p] + c * 16;
        uint8_t *d = dst + dst_off[p] + c * 16;
        uint64_t m = reclen - c * 16;
        if (m >= 16) {
            uint32_t w0, w1, w2, w3;
            __builtin_memcpy(&w0, s, 4);
            __builtin_memcpy(&w1, s + 4, 4);
            __builtin_memcpy(&w2, s + 8, 4);
            __builtin_memcpy(&w3, s + 12, 4);
            __builtin_memcpy(d, &w0, 4);
            __builtin_memcpy(d + 4, &w1, 4);
            __builtin_memcpy(d + 8, &w2, 4);
            __builtin_memcpy(d + 12, &w3, 4);
        } else {
            for (uint64_t b = 0; b < m; b++)
                d[b] = s[b];
        }
    }
}

/* v4: thread-per-record, 16B vector loop (uncoalesced across lanes but max ILP) */
__global__ void k_v4(const uint64_t *src_off, const uint8_t *src, const uint64_t *dst_off,
                     uint8_t *dst, uint64_t n, uint64_t reclen)
{
    for (uint64_t p = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; p < n;
         p += gridDim.x * (uint64_t)blockDim.x) {
        const uint8_t *s = src + src_off[p];
        uint8_t *d = dst + dst_off[p];
        uint64_t b = 0;
        for (; b + 16 <= reclen; b += 16) {
            uint32_t w[4];
            __builtin_memcpy(w, s + b, 16);
            __builtin_memcpy(d + b, w, 16);
        }
        for (; b < reclen; b++)
            d[b] = s[b];
    }
}

int main()
{
    const uint64_t N = 3500000, RECLEN = 130, RUNS = 8;
    const uint64_t BYTES = N * RECLEN;
    /* source: 8 run regions; rank order interleaves them */
    uint8_t *d_src, *d_dst;
    uint64_t *d_soff, *d_doff;
    HIP_OK(hipMalloc(&d_src, BYTES));
    HIP_OK(hipMalloc(&d_dst, BYTES));
    HIP_OK(hipMalloc(&d_soff, N * 8));
    HIP_OK(hipMalloc(&d_doff, N * 8));
    std::vector<uint64_t> soff(N), doff(N);
    std::mt19937_64 rng(1);
    /* records assigned round-robin-ish to runs, in rank order */
    std::vector<uint64_t> run_pos(RUNS);
    uint64_t per_run = N / RUNS;
    for (uint64_t r = 0; r < RUNS; r++)
        run_pos[r] = r * per_run * RECLEN;
    for (uint64_t p = 0; p < N; p++) {
        uint64_t r = rng() % RUNS;
        while (run_pos[r] >= (r + 1) * per_run * RECLEN && r + 1 < RUNS)
            r++;
        if (run_pos[r] >= (r + 1) * per_run * RECLEN)
            r = 0;
        soff[p] = run_pos[r];
        run_pos[r] += RECLEN;
        doff[p] = p * RECLEN;
    }
    HIP_OK(hipMemcpy(d_soff, soff.data(), N * 8, hipMemcpyHostToDevice));
    HIP_OK(hipMemcpy(d_doff, doff.data(), N * 8, hipMemcpyHostToDevice));

    auto bench = [&](const char *name, auto kern, int grid) {
        hipEvent_t a, b;
        hipEventCreate(&a);
        hipEventCreate(&b);
        /* warmup */
        kern<<<grid, BLOCK>>>(d_soff, d_src, d_doff, d_dst, N, RECLEN);
        HIP_OK(hipDeviceSynchronize());
        hipEventRecord(a);
        for (int i = 0; i < 5; i++)
            kern<<<grid, BLOCK>>>(d_soff, d_src, d_doff, d_dst, N, RECLEN);
        hipEventRecord(b);
        HIP_OK(hipDeviceSynchronize());
        float ms;
        hipEventElapsedTime(&ms, a, b);
        ms /= 5;
        printf("%-28s grid=%5d  %7.3f ms  %7.1f GB/s (2x%luMB)\n", name, grid, ms,
               2.0 * BYTES / (ms * 1e-3) / 1e9, BYTES >> 20);
        hipEventDestroy(a);
        hipEventDestroy(b);
    };
    for (int grid : {2048, 8192}) {
        bench("v0 wave/rec byte lanes", k_v0, grid);
        bench("v1 wave/rec 4B lanes", k_v1, grid);
        bench("v2 2rec/wave 4B", k_v2, grid);
        bench("v3 thread/16B chunk", k_v3, grid);
        bench("v4 thread/rec 16B ILP", k_v4, grid);
    }
    return 0;
}
