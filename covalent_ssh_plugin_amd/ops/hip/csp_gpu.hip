// CDNA4 (gfx950 / MI355X) GPU library for the covalent SSH executor.
//
// New MI355X-native component with no reference counterpart (the
// reference plugin, /root/reference/covalent_ssh_plugin, has zero native
// code — SURVEY.md §2.4).  Provides, behind a plain C ABI consumed via
// ctypes (so the remote stub needs no torch and no Python extension ABI):
//
//   * csp_probe_json  — device probe + measured HBM bandwidth and bf16
//                       MFMA throughput from the two hand-written
//                       warm-up kernels below,
//   * csp_warmup      — clock/cache warm-up (MFMA spin + HBM sweep)
//                       bounded by a millisecond budget,
//   * csp_host_alloc / csp_host_free / csp_staging_get /
//     csp_memcpy_d2h / csp_memcpy_h2d
//                     — hipHostMalloc-pinned staging for large tensor
//                       results on the SFTP/stdout return path.
//
// Kernel design notes (per /opt/skills/guides/MI355X_MICROARCH.md):
//   - wave64; 256-thread blocks = 4 waves;
//   - the MFMA spin uses v_mfma_f32_32x32x16_bf16 (32768 FLOP/instr,
//     ~32 cyc back-to-back issue per SIMD) with 4 independent
//     accumulators per wave so issue is never dependency-stalled;
//   - the HBM sweep reads+writes float4 (16 B/lane) grid-stride with
//     >=1024 workgroups so all 8 XCDs are saturated;
//   - gfx950-only: built with --offload-arch=gfx950, no other targets.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdio>
#include <cstring>
#include <algorithm>
#include <mutex>
#include <vector>
#include <utility>

// ---------------------------------------------------------------------------
// Error plumbing
// ---------------------------------------------------------------------------

static thread_local char g_err[512];

static int set_err(const char* what, hipError_t e) {
    snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
    return -1;
}

#define HIP_TRY(expr)                                    \
    do {                                                 \
        hipError_t _e = (expr);                          \
        if (_e != hipSuccess) return set_err(#expr, _e); \
    } while (0)

extern "C" const char* csp_last_error() { return g_err; }

// ---------------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------------

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// MFMA spin: each wave issues `iters` v_mfma_f32_32x32x16_bf16 across 4
// independent accumulators (issue-rate bound, no dependency stall).
__global__ __launch_bounds__(256) void csp_mfma_spin_kernel(
    float* __restrict__ out, int iters) {
    bf16x8 a, b;
    const int lane = threadIdx.x;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
        a[i] = (__bf16)((lane + i) & 7);
        b[i] = (__bf16)((lane * 3 + i) & 7);
    }
    f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
    for (int i = 0; i < iters; i += 4) {
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
        acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
    }
    float s = 0.f;
#pragma unroll
    for (int i = 0; i < 16; ++i) s += acc0[i] + acc1[i] + acc2[i] + acc3[i];
    if (out && s == -1.0f) out[blockIdx.x] = s;  // never true: defeats DCE only
}

// HBM sweep: grid-stride float4 copy (read 16 B + write 16 B per lane per
// element).  Doubles as the bandwidth probe and the HBM warm-up.
// Template knobs measured on-box (tools: csp_hbm_bench):
//   UNROLL — independent loads in flight per lane (ILP on top of the
//            grid's TLP; HBM latency ~900 cyc wants >1 per lane),
//   NT     — nontemporal load+store (aux=2): a pure streaming copy never
//            re-reads, so bypassing L2 residency avoids evicting it.
template <int UNROLL, bool NT>
__global__ __launch_bounds__(256) void csp_hbm_sweep_t(
    const float4* __restrict__ src, float4* __restrict__ dst, size_t n4) {
    const size_t stride = (size_t)gridDim.x * blockDim.x;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    // nontemporal builtins need a true vector type, not HIP's float4 class
    typedef float f32x4v __attribute__((ext_vector_type(4)));
    const f32x4v* __restrict__ srcv = reinterpret_cast<const f32x4v*>(src);
    f32x4v* __restrict__ dstv = reinterpret_cast<f32x4v*>(dst);
    for (; i + (UNROLL - 1) * stride < n4; i += UNROLL * stride) {
        f32x4v v[UNROLL];
#pragma unroll
        for (int u = 0; u < UNROLL; ++u)
            v[u] = NT ? __builtin_nontemporal_load(&srcv[i + u * stride])
                      : srcv[i + u * stride];
#pragma unroll
        for (int u = 0; u < UNROLL; ++u) {
            if (NT)
                __builtin_nontemporal_store(v[u], &dstv[i + u * stride]);
            else
                dstv[i + u * stride] = v[u];
        }
    }
    for (; i < n4; i += stride) dst[i] = src[i];
}

using sweep_fn = void (*)(const float4*, float4*, size_t);

static sweep_fn sweep_variant(int variant) {
    switch (variant) {
        case 1: return csp_hbm_sweep_t<4, false>;
        case 2: return csp_hbm_sweep_t<1, true>;
        case 3: return csp_hbm_sweep_t<4, true>;
        case 4: return csp_hbm_sweep_t<8, true>;
        default: return csp_hbm_sweep_t<1, false>;
    }
}

// Default variant for probe/warm-up; selected from on-box measurements
// (profiles/hbm_sweep_variants.md): nontemporal/unroll-1 at 1024
// workgroups (4 per CU), fixed copy direction = 6.0 TB/s, 95% of the
// measured float4-copy ceiling.
static int g_sweep_variant = 2;

extern "C" void csp_set_sweep_variant(int v) { g_sweep_variant = v; }

// ---------------------------------------------------------------------------
// Device / measurement helpers
// ---------------------------------------------------------------------------

extern "C" int csp_device_count() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) {
        set_err("hipGetDeviceCount", e);
        return -1;
    }
    return n;
}

static int run_mfma_spin(int iters_per_wave, int blocks, float* ms_out) {
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL(csp_mfma_spin_kernel, dim3(blocks), dim3(256), 0, 0,
                       nullptr, iters_per_wave);
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    HIP_TRY(hipEventElapsedTime(ms_out, t0, t1));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

static int run_hbm_sweep(float4* buf_a, float4* buf_b, size_t n4, int reps,
                         float* ms_out, int variant = -1, int blocks = 1024,
                         bool pingpong = false) {
    sweep_fn kern = sweep_variant(variant < 0 ? g_sweep_variant : variant);
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    for (int r = 0; r < reps; ++r) {
        // ping-pong alternates direction; fixed always copies a -> b
        const float4* s = (pingpong && (r & 1)) ? buf_b : buf_a;
        float4* d = (pingpong && (r & 1)) ? buf_a : buf_b;
        hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, 0, s, d, n4);
    }
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    HIP_TRY(hipEventElapsedTime(ms_out, t0, t1));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

// FLOP per v_mfma_f32_32x32x16_bf16: 2*32*32*16
static const double kFlopPerMfma = 32768.0;

extern "C" int csp_warmup(int device, int budget_ms) {
    HIP_TRY(hipSetDevice(device));
    if (budget_ms <= 0) budget_ms = 50;
    // One short HBM touch (256 MiB) + an MFMA spin sized to the budget.
    size_t bytes = 256u * 1024u * 1024u;
    size_t n4 = bytes / sizeof(float4);
    float4 *a = nullptr, *b = nullptr;
    HIP_TRY(hipMalloc(&a, bytes));
    HIP_TRY(hipMalloc(&b, bytes));
    HIP_TRY(hipMemsetAsync(a, 1, bytes));
    float ms = 0.f;
    int rc = run_hbm_sweep(a, b, n4, 2, &ms);
    if (rc == 0) {
        // ~40 TFLOP per budget at low clocks -> iters sized for ~budget_ms
        // at an assumed sub-peak 1 PF/s warm-from-idle rate.
        double flops_target = 1.0e12 * (budget_ms / 1000.0) * 2.0;
        int blocks = 1024;
        double waves = (double)blocks * 4.0;
        int iters = (int)(flops_target / (waves * kFlopPerMfma));
        if (iters < 64) iters = 64;
        rc = run_mfma_spin(iters, blocks, &ms);
    }
    (void)hipFree(a);
    (void)hipFree(b);
    return rc;
}

// Properties-only probe: no measurement kernels, no large allocations —
// cheap enough for per-task use (the stub's prologue).  Measured
// bandwidth/TFLOPs fields are 0; csp_probe_json fills them.
extern "C" int csp_probe_props_json(int device, char* buf, size_t buflen) {
    HIP_TRY(hipSetDevice(device));
    hipDeviceProp_t props;
    HIP_TRY(hipGetDeviceProperties(&props, device));
    size_t mem_free = 0, mem_total = 0;
    HIP_TRY(hipMemGetInfo(&mem_free, &mem_total));
    int written = snprintf(
        buf, buflen,
        "{\"name\":\"%s\",\"gcn_arch\":\"%s\",\"device\":%d,"
        "\"cu_count\":%d,\"max_clock_mhz\":%d,\"lds_per_cu_kb\":%zu,"
        "\"wavefront_size\":%d,\"hbm_total_gb\":%.1f,\"hbm_free_gb\":%.1f,"
        "\"hbm_bw_gbps\":0,\"mfma_bf16_tflops\":0}",
        props.name, props.gcnArchName, device, props.multiProcessorCount,
        props.clockRate / 1000, (size_t)props.maxSharedMemoryPerMultiProcessor / 1024,
        props.warpSize, (double)mem_total / 1.0e9, (double)mem_free / 1.0e9);
    if (written < 0 || (size_t)written >= buflen) {
        snprintf(g_err, sizeof(g_err), "probe json buffer too small");
        return -2;
    }
    return 0;
}

// Lightweight per-task telemetry (worker meta): HBM occupancy without
// the full props query.  One hipMemGetInfo round trip.
extern "C" int csp_mem_info(int device, double* free_gb, double* total_gb) {
    HIP_TRY(hipSetDevice(device));
    size_t mem_free = 0, mem_total = 0;
    HIP_TRY(hipMemGetInfo(&mem_free, &mem_total));
    if (free_gb) *free_gb = (double)mem_free / 1.0e9;
    if (total_gb) *total_gb = (double)mem_total / 1.0e9;
    return 0;
}

extern "C" int csp_probe_json(int device, char* buf, size_t buflen) {
    HIP_TRY(hipSetDevice(device));
    hipDeviceProp_t props;
    HIP_TRY(hipGetDeviceProperties(&props, device));
    size_t mem_free = 0, mem_total = 0;
    HIP_TRY(hipMemGetInfo(&mem_free, &mem_total));

    // --- measured HBM bandwidth: 1 GiB ping-pong copy ------------------
    size_t bytes = 1024u * 1024u * 1024u;
    size_t n4 = bytes / sizeof(float4);
    float4 *a = nullptr, *b = nullptr;
    HIP_TRY(hipMalloc(&a, bytes));
    HIP_TRY(hipMalloc(&b, bytes));
    HIP_TRY(hipMemsetAsync(a, 1, bytes));
    float warm_ms = 0.f, ms = 0.f;
    int rc = run_hbm_sweep(a, b, n4, 2, &warm_ms);  // warm
    if (rc == 0) rc = run_hbm_sweep(a, b, n4, 6, &ms);
    (void)hipFree(a);
    (void)hipFree(b);
    if (rc != 0) return rc;
    // each rep reads + writes `bytes`
    double hbm_gbps = (2.0 * (double)bytes * 6.0 / 1.0e9) / ((double)ms / 1000.0);

    // --- measured bf16 MFMA throughput ---------------------------------
    const int blocks = 2048;  // 8 WGs/CU; best measured (profiles/)
    const int iters = 8192;   // per wave
    rc = run_mfma_spin(iters / 4, blocks, &warm_ms);  // warm clocks
    if (rc != 0) return rc;
    rc = run_mfma_spin(iters, blocks, &ms);
    if (rc != 0) return rc;
    double waves = (double)blocks * 4.0;
    double tflops = waves * (double)iters * kFlopPerMfma / ((double)ms / 1000.0) / 1.0e12;

    int written = snprintf(
        buf, buflen,
        "{\"name\":\"%s\",\"gcn_arch\":\"%s\",\"device\":%d,"
        "\"cu_count\":%d,\"max_clock_mhz\":%d,\"lds_per_cu_kb\":%zu,"
        "\"wavefront_size\":%d,\"hbm_total_gb\":%.1f,\"hbm_free_gb\":%.1f,"
        "\"hbm_bw_gbps\":%.1f,\"mfma_bf16_tflops\":%.1f}",
        props.name, props.gcnArchName, device, props.multiProcessorCount,
        props.clockRate / 1000, (size_t)props.maxSharedMemoryPerMultiProcessor / 1024,
        props.warpSize, (double)mem_total / 1.0e9, (double)mem_free / 1.0e9,
        hbm_gbps, tflops);
    if (written < 0 || (size_t)written >= buflen) {
        snprintf(g_err, sizeof(g_err), "probe json buffer too small");
        return -2;
    }
    return 0;
}

// ---------------------------------------------------------------------------
// hipHostMalloc-pinned staging
// ---------------------------------------------------------------------------

extern "C" void* csp_host_alloc(size_t nbytes) {
    void* p = nullptr;
    hipError_t e = hipHostMalloc(&p, nbytes, hipHostMallocDefault);
    if (e != hipSuccess) {
        set_err("hipHostMalloc", e);
        return nullptr;
    }
    return p;
}

extern "C" int csp_host_free(void* p) {
    HIP_TRY(hipHostFree(p));
    return 0;
}

// Pooled pinned staging allocator.  A task result may contain several
// large tensors, each needing its own live pinned block until the reply
// is written — so this is a freelist of hipHostMalloc blocks, recycled
// across electrons (pinning 1 GiB costs ~100s of ms; reuse makes it
// one-time).  csp_staging_release_all() returns every outstanding block
// to the freelist; blocks beyond the cache cap are actually freed.
static std::mutex g_staging_mu;
static std::vector<std::pair<void*, size_t>> g_staging_free;
static std::vector<std::pair<void*, size_t>> g_staging_in_use;
static const size_t kStagingCacheCap = size_t(6) << 30;  // 6 GiB pinned cache

extern "C" void* csp_staging_alloc(size_t nbytes) {
    std::lock_guard<std::mutex> lock(g_staging_mu);
    // best-fit from the freelist
    int best = -1;
    for (int i = 0; i < (int)g_staging_free.size(); ++i) {
        if (g_staging_free[i].second >= nbytes &&
            (best < 0 || g_staging_free[i].second < g_staging_free[best].second))
            best = i;
    }
    if (best >= 0) {
        auto blk = g_staging_free[best];
        g_staging_free.erase(g_staging_free.begin() + best);
        g_staging_in_use.push_back(blk);
        return blk.first;
    }
    void* p = nullptr;
    hipError_t e = hipHostMalloc(&p, nbytes, hipHostMallocDefault);
    if (e != hipSuccess) {
        set_err("hipHostMalloc(staging)", e);
        return nullptr;
    }
    g_staging_in_use.push_back({p, nbytes});
    return p;
}

// Back-compat alias (same semantics as alloc).
extern "C" void* csp_staging_get(size_t nbytes) { return csp_staging_alloc(nbytes); }

extern "C" int csp_staging_release_all() {
    std::lock_guard<std::mutex> lock(g_staging_mu);
    for (auto& blk : g_staging_in_use) g_staging_free.push_back(blk);
    g_staging_in_use.clear();
    // trim the cache: keep the largest blocks up to the cap
    std::sort(g_staging_free.begin(), g_staging_free.end(),
              [](auto& a, auto& b) { return a.second > b.second; });
    size_t kept = 0;
    std::vector<std::pair<void*, size_t>> keep;
    for (auto& blk : g_staging_free) {
        if (kept + blk.second <= kStagingCacheCap) {
            kept += blk.second;
            keep.push_back(blk);
        } else {
            HIP_TRY(hipHostFree(blk.first));
        }
    }
    g_staging_free.swap(keep);
    return 0;
}

extern "C" int csp_staging_reset() {
    std::lock_guard<std::mutex> lock(g_staging_mu);
    for (auto& blk : g_staging_free) HIP_TRY(hipHostFree(blk.first));
    for (auto& blk : g_staging_in_use) HIP_TRY(hipHostFree(blk.first));
    g_staging_free.clear();
    g_staging_in_use.clear();
    return 0;
}

// Dedicated copy stream so staging copies never serialize behind the
// caller's compute stream.
static hipStream_t copy_stream() {
    static hipStream_t s = nullptr;
    static std::once_flag once;
    std::call_once(once, [] {
        if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess)
            s = nullptr;  // fall back to the null stream
    });
    return s;
}

extern "C" int csp_memcpy_d2h(void* dst_host, const void* src_dev, size_t n) {
    hipStream_t s = copy_stream();
    HIP_TRY(hipMemcpyAsync(dst_host, src_dev, n, hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    return 0;
}

extern "C" int csp_memcpy_h2d(void* dst_dev, const void* src_host, size_t n) {
    hipStream_t s = copy_stream();
    HIP_TRY(hipMemcpyAsync(dst_dev, src_host, n, hipMemcpyHostToDevice, s));
    HIP_TRY(hipStreamSynchronize(s));
    return 0;
}

// On-box variant exploration for the HBM sweep (tools only; the probe
// uses g_sweep_variant).  Returns measured GB/s via *gbps.
extern "C" int csp_hbm_bench(int device, int variant, size_t bytes, int reps,
                             double* gbps) {
    HIP_TRY(hipSetDevice(device));
    size_t n4 = bytes / sizeof(float4);
    float4 *a = nullptr, *b = nullptr;
    HIP_TRY(hipMalloc(&a, bytes));
    HIP_TRY(hipMalloc(&b, bytes));
    HIP_TRY(hipMemsetAsync(a, 1, bytes));
    float warm_ms = 0.f, ms = 0.f;
    int rc = run_hbm_sweep(a, b, n4, 2, &warm_ms, variant);
    if (rc == 0) rc = run_hbm_sweep(a, b, n4, reps, &ms, variant);
    (void)hipFree(a);
    (void)hipFree(b);
    if (rc != 0) return rc;
    *gbps = (2.0 * (double)bytes * reps / 1.0e9) / ((double)ms / 1000.0);
    return 0;
}

// Extended exploration: grid size + copy direction knobs.
extern "C" int csp_hbm_bench2(int device, int variant, size_t bytes, int reps,
                              int blocks, int pingpong, double* gbps) {
    HIP_TRY(hipSetDevice(device));
    size_t n4 = bytes / sizeof(float4);
    float4 *a = nullptr, *b = nullptr;
    HIP_TRY(hipMalloc(&a, bytes));
    HIP_TRY(hipMalloc(&b, bytes));
    HIP_TRY(hipMemsetAsync(a, 1, bytes));
    float warm_ms = 0.f, ms = 0.f;
    int rc = run_hbm_sweep(a, b, n4, 2, &warm_ms, variant, blocks, pingpong != 0);
    if (rc == 0)
        rc = run_hbm_sweep(a, b, n4, reps, &ms, variant, blocks, pingpong != 0);
    (void)hipFree(a);
    (void)hipFree(b);
    if (rc != 0) return rc;
    *gbps = (2.0 * (double)bytes * reps / 1.0e9) / ((double)ms / 1000.0);
    return 0;
}

// MFMA spin exploration: blocks knob (TF/s out).
extern "C" int csp_mfma_bench(int device, int iters, int blocks, double* tflops) {
    HIP_TRY(hipSetDevice(device));
    float ms = 0.f;
    int rc = run_mfma_spin(iters / 4, blocks, &ms);  // warm clocks
    if (rc == 0) rc = run_mfma_spin(iters, blocks, &ms);
    if (rc != 0) return rc;
    double waves = (double)blocks * 4.0;
    *tflops = waves * (double)iters * kFlopPerMfma / ((double)ms / 1000.0) / 1.0e12;
    return 0;
}

// ---------------------------------------------------------------------------
// MFMA numerics check: one v_mfma_f32_32x32x16_bf16 with self-described
// operand layouts.  Each lane scatters the exact operand values it feeds
// the instruction into global A (32x16) and B (16x32), and its
// accumulator fragment into D (32x32) using the C/D mapping
// (col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)).  The host
// test then asserts D == A @ B against a PyTorch reference — validating
// the assumed A/B lane mapping AND the instruction together.  `layout`
// selects the A/B k-mapping candidate:
//   0: k = (lane>>5)*8 + j           (contiguous 8)
//   1: k = (lane>>5)*4 + (j&3) + (j>>2)*8   (two 4-element halves)
__global__ __launch_bounds__(64) void csp_mfma_check_kernel(
    float* __restrict__ D, float* __restrict__ A, float* __restrict__ B,
    int layout) {
    const int l = threadIdx.x;
    bf16x8 a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        const int k = (layout == 0) ? ((l >> 5) * 8 + j)
                                    : ((l >> 5) * 4 + (j & 3) + (j >> 2) * 8);
        const int i = l & 31;  // A row
        const float av = (float)(((i * 3 + k) % 7) - 3);  // integer-valued
        a[j] = (__bf16)av;
        A[i * 16 + k] = av;
        const int col = l & 31;  // B column
        const float bv = (float)(((k * 5 + col) % 5) - 2);
        b[j] = (__bf16)bv;
        B[k * 32 + col] = bv;
    }
    f32x16 acc = {};
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
        D[row * 32 + (l & 31)] = acc[r];
    }
}

extern "C" int csp_mfma_check(int device, int layout, float* d_out,
                              float* a_out, float* b_out) {
    HIP_TRY(hipSetDevice(device));
    float *D = nullptr, *A = nullptr, *B = nullptr;
    HIP_TRY(hipMalloc(&D, 32 * 32 * sizeof(float)));
    HIP_TRY(hipMalloc(&A, 32 * 16 * sizeof(float)));
    HIP_TRY(hipMalloc(&B, 16 * 32 * sizeof(float)));
    hipLaunchKernelGGL(csp_mfma_check_kernel, dim3(1), dim3(64), 0, 0, D, A, B,
                       layout);
    HIP_TRY(hipMemcpy(d_out, D, 32 * 32 * sizeof(float), hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(a_out, A, 32 * 16 * sizeof(float), hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(b_out, B, 16 * 32 * sizeof(float), hipMemcpyDeviceToHost));
    (void)hipFree(D);
    (void)hipFree(A);
    (void)hipFree(B);
    return 0;
}
