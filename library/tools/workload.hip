/* workload.hip — GPU workload generator for MI355X (gfx950).
 *
 * The vGPU stack's test/bench workload: a tunable busy kernel (FMA spin
 * calibrated by cycle count, ~2.4 GHz on MI355X) plus thin C wrappers
 * over hipMalloc/hipFree/hipMemGetInfo, loaded via ctypes by bench.py
 * and the GPU tests.  When the process runs under LD_PRELOAD of
 * libvgpu-control.so every call here goes through the shim — this IS
 * the interception-overhead measurement vehicle.
 *
 * (Parity: reference library/tools/gpu_busy.cu + mem tools, rebuilt
 * as one HIP library; CDNA4: wave64, s_memtime-free portable spin.)
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define API extern "C" __attribute__((visibility("default")))

__global__ void busy_kernel(int64_t spin, float *sink) {
    /* dependent-FMA spin: ~2 cycles per iteration per lane (VALU
     * f32 FMA throughput on CDNA4), immune to DCE via sink store.     */
    float a = 1.0009765625f;                /* 1 + 2^-10               */
    float x = (float)(threadIdx.x + 1);
    for (int64_t i = 0; i < spin; i++) x = __builtin_fmaf(x, a, 0.25f);
    if (x == 0.0f) sink[threadIdx.x] = x;   /* never true              */
}

__global__ void touch_kernel(float *buf, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) buf[i] = buf[i] * 2.0f + 1.0f;
}

static float *g_sink = nullptr;

API int wk_init(int dev) {
    if (hipSetDevice(dev) != hipSuccess) return -1;
    if (!g_sink && hipMalloc((void **)&g_sink, 4096) != hipSuccess)
        return -2;
    return 0;
}

/* launch `count` busy kernels of `grid` blocks spinning `spin_iters`   */
API int wk_launch_busy(int count, int grid, int block, int64_t spin_iters) {
    for (int i = 0; i < count; i++) {
        hipLaunchKernelGGL(busy_kernel, dim3((uint32_t)grid), dim3((uint32_t)block), 0, 0,
                           spin_iters, g_sink);
        hipError_t e = hipGetLastError();
        if (e != hipSuccess) return (int)e;
    }
    return 0;
}

API int wk_touch(void *buf, size_t floats) {
    int grid = (int)((floats + 255) / 256);
    hipLaunchKernelGGL(touch_kernel, dim3(grid), dim3(256), 0, 0,
                       (float *)buf, floats);
    return (int)hipGetLastError();
}

API int wk_sync(void) { return (int)hipDeviceSynchronize(); }

API void *wk_malloc(size_t bytes) {
    void *p = nullptr;
    if (hipMalloc(&p, bytes) != hipSuccess) return nullptr;
    return p;
}

API int wk_free(void *p) { return (int)hipFree(p); }

API long long wk_mem_total(void) {
    size_t f = 0, t = 0;
    if (hipMemGetInfo(&f, &t) != hipSuccess) return -1;
    return (long long)t;
}

API long long wk_mem_free(void) {
    size_t f = 0, t = 0;
    if (hipMemGetInfo(&f, &t) != hipSuccess) return -1;
    return (long long)f;
}

API int wk_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return -1;
    return n;
}

/* managed-memory prefetch probe (reference mem_prefetch.cu parity)    */
API void *wk_malloc_managed(size_t bytes) {
    void *p = nullptr;
    if (hipMallocManaged(&p, bytes, hipMemAttachGlobal) != hipSuccess)
        return nullptr;
    return p;
}

API int wk_prefetch(void *p, size_t bytes, int dev) {
    if (hipMemPrefetchAsync(p, bytes, dev, 0) != hipSuccess) return -1;
    return (int)hipDeviceSynchronize();
}
