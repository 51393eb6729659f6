// Stencil probe: 5-point fp32 Laplacian variants on gfx950.
// Finds the right codegen strategy for the shifted-slice stencil family
// (BASELINE configs[3]).  Traffic: 8 B/elem algorithmic (1 read + 1 write;
// row reuse must come from caches or LDS).
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_stencil.hip
//        -o tools/_build/probe_stencil
// Run:   probe_stencil [n] [grid]

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define CHECK(x)                                                         \
    do {                                                                 \
        hipError_t e = (x);                                              \
        if (e != hipSuccess) {                                           \
            fprintf(stderr, "%s: %s\n", #x, hipGetErrorString(e));       \
            exit(1);                                                     \
        }                                                                \
    } while (0)

typedef long long i64;
typedef __attribute__((ext_vector_type(4))) float f4_t;

// interior (1..n-2)^2; row stride = pitch (elements), 128B-aligned rows.

// v0: one element per thread, scalar loads (what the generic codegen emits)
__global__ __launch_bounds__(256) void st_naive(
    const float *__restrict__ A, float *__restrict__ B, int n, i64 pitch) {
    i64 x = (i64)blockIdx.x * 256 + threadIdx.x + 1;
    i64 y = blockIdx.y + 1;
    if (x >= n - 1) return;
    for (; y < n - 1; y += gridDim.y) {
        const float *r0 = A + (y - 1) * pitch;
        const float *r1 = A + y * pitch;
        const float *r2 = A + (y + 1) * pitch;
        B[y * pitch + x] = r0[x] + r2[x] + r1[x - 1] + r1[x + 1]
                           - 4.0f * r1[x];
    }
}

// v1: 4 elements per thread via scalar loads (codegen's V=4 'u' class)
__global__ __launch_bounds__(256) void st_v4scalar(
    const float *__restrict__ A, float *__restrict__ B, int n, i64 pitch) {
    i64 x = ((i64)blockIdx.x * 256 + threadIdx.x) * 4 + 1;
    i64 y = blockIdx.y + 1;
    for (; y < n - 1; y += gridDim.y) {
        const float *r0 = A + (y - 1) * pitch;
        const float *r1 = A + y * pitch;
        const float *r2 = A + (y + 1) * pitch;
        for (int l = 0; l < 4; ++l) {
            i64 xx = x + l;
            if (xx < n - 1)
                B[y * pitch + xx] = r0[xx] + r2[xx] + r1[xx - 1] + r1[xx + 1]
                                    - 4.0f * r1[xx];
        }
    }
}

// v2: vector loads — each thread owns an aligned float4; the x-1 / x+1
// neighbours come from the own vector + one scalar load at each end.
__global__ __launch_bounds__(256) void st_vec4(
    const float *__restrict__ A, float *__restrict__ B, int n, i64 pitch) {
    i64 x4 = ((i64)blockIdx.x * 256 + threadIdx.x) * 4;  // aligned start
    if (x4 + 4 > n) return;
    i64 y = blockIdx.y + 1;
    for (; y < n - 1; y += gridDim.y) {
        const float *r0 = A + (y - 1) * pitch;
        const float *r1 = A + y * pitch;
        const float *r2 = A + (y + 1) * pitch;
        f4_t a0 = *(const f4_t *)&r0[x4];
        f4_t a1 = *(const f4_t *)&r1[x4];
        f4_t a2 = *(const f4_t *)&r2[x4];
        float left = x4 > 0 ? r1[x4 - 1] : 0.0f;
        float right = x4 + 4 < n ? r1[x4 + 4] : 0.0f;
        f4_t out;
        out[0] = a0[0] + a2[0] + left + a1[1] - 4.0f * a1[0];
        out[1] = a0[1] + a2[1] + a1[0] + a1[2] - 4.0f * a1[1];
        out[2] = a0[2] + a2[2] + a1[1] + a1[3] - 4.0f * a1[2];
        out[3] = a0[3] + a2[3] + a1[2] + right - 4.0f * a1[3];
        // interior mask: x in [1, n-2]
        if (x4 >= 1 && x4 + 4 <= n - 1) {
            *(f4_t *)&B[y * pitch + x4] = out;
        } else {
            for (int l = 0; l < 4; ++l) {
                i64 xx = x4 + l;
                if (xx >= 1 && xx < n - 1) B[y * pitch + xx] = out[l];
            }
        }
    }
}

// v3: y-blocked vector kernel — each thread processes YB consecutive rows,
// carrying row registers down (each row loaded once per thread).
template <int YB>
__global__ __launch_bounds__(256) void st_ycarry(
    const float *__restrict__ A, float *__restrict__ B, int n, i64 pitch) {
    i64 x4 = ((i64)blockIdx.x * 256 + threadIdx.x) * 4;
    if (x4 + 4 > n) return;
    i64 y0 = (i64)blockIdx.y * YB + 1;
    if (y0 >= n - 1) return;
    f4_t rm = *(const f4_t *)&A[(y0 - 1) * pitch + x4];
    f4_t rc = *(const f4_t *)&A[y0 * pitch + x4];
    bool edge = !(x4 >= 1 && x4 + 4 <= n - 1);
    for (int k = 0; k < YB; ++k) {
        i64 y = y0 + k;
        if (y >= n - 1) break;
        f4_t rp = *(const f4_t *)&A[(y + 1) * pitch + x4];
        float left = x4 > 0 ? A[y * pitch + x4 - 1] : 0.0f;
        float right = x4 + 4 < n ? A[y * pitch + x4 + 4] : 0.0f;
        f4_t out;
        out[0] = rm[0] + rp[0] + left + rc[1] - 4.0f * rc[0];
        out[1] = rm[1] + rp[1] + rc[0] + rc[2] - 4.0f * rc[1];
        out[2] = rm[2] + rp[2] + rc[1] + rc[3] - 4.0f * rc[2];
        out[3] = rm[3] + rp[3] + rc[2] + right - 4.0f * rc[3];
        if (!edge) {
            *(f4_t *)&B[y * pitch + x4] = out;
        } else {
            for (int l = 0; l < 4; ++l) {
                i64 xx = x4 + l;
                if (xx >= 1 && xx < n - 1) B[y * pitch + xx] = out[l];
            }
        }
        rm = rc;
        rc = rp;
    }
}

// v4: unrolled y-block, addresses from ONE base pointer with literal row
// deltas -- tests whether LLVM GVN dedupes the overlapping vector loads
// across unrolled rows (the codegen-friendly alternative to an explicit
// register window)
template <int YB>
__global__ __launch_bounds__(256) void st_yunroll(
    const float *__restrict__ A, float *__restrict__ B, int n, i64 pitch) {
    i64 x4 = ((i64)blockIdx.x * 256 + threadIdx.x) * 4;
    if (x4 + 4 > n) return;
    i64 y0 = (i64)blockIdx.y * YB + 1;
    if (y0 + YB > n - 1) {
        for (i64 y = y0; y < n - 1; ++y) {
            f4_t a0 = *(const f4_t *)&A[(y - 1) * pitch + x4];
            f4_t a1 = *(const f4_t *)&A[y * pitch + x4];
            f4_t a2 = *(const f4_t *)&A[(y + 1) * pitch + x4];
            float left = x4 > 0 ? A[y * pitch + x4 - 1] : 0.0f;
            float right = x4 + 4 < n ? A[y * pitch + x4 + 4] : 0.0f;
            f4_t out;
            out[0] = a0[0] + a2[0] + left + a1[1] - 4.0f * a1[0];
            out[1] = a0[1] + a2[1] + a1[0] + a1[2] - 4.0f * a1[1];
            out[2] = a0[2] + a2[2] + a1[1] + a1[3] - 4.0f * a1[2];
            out[3] = a0[3] + a2[3] + a1[2] + right - 4.0f * a1[3];
            if (x4 >= 1 && x4 + 4 <= n - 1) *(f4_t *)&B[y * pitch + x4] = out;
        }
        return;
    }
#pragma unroll
    for (int ky = 0; ky < YB; ++ky) {
        i64 y = y0 + ky;
        f4_t a0 = *(const f4_t *)&A[(y - 1) * pitch + x4];
        f4_t a1 = *(const f4_t *)&A[y * pitch + x4];
        f4_t a2 = *(const f4_t *)&A[(y + 1) * pitch + x4];
        float left = x4 > 0 ? A[y * pitch + x4 - 1] : 0.0f;
        float right = x4 + 4 < n ? A[y * pitch + x4 + 4] : 0.0f;
        f4_t out;
        out[0] = a0[0] + a2[0] + left + a1[1] - 4.0f * a1[0];
        out[1] = a0[1] + a2[1] + a1[0] + a1[2] - 4.0f * a1[1];
        out[2] = a0[2] + a2[2] + a1[1] + a1[3] - 4.0f * a1[2];
        out[3] = a0[3] + a2[3] + a1[2] + right - 4.0f * a1[3];
        if (x4 >= 1 && x4 + 4 <= n - 1) *(f4_t *)&B[y * pitch + x4] = out;
    }
}

template <typename F>
static double timeit(F f, int iters) {
    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    f();
    CHECK(hipDeviceSynchronize());
    CHECK(hipEventRecord(e0, 0));
    for (int i = 0; i < iters; ++i) f();
    CHECK(hipEventRecord(e1, 0));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    return ms / iters;
}

__global__ void init_k(float *A, i64 total) {
    i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x;
    i64 s = (i64)gridDim.x * blockDim.x;
    for (; i < total; i += s) A[i] = (float)(i % 977) * 0.25f;
}

int main(int argc, char **argv) {
    int n = argc > 1 ? atoi(argv[1]) : 16384;
    int gy = argc > 2 ? atoi(argv[2]) : 4096;
    i64 pitch = ((i64)n + 31) / 32 * 32;
    i64 total = pitch * n;
    float *A, *B;
    CHECK(hipMalloc(&A, total * 4));
    CHECK(hipMalloc(&B, total * 4));
    hipLaunchKernelGGL(init_k, dim3(4096), dim3(256), 0, 0, A, total);
    CHECK(hipDeviceSynchronize());
    double bytes = (double)(n - 2) * (n - 2) * 8.0;
    double ms;
    int gx1 = (n + 255) / 256;
    int gx4 = (n + 1023) / 1024;

#define RUN(name, launch)                                                 \
    ms = timeit([&] { launch; }, 5);                                      \
    printf("%-12s %8.3f ms   %8.1f GB/s\n", name, ms, bytes / ms * 1e-6);

    RUN("naive", hipLaunchKernelGGL(st_naive, dim3(gx1, (unsigned)gy), dim3(256),
                                    0, 0, A, B, n, pitch));
    RUN("v4scalar", hipLaunchKernelGGL(st_v4scalar, dim3(gx4, (unsigned)gy), dim3(256),
                                       0, 0, A, B, n, pitch));
    RUN("vec4", hipLaunchKernelGGL(st_vec4, dim3(gx4, (unsigned)gy), dim3(256),
                                   0, 0, A, B, n, pitch));
    int yb = 8;
    RUN("ycarry8", hipLaunchKernelGGL(st_ycarry<8>, dim3(gx4, (n + yb - 1) / yb),
                                      dim3(256), 0, 0, A, B, n, pitch));
    RUN("ycarry16", hipLaunchKernelGGL(st_ycarry<16>, dim3(gx4, (n + 15) / 16),
                                       dim3(256), 0, 0, A, B, n, pitch));
    RUN("yunroll4", hipLaunchKernelGGL(st_yunroll<4>, dim3(gx4, (n + 3) / 4),
                                       dim3(256), 0, 0, A, B, n, pitch));
    RUN("yunroll8", hipLaunchKernelGGL(st_yunroll<8>, dim3(gx4, (n + 7) / 8),
                                       dim3(256), 0, 0, A, B, n, pitch));
    return 0;
}
