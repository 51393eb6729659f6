// Bandwidth probe: what does gfx950 sustain for the flagship access mix?
// Calibrates the bench roofline (1 read + 3 write fp64 streams) against
// pure copy and store-only patterns, and isolates the sincos VALU cost.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_streams.hip
//        -o tools/_build/probe_streams
// Run:   probe_streams [elems]     (default 5e8; needs 4*8*elems bytes)

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
typedef __attribute__((ext_vector_type(2))) double d2_t;

__device__ __forceinline__ void rt_sincos(double x, double *sr, double *cr) {
    const double invpio2 = 6.36619772367581382433e-01;
    const double pio2_1 = 1.57079632673412561417e+00;
    const double pio2_1t = 6.07710050650619224932e-11;
    double fn = __builtin_rint(x * invpio2);
    int n = (int)fn;
    double r = __builtin_fma(-fn, pio2_1, x);
    double w = fn * pio2_1t;
    double y0 = r - w, y1 = (r - y0) - w;
    const double S1 = -1.66666666666666324348e-01, S2 = 8.33333333332248946124e-03,
                 S3 = -1.98412698298579493134e-04, S4 = 2.75573137070700676789e-06,
                 S5 = -2.50507602534068634195e-08, S6 = 1.58969099521155010221e-10;
    double z = y0 * y0, v = z * y0;
    double rs = S2 + z * (S3 + z * (S4 + z * (S5 + z * S6)));
    double ks = y0 - ((z * (0.5 * y1 - v * rs) - y1) - v * S1);
    const double C1 = 4.16666666666666019037e-02, C2 = -1.38888888888741095749e-03,
                 C3 = 2.48015872894767294178e-05, C4 = -2.75573143513906633035e-07,
                 C5 = 2.08757232129817482790e-09, C6 = -1.13596475577881948265e-11;
    double rc = z * (C1 + z * (C2 + z * (C3 + z * (C4 + z * (C5 + z * C6)))));
    double hz = 0.5 * z, wc = 1.0 - hz;
    double kc = wc + (((1.0 - wc) - hz) + (z * rc - y0 * y1));
    switch (n & 3) {
        case 0: *sr = ks; *cr = kc; break;
        case 1: *sr = kc; *cr = -ks; break;
        case 2: *sr = -ks; *cr = -kc; break;
        default: *sr = -kc; *cr = ks; break;
    }
}

// grid-stride, V=2, like the generated kernels
#define LOOP(...)                                                        \
    i64 vb = ((i64)blockIdx.x * 256 + threadIdx.x) * 2;                   \
    const i64 xs = (i64)gridDim.x * 256 * 2;                              \
    for (; vb + 2 <= n; vb += xs) { __VA_ARGS__ }

__global__ __launch_bounds__(256) void k_init(double *A, i64 n) {
    i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x;
    i64 s = (i64)gridDim.x * blockDim.x;
    for (; i < n; i += s) A[i] = (double)i * 0.001;
}

__global__ __launch_bounds__(256) void k_copy(const double *__restrict__ A, double *__restrict__ B,
                                              i64 n) {
    LOOP(*(d2_t *)&B[vb] = *(const d2_t *)&A[vb];)
}

__global__ __launch_bounds__(256) void k_store3(double *__restrict__ B, double *__restrict__ C,
                                                double *__restrict__ D, i64 n) {
    d2_t one = {1.0, 2.0};
    LOOP(*(d2_t *)&B[vb] = one; *(d2_t *)&C[vb] = one; *(d2_t *)&D[vb] = one;)
}

__global__ __launch_bounds__(256) void k_1r3w(const double *__restrict__ A, double *__restrict__ B,
                                              double *__restrict__ C, double *__restrict__ D, i64 n) {
    LOOP(d2_t a = *(const d2_t *)&A[vb]; d2_t b, c, d;
         b[0] = a[0] + 1.0; b[1] = a[1] + 1.0;
         c[0] = a[0] * 2.0; c[1] = a[1] * 2.0;
         d[0] = a[0] + 3.0; d[1] = a[1] + 3.0;
         *(d2_t *)&B[vb] = b; *(d2_t *)&C[vb] = c; *(d2_t *)&D[vb] = d;)
}

__global__ __launch_bounds__(256) void k_1r3w_nt(const double *__restrict__ A, double *__restrict__ B,
                                                 double *__restrict__ C, double *__restrict__ D,
                                                 i64 n) {
    LOOP(d2_t a = *(const d2_t *)&A[vb];
         __builtin_nontemporal_store(a[0] + 1.0, &B[vb]);
         __builtin_nontemporal_store(a[1] + 1.0, &B[vb + 1]);
         __builtin_nontemporal_store(a[0] * 2.0, &C[vb]);
         __builtin_nontemporal_store(a[1] * 2.0, &C[vb + 1]);
         __builtin_nontemporal_store(a[0] + 3.0, &D[vb]);
         __builtin_nontemporal_store(a[1] + 3.0, &D[vb + 1]);)
}

__global__ __launch_bounds__(256) void k_flagship(const double *__restrict__ A, double *__restrict__ B,
                                                  double *__restrict__ C, double *__restrict__ D,
                                                  i64 n) {
    LOOP(d2_t a = *(const d2_t *)&A[vb]; d2_t b, c, d;
         double s0, c0, s1, c1;
         rt_sincos(a[0], &s0, &c0); rt_sincos(a[1], &s1, &c1);
         b[0] = s0; b[1] = s1; c[0] = c0; c[1] = c1;
         d[0] = s0 * s0 + c0 * c0; d[1] = s1 * s1 + c1 * c1;
         *(d2_t *)&B[vb] = b; *(d2_t *)&C[vb] = c; *(d2_t *)&D[vb] = d;)
}

__global__ __launch_bounds__(256) void k_flagship_nt(const double *__restrict__ A,
                                                     double *__restrict__ B, double *__restrict__ C,
                                                     double *__restrict__ D, i64 n) {
    LOOP(d2_t a = *(const d2_t *)&A[vb];
         double s0, c0, s1, c1;
         rt_sincos(a[0], &s0, &c0); rt_sincos(a[1], &s1, &c1);
         d2_t b, c, d;
         b[0] = s0; b[1] = s1; c[0] = c0; c[1] = c1;
         d[0] = s0 * s0 + c0 * c0; d[1] = s1 * s1 + c1 * c1;
         __builtin_nontemporal_store(b, (d2_t*)&B[vb]);
         __builtin_nontemporal_store(c, (d2_t*)&C[vb]);
         __builtin_nontemporal_store(d, (d2_t*)&D[vb]);)
}

// V=4 variant (32 B per lane per stream)
__global__ __launch_bounds__(256) void k_flagship_v4(const double *__restrict__ A,
                                                     double *__restrict__ B, double *__restrict__ C,
                                                     double *__restrict__ D, i64 n) {
    typedef __attribute__((ext_vector_type(4))) double d4_t;
    i64 vb = ((i64)blockIdx.x * 256 + threadIdx.x) * 4;
    const i64 xs = (i64)gridDim.x * 256 * 4;
    for (; vb + 4 <= n; vb += xs) {
        d4_t a = *(const d4_t *)&A[vb];
        d4_t b, c, d;
        for (int l = 0; l < 4; ++l) {
            double s, cc;
            rt_sincos(a[l], &s, &cc);
            b[l] = s; c[l] = cc; d[l] = s * s + cc * cc;
        }
        *(d4_t *)&B[vb] = b;
        *(d4_t *)&C[vb] = c;
        *(d4_t *)&D[vb] = d;
    }
}


// unroll-4: four independent 16B chunks in flight per thread
__global__ __launch_bounds__(256) void k_copy_u4(const double *__restrict__ A,
                                                 double *__restrict__ B, i64 n) {
    i64 i = ((i64)blockIdx.x * 256 + threadIdx.x) * 2;
    const i64 G = (i64)gridDim.x * 256 * 2;
    for (; i + 3 * G + 2 <= n; i += 4 * G) {
        d2_t a0 = *(const d2_t *)&A[i];
        d2_t a1 = *(const d2_t *)&A[i + G];
        d2_t a2 = *(const d2_t *)&A[i + 2 * G];
        d2_t a3 = *(const d2_t *)&A[i + 3 * G];
        *(d2_t *)&B[i] = a0;
        *(d2_t *)&B[i + G] = a1;
        *(d2_t *)&B[i + 2 * G] = a2;
        *(d2_t *)&B[i + 3 * G] = a3;
    }
    for (; i + 2 <= n; i += G) *(d2_t *)&B[i] = *(const d2_t *)&A[i];
}

__global__ __launch_bounds__(256) void k_1r3w_u2(const double *__restrict__ A,
                                                 double *__restrict__ B,
                                                 double *__restrict__ C,
                                                 double *__restrict__ D, i64 n) {
    i64 i = ((i64)blockIdx.x * 256 + threadIdx.x) * 2;
    const i64 G = (i64)gridDim.x * 256 * 2;
    for (; i + G + 2 <= n; i += 2 * G) {
        d2_t a0 = *(const d2_t *)&A[i];
        d2_t a1 = *(const d2_t *)&A[i + G];
        d2_t b0, c0, d0, b1, c1, d1;
        b0[0] = a0[0] + 1.0; b0[1] = a0[1] + 1.0;
        c0[0] = a0[0] * 2.0; c0[1] = a0[1] * 2.0;
        d0[0] = a0[0] + 3.0; d0[1] = a0[1] + 3.0;
        b1[0] = a1[0] + 1.0; b1[1] = a1[1] + 1.0;
        c1[0] = a1[0] * 2.0; c1[1] = a1[1] * 2.0;
        d1[0] = a1[0] + 3.0; d1[1] = a1[1] + 3.0;
        *(d2_t *)&B[i] = b0; *(d2_t *)&C[i] = c0; *(d2_t *)&D[i] = d0;
        *(d2_t *)&B[i + G] = b1; *(d2_t *)&C[i + G] = c1;
        *(d2_t *)&D[i + G] = d1;
    }
    for (; i + 2 <= n; i += G) {
        d2_t a = *(const d2_t *)&A[i];
        d2_t b, c, d;
        b[0] = a[0] + 1.0; b[1] = a[1] + 1.0;
        c[0] = a[0] * 2.0; c[1] = a[1] * 2.0;
        d[0] = a[0] + 3.0; d[1] = a[1] + 3.0;
        *(d2_t *)&B[i] = b; *(d2_t *)&C[i] = c; *(d2_t *)&D[i] = d;
    }
}

__global__ __launch_bounds__(1024) void k_copy_b1024(const double *__restrict__ A,
                                                     double *__restrict__ B, i64 n) {
    i64 vb = ((i64)blockIdx.x * 1024 + threadIdx.x) * 2;
    const i64 xs = (i64)gridDim.x * 1024 * 2;
    for (; vb + 2 <= n; vb += xs) *(d2_t *)&B[vb] = *(const d2_t *)&A[vb];
}

__global__ __launch_bounds__(256) void k_copy_ntl(const double *__restrict__ A,
                                                  double *__restrict__ B, i64 n) {
    LOOP(d2_t a = __builtin_nontemporal_load((const d2_t *)&A[vb]);
         __builtin_nontemporal_store(a, (d2_t *)&B[vb]);)
}

typedef __attribute__((ext_vector_type(4))) float f4v;
__global__ __launch_bounds__(256) void k_1r3w_sc1(const double *__restrict__ A,
                                                  double *__restrict__ B,
                                                  double *__restrict__ C,
                                                  double *__restrict__ D, i64 n) {
    auto rb = __builtin_amdgcn_make_buffer_rsrc((void*)B, 0, (unsigned)(n * 8), 0x00020000);
    auto rc = __builtin_amdgcn_make_buffer_rsrc((void*)C, 0, (unsigned)(n * 8), 0x00020000);
    auto rd = __builtin_amdgcn_make_buffer_rsrc((void*)D, 0, (unsigned)(n * 8), 0x00020000);
    LOOP(d2_t a = *(const d2_t *)&A[vb];
         d2_t b, c, d;
         b[0] = a[0] + 1.0; b[1] = a[1] + 1.0;
         c[0] = a[0] * 2.0; c[1] = a[1] * 2.0;
         d[0] = a[0] + 3.0; d[1] = a[1] + 3.0;
         unsigned off = (unsigned)(vb * 8);
         __builtin_amdgcn_raw_buffer_store_b128(__builtin_bit_cast(f4v, b), rb, off, 0, 16);
         __builtin_amdgcn_raw_buffer_store_b128(__builtin_bit_cast(f4v, c), rc, off, 0, 16);
         __builtin_amdgcn_raw_buffer_store_b128(__builtin_bit_cast(f4v, d), rd, off, 0, 16);)
}

__global__ __launch_bounds__(256) void k_1r3w_ntl(const double *__restrict__ A,
                                                  double *__restrict__ B,
                                                  double *__restrict__ C,
                                                  double *__restrict__ D, i64 n) {
    LOOP(d2_t a = __builtin_nontemporal_load((const d2_t *)&A[vb]);
         d2_t b, c, d;
         b[0] = a[0] + 1.0; b[1] = a[1] + 1.0;
         c[0] = a[0] * 2.0; c[1] = a[1] * 2.0;
         d[0] = a[0] + 3.0; d[1] = a[1] + 3.0;
         *(d2_t *)&B[vb] = b; *(d2_t *)&C[vb] = c; *(d2_t *)&D[vb] = d;)
}

template <typename F>
static double timeit(F f, int iters) {
    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    f();  // warm
    CHECK(hipDeviceSynchronize());
    CHECK(hipEventRecord(e0, 0));
    for (int i = 0; i < iters; ++i) f();
    CHECK(hipEventRecord(e1, 0));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    return ms / iters;
}

int main(int argc, char **argv) {
    i64 n = argc > 1 ? atoll(argv[1]) : 500000000LL;
    int grid = argc > 2 ? atoi(argv[2]) : 8192;
    double *A, *B, *C, *D;
    CHECK(hipMalloc(&A, n * 8));
    CHECK(hipMalloc(&B, n * 8));
    CHECK(hipMalloc(&C, n * 8));
    CHECK(hipMalloc(&D, n * 8));
    // init A = i*0.001
    {
        hipLaunchKernelGGL(k_init, dim3(4096), dim3(256), 0, 0, A, n);
        CHECK(hipDeviceSynchronize());
    }
    struct Row { const char *name; double ms; double gb; };
    double ms;

#define RUN(name, bytes, launch)                                          \
    ms = timeit([&] { launch; }, 5);                                      \
    printf("%-14s %8.3f ms   %8.1f GB/s (algorithmic %.1f GB)\n", name,  \
           ms, (bytes) / ms * 1e-6, (bytes) / 1e9);

    RUN("memcpyDtoD", n * 16.0,
        CHECK(hipMemcpyAsync(B, A, n * 8, hipMemcpyDeviceToDevice, 0)));
    RUN("copy_1r1w", n * 16.0,
        hipLaunchKernelGGL(k_copy, dim3(grid), dim3(256), 0, 0, A, B, n));
    RUN("store3", n * 24.0,
        hipLaunchKernelGGL(k_store3, dim3(grid), dim3(256), 0, 0, B, C, D, n));
    RUN("1r3w", n * 32.0,
        hipLaunchKernelGGL(k_1r3w, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("1r3w_nt", n * 32.0,
        hipLaunchKernelGGL(k_1r3w_nt, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("flagship", n * 32.0,
        hipLaunchKernelGGL(k_flagship, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("flagship_nt", n * 32.0,
        hipLaunchKernelGGL(k_flagship_nt, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("flagship_v4", n * 32.0,
        hipLaunchKernelGGL(k_flagship_v4, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("copy_u4", n * 16.0,
        hipLaunchKernelGGL(k_copy_u4, dim3(grid), dim3(256), 0, 0, A, B, n));
    RUN("1r3w_u2", n * 32.0,
        hipLaunchKernelGGL(k_1r3w_u2, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("copy_b1024", n * 16.0,
        hipLaunchKernelGGL(k_copy_b1024, dim3(grid / 4), dim3(1024), 0, 0, A, B, n));
    RUN("copy_ntl", n * 16.0,
        hipLaunchKernelGGL(k_copy_ntl, dim3(grid), dim3(256), 0, 0, A, B, n));
    RUN("1r3w_sc1", n * 32.0,
        hipLaunchKernelGGL(k_1r3w_sc1, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    RUN("1r3w_ntl", n * 32.0,
        hipLaunchKernelGGL(k_1r3w_ntl, dim3(grid), dim3(256), 0, 0, A, B, C, D, n));
    return 0;
}
