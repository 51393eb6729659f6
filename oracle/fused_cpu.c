/* oracle/fused_cpu.c — TEST INFRASTRUCTURE (see oracle/__init__.py).
 *
 * C/OpenMP restatement of the north-star fused loop, used ONLY as
 * bench.py's `cpu_baseline` leg (kind="port") per BASELINE.md: the
 * reference's own Numba+MPI path cannot run in this environment (numba and
 * mpi4py are absent, no network), so this is the "best-case CPU" stand-in
 * for the reference's fused Numba loop — the same loop body the reference
 * generates and JITs for sample/test-ramba.py (one read of A, three
 * stores; codelines per ramba/ramba.py:8246-8265 / op tables 7950-7956):
 *     b = sin(a); c = cos(a); d = b*b + c*c
 *
 * Build: gcc -O3 -fopenmp oracle/fused_cpu.c -o oracle/_build/fused_cpu -lm
 * Run:   fused_cpu <nelems> <iters>      (prints one JSON line)
 */

#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>
#ifdef _OPENMP
#include <omp.h>
#endif

static double now(void) {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
}

int main(int argc, char **argv) {
    long long n = argc > 1 ? atoll(argv[1]) : 100000000LL;
    int iters = argc > 2 ? atoi(argv[2]) : 3;
    double *A = malloc(n * sizeof(double));
    double *B = malloc(n * sizeof(double));
    double *C = malloc(n * sizeof(double));
    double *D = malloc(n * sizeof(double));
    if (!A || !B || !C || !D) {
        fprintf(stderr, "alloc failed\n");
        return 1;
    }
#pragma omp parallel for schedule(static)
    for (long long i = 0; i < n; ++i) A[i] = (double)i * 0.001;

    /* one untimed warmup iteration (pages + frequency), matching the
     * reference's excluded first/JIT iteration (README.md:63) */
    for (int it = -1; it < iters; ++it) {
        double t0 = now();
#pragma omp parallel for schedule(static)
        for (long long i = 0; i < n; ++i) {
            double a = A[i];
            double s = sin(a), c = cos(a);
            B[i] = s;
            C[i] = c;
            D[i] = s * s + c * c;
        }
        double t1 = now();
        if (it == iters - 1) {
            int threads = 1;
#ifdef _OPENMP
            threads = omp_get_max_threads();
#endif
            /* checksum keeps the loop un-eliminated */
            double chk = B[n / 2] + C[n / 3] + D[n - 1];
            printf("{\"elems\": %lld, \"iters\": %d, \"last_iter_secs\": "
                   "%.6f, \"threads\": %d, \"check\": %.6f}\n",
                   n, iters, t1 - t0, threads, chk);
        }
    }
    free(A); free(B); free(C); free(D);
    return 0;
}
