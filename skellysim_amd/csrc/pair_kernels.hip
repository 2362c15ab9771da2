/* pair_kernels.hip — hand-written CDNA4 (gfx950/MI355X) fp64 pair kernels for
 * SkellySim's hydrodynamic hot path.
 *
 * Built from scratch for MI355X; the math restates the reference
 * (flatironinstitute/SkellySim) formula by formula:
 *   Stokeslet            src/core/kernels.cu:62-76
 *   Stresslet (DL)       src/core/kernels.cu:29-54
 *   Regularized Oseen    src/core/kernels.cpp:85-131
 *   Rotlet               src/core/kernels.cpp:206-242
 *
 * Design (nothing shared with the reference's CUDA driver, kernels.cu:80-123,
 * which used 32-thread blocks and accumulated into global memory every tile):
 *   - 256-thread blocks (4 wave64s), TPT targets per thread held in VGPRs;
 *     velocity accumulates in registers and is written once.
 *   - Sources staged through LDS in TILE=512-point tiles (24 KB stokeslet,
 *     48 KB stresslet). Inner-loop LDS reads are wave-uniform -> broadcast,
 *     conflict-free by construction.
 *   - fp64 rsqrt: raw v_rsq_f64 + one 4-op Newton step (see rsq_refined),
 *     skipping libm's fp-class edge handling because r^2 is a finite sum of
 *     squares and the r^2==0 case is masked explicitly (reference masks it
 *     too, kernels.cu:39,70).
 *   - Accumulation order per target is source order (tile-major), fixed ->
 *     bit-reproducible for a given shard layout and slice count.
 *   - Small target counts (< 512 workgroups at one target/thread) switch to
 *     a source-split launch: gridDim.y source slices accumulate partials
 *     that a second kernel reduces in fixed slice order (deterministic).
 *
 * Roofline: compute-bound on the fp64 VALU. The kernels are
 * deliberately not GEMM-shaped: gfx950's fp64 matrix rate equals its vector
 * rate, so MFMA buys nothing here.
 */

#include <hip/hip_runtime.h>

#include <cstdlib>
#include <map>
#include <mutex>
#include <vector>

#define BLOCK 256
#define TILE 512

/* v_rsq_f64 + one refinement step. Inputs here are finite and >= 0;
 * x == 0 must be masked by the caller. */
__device__ inline double rsq_refined(double x) {
    double y;
    asm("v_rsq_f64 %0, %1" : "=v"(y) : "v"(x));
#ifdef SKELLY_RSQ_PRECISE
    /* 5-op Householder (2nd order) step — the same polish ROCm libm applies:
     * cubic convergence, <= 2 ulp of fp64. */
    const double e = __builtin_fma(-x * y, y, 1.0);
    const double c = __builtin_fma(e, 0.375, 0.5);
    return __builtin_fma(y * e, c, y);
#else
    /* Default: 4-op quadratic Newton step, y' = -(0.5*y) * ((x*y)*y - 3).
     * Measured on MI355X (profiles/rocprof_r01.md): identical norm-relative
     * parity vs the CPU oracle (~5e-16 at 2e4x4096 incl. near-coincident
     * pairs) and +4.6%/+2.9%/+2.2% on stokeslet/stresslet/oseen over the
     * Householder form — v_rsq_f64's seed is accurate enough that one
     * quadratic step reaches the fp64 rounding floor of the summation. */
    const double t = x * y;
    const double w = __builtin_fma(t, y, -3.0);
    return (-0.5 * y) * w;
#endif
}

/* ---- kernel functors ------------------------------------------------- */

struct Stokeslet {
    static constexpr int SRCDIM = 3;
    struct Params {
        double scale; /* 1/(8*pi) [ / eta when folded] */
    };
    /* u += (1/r)(f + rhat (f.rhat)); r = t - s; r==0 -> 0 (kernels.cu:62-76) */
    __device__ static inline void pair(const double t[3], const double sp[3], const double f[3],
                                       double acc[3], const Params &) {
        const double dx = t[0] - sp[0];
        const double dy = t[1] - sp[1];
        const double dz = t[2] - sp[2];
        double r2 = dx * dx;
        r2 = __builtin_fma(dy, dy, r2);
        r2 = __builtin_fma(dz, dz, r2);
        double rinv = rsq_refined(r2);
        rinv = (r2 == 0.0) ? 0.0 : rinv; /* kernels.cu:70 */
        const double rinv2 = rinv * rinv;
        double fdotr = f[0] * dx;
        fdotr = __builtin_fma(f[1], dy, fdotr);
        fdotr = __builtin_fma(f[2], dz, fdotr);
        const double inner = fdotr * rinv2;
        acc[0] = __builtin_fma(rinv, __builtin_fma(dx, inner, f[0]), acc[0]);
        acc[1] = __builtin_fma(rinv, __builtin_fma(dy, inner, f[1]), acc[1]);
        acc[2] = __builtin_fma(rinv, __builtin_fma(dz, inner, f[2]), acc[2]);
    }
    __device__ static inline double finish(double a, const Params &p) { return a * p.scale; }
};

struct Stresslet {
    static constexpr int SRCDIM = 9;
    struct Params {
        double scale; /* 1/(8*pi) [ / eta when folded] */
    };
    /* u += -3 (d^T S d)/r^5 d; d = t - s; r==0 -> 0 (kernels.cu:29-54) */
    __device__ static inline void pair(const double t[3], const double sp[3], const double f[9],
                                       double acc[3], const Params &) {
        const double dx = t[0] - sp[0];
        const double dy = t[1] - sp[1];
        const double dz = t[2] - sp[2];
        const double dxx = dx * dx, dyy = dy * dy, dzz = dz * dz;
        double dr2 = dxx + dyy;
        dr2 += dzz;
        const double rinv = rsq_refined(dr2);
        const double rinv2 = rinv * rinv;
        double rinv5 = rinv * rinv2 * rinv2;
        rinv5 = (dr2 == 0.0) ? 0.0 : rinv5; /* kernels.cu:39 */
        /* coeff = sxx dx^2 + syy dy^2 + szz dz^2 + (sxy+syx) dx dy
         *       + (sxz+szx) dx dz + (syz+szy) dy dz   (kernels.cu:45-48) */
        double coeff = f[0] * dxx;
        coeff = __builtin_fma(f[4], dyy, coeff);
        coeff = __builtin_fma(f[8], dzz, coeff);
        coeff = __builtin_fma(f[1] + f[3], dx * dy, coeff);
        coeff = __builtin_fma(f[2] + f[6], dx * dz, coeff);
        coeff = __builtin_fma(f[5] + f[7], dy * dz, coeff);
        coeff *= -3.0 * rinv5;
        acc[0] = __builtin_fma(dx, coeff, acc[0]);
        acc[1] = __builtin_fma(dy, coeff, acc[1]);
        acc[2] = __builtin_fma(dz, coeff, acc[2]);
    }
    __device__ static inline double finish(double a, const Params &p) { return a * p.scale; }
};

struct OseenContract {
    static constexpr int SRCDIM = 3;
    struct Params {
        double scale; /* factor = 1/(8*pi*eta), hoisted out of the pair loop */
        double reg2;
        double eps2;
    };
    /* kernels.cpp:96-127: dr==0 skipped; dr > eps -> fr=1/dr, gr=1/dr^3;
     * else fr=1/sqrt(dr^2+reg^2), gr that cubed. The dr > eps comparison is
     * evaluated as dr2 > eps^2 (monotone equivalent for exact values; can
     * differ only when sqrt rounding straddles eps itself). `factor` is
     * applied once at the end (linear in the sum). */
    __device__ static inline void pair(const double t[3], const double sp[3], const double rho[3],
                                       double acc[3], const Params &p) {
        const double dx = sp[0] - t[0]; /* src - trg, kernels.cpp:99-101 */
        const double dy = sp[1] - t[1];
        const double dz = sp[2] - t[2];
        double dr2 = dx * dx;
        dr2 = __builtin_fma(dy, dy, dr2);
        dr2 = __builtin_fma(dz, dz, dr2);
        const double denom2 = (dr2 > p.eps2) ? dr2 : dr2 + p.reg2;
        double y = rsq_refined(denom2);
        y = (dr2 == 0.0) ? 0.0 : y; /* kernels.cpp:105-106 skip */
        const double y3 = y * (y * y);
        double ddotrho = dx * rho[0];
        ddotrho = __builtin_fma(dy, rho[1], ddotrho);
        ddotrho = __builtin_fma(dz, rho[2], ddotrho);
        const double c = y3 * ddotrho;
        acc[0] = __builtin_fma(y, rho[0], acc[0]);
        acc[1] = __builtin_fma(y, rho[1], acc[1]);
        acc[2] = __builtin_fma(y, rho[2], acc[2]);
        acc[0] = __builtin_fma(dx, c, acc[0]);
        acc[1] = __builtin_fma(dy, c, acc[1]);
        acc[2] = __builtin_fma(dz, c, acc[2]);
    }
    __device__ static inline double finish(double a, const Params &p) { return a * p.scale; }
};

struct Rotlet {
    static constexpr int SRCDIM = 3;
    struct Params {
        double scale; /* factor = 1/(8*pi*eta), applied at the end as the
                         reference does (kernels.cpp:239) */
        double reg2;
        double eps2;
    };
    /* kernels.cpp:218-236: dr = dr2 < eps^2 ? sqrt(reg2+dr2) : sqrt(dr2)
     * (no dr==0 skip); fr = 1/dr^3; u += fr * (d x' rho) with the reference's
     * sign convention u_x += fr*(dz*rho_y - dy*rho_z), d = trg - src. */
    __device__ static inline void pair(const double t[3], const double sp[3], const double rho[3],
                                       double acc[3], const Params &p) {
        const double dx = t[0] - sp[0];
        const double dy = t[1] - sp[1];
        const double dz = t[2] - sp[2];
        double dr2 = dx * dx;
        dr2 = __builtin_fma(dy, dy, dr2);
        dr2 = __builtin_fma(dz, dz, dr2);
        const double denom2 = (dr2 < p.eps2) ? dr2 + p.reg2 : dr2;
        const double y = rsq_refined(denom2);
        const double fr = y * (y * y);
        const double cx = __builtin_fma(dz, rho[1], -(dy * rho[2]));
        const double cy = __builtin_fma(dx, rho[2], -(dz * rho[0]));
        const double cz = __builtin_fma(dy, rho[0], -(dx * rho[1]));
        acc[0] = __builtin_fma(fr, cx, acc[0]);
        acc[1] = __builtin_fma(fr, cy, acc[1]);
        acc[2] = __builtin_fma(fr, cz, acc[2]);
    }
    __device__ static inline double finish(double a, const Params &p) { return a * p.scale; }
};

struct StressletNormalDensity {
    static constexpr int SRCDIM = 6; /* normal[3] + density[3] per source */
    struct Params {
        double scale; /* -3/(4*pi), kernels.cpp:311 (no eta dependence) */
        double reg2;
        double eps2;
    };
    /* kernels::stresslet_times_normal_times_density (kernels.cpp:307-334):
     * Sdn_i += (d.rho_j)(d.n_j)/r^5 d, d = trg - src; r_norm < eps ->
     * sqrt(r^2+reg^2) (kernels.cpp:320-323). The reference's i==j skip is a
     * d==0 mask here (identical: the numerator is ~d^3). */
    __device__ static inline void pair(const double t[3], const double sp[3], const double f[6],
                                       double acc[3], const Params &p) {
        const double dx = t[0] - sp[0];
        const double dy = t[1] - sp[1];
        const double dz = t[2] - sp[2];
        double dr2 = dx * dx;
        dr2 = __builtin_fma(dy, dy, dr2);
        dr2 = __builtin_fma(dz, dz, dr2);
        const double denom2 = (dr2 < p.eps2) ? dr2 + p.reg2 : dr2;
        double y = rsq_refined(denom2);
        y = (dr2 == 0.0) ? 0.0 : y;
        const double y2 = y * y;
        const double rinv5 = y * y2 * y2;
        double ddn = dx * f[0];
        ddn = __builtin_fma(dy, f[1], ddn);
        ddn = __builtin_fma(dz, f[2], ddn);
        double ddrho = dx * f[3];
        ddrho = __builtin_fma(dy, f[4], ddrho);
        ddrho = __builtin_fma(dz, f[5], ddrho);
        const double f0 = ddrho * ddn * rinv5;
        acc[0] = __builtin_fma(f0, dx, acc[0]);
        acc[1] = __builtin_fma(f0, dy, acc[1]);
        acc[2] = __builtin_fma(f0, dz, acc[2]);
    }
    __device__ static inline double finish(double a, const Params &p) { return a * p.scale; }
};

/* ---- driver ----------------------------------------------------------- */

/* PARTIAL=false: each block sums ALL sources for its targets and writes the
 * finished (scaled) velocities. PARTIAL=true (source-split, used when small
 * target counts would underfill the 256 CUs): gridDim.y slices partition the
 * sources; block (x, y) sums slice y for target tile x into
 * u_trg[y * 3*n_trg ...] UNSCALED; reduce_partials then sums the slices in
 * fixed slice order (deterministic) and applies the finish scale. */
template <typename K, int TPT, bool PARTIAL = false>
__global__ __launch_bounds__(BLOCK) void pair_driver(const double *__restrict__ r_src,
                                                     const double *__restrict__ f_src,
                                                     const double *__restrict__ r_trg,
                                                     double *__restrict__ u_trg,
                                                     long long n_src, long long n_trg,
                                                     typename K::Params params) {
    __shared__ double lds[TILE * (3 + K::SRCDIM)];
    double *lds_r = lds;
    double *lds_f = lds + TILE * 3;
    const int tid = threadIdx.x;
    const long long base = (long long)blockIdx.x * (BLOCK * TPT);

    double tp[TPT][3];
    double acc[TPT][3];
#pragma unroll
    for (int k = 0; k < TPT; ++k) {
        long long it = base + (long long)k * BLOCK + tid;
        const long long itc = it < n_trg ? it : (n_trg > 0 ? n_trg - 1 : 0);
        tp[k][0] = r_trg[3 * itc + 0];
        tp[k][1] = r_trg[3 * itc + 1];
        tp[k][2] = r_trg[3 * itc + 2];
        acc[k][0] = acc[k][1] = acc[k][2] = 0.0;
    }

    long long src_begin = 0, src_end = n_src;
    if (PARTIAL) {
        const long long chunk = (n_src + gridDim.y - 1) / gridDim.y;
        src_begin = (long long)blockIdx.y * chunk;
        src_end = src_begin + chunk < n_src ? src_begin + chunk : n_src;
    }

    for (long long tile0 = src_begin; tile0 < src_end; tile0 += TILE) {
        const int m = (int)((src_end - tile0 < TILE) ? (src_end - tile0) : TILE);
        __syncthreads();
        for (int i = tid; i < m * 3; i += BLOCK)
            lds_r[i] = r_src[tile0 * 3 + i];
        for (int i = tid; i < m * K::SRCDIM; i += BLOCK)
            lds_f[i] = f_src[tile0 * K::SRCDIM + i];
        __syncthreads();

        /* Unroll by U sources: all U sources' LDS reads issue together (one
         * lgkmcnt wait per U sources instead of several per source) and
         * their v_rsq/refine chains interleave. */
#ifndef SKELLY_UNROLL
#define SKELLY_UNROLL 4
#endif
        constexpr int U = SKELLY_UNROLL;
        int s = 0;
        for (; s + U <= m; s += U) {
            double sp[U][3], sf[U][K::SRCDIM];
#pragma unroll
            for (int u = 0; u < U; ++u) {
#pragma unroll
                for (int j = 0; j < 3; ++j)
                    sp[u][j] = lds_r[3 * (s + u) + j];
#pragma unroll
                for (int j = 0; j < K::SRCDIM; ++j)
                    sf[u][j] = lds_f[K::SRCDIM * (s + u) + j];
            }
#pragma unroll
            for (int u = 0; u < U; ++u)
#pragma unroll
                for (int k = 0; k < TPT; ++k)
                    K::pair(tp[k], sp[u], sf[u], acc[k], params);
        }
        for (; s < m; ++s) {
            double sp[3], sf[K::SRCDIM];
#pragma unroll
            for (int j = 0; j < 3; ++j)
                sp[j] = lds_r[3 * s + j];
#pragma unroll
            for (int j = 0; j < K::SRCDIM; ++j)
                sf[j] = lds_f[K::SRCDIM * s + j];
#pragma unroll
            for (int k = 0; k < TPT; ++k)
                K::pair(tp[k], sp, sf, acc[k], params);
        }
    }

#pragma unroll
    for (int k = 0; k < TPT; ++k) {
        const long long it = base + (long long)k * BLOCK + tid;
        if (it < n_trg) {
            if (PARTIAL) {
                double *p = u_trg + (long long)blockIdx.y * 3 * n_trg;
                p[3 * it + 0] = acc[k][0];
                p[3 * it + 1] = acc[k][1];
                p[3 * it + 2] = acc[k][2];
            } else {
                u_trg[3 * it + 0] = K::finish(acc[k][0], params);
                u_trg[3 * it + 1] = K::finish(acc[k][1], params);
                u_trg[3 * it + 2] = K::finish(acc[k][2], params);
            }
        }
    }
}

/* Sum source-slice partials in fixed slice order and apply the finish scale.
 * partial layout: [n_slices][n_trg][3]; one thread per (target, component). */
template <typename K>
__global__ __launch_bounds__(BLOCK) void reduce_partials(const double *__restrict__ partial,
                                                         double *__restrict__ u_trg,
                                                         long long n_trg, int n_slices,
                                                         typename K::Params params) {
    const long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; /* 3*n_trg elems */
    if (i >= 3 * n_trg)
        return;
    double s = 0.0;
    for (int y = 0; y < n_slices; ++y)
        s += partial[(long long)y * 3 * n_trg + i];
    u_trg[i] = K::finish(s, params);
}

/* ---- launch helpers (host) -------------------------------------------- */

namespace skelly {

/* Optional persistent split-K workspace (SKELLY_PERSISTENT_WS=1): replaces
 * the per-launch hipMallocAsync/hipFreeAsync pair with a grow-only buffer
 * cached per stream. Reuse across launches is safe because launches on one
 * stream are ordered; outgrown blocks are RETIRED (never freed or reused —
 * queued work may still reference them; geometric growth bounds the leak at
 * ~2x the final size). Default OFF: the stream-ordered mempool path is the
 * round-1-validated behavior, and deep queues of outstanding async
 * alloc/free pairs are the prime suspect for the sync-cadence corruption
 * documented in gmres.py / DESIGN.md — flip this on to test that
 * hypothesis. */
static int g_persistent_ws_override = -1; /* -1 = follow env var */

static bool persistent_ws_enabled() {
    if (g_persistent_ws_override >= 0)
        return g_persistent_ws_override != 0;
    static const bool on = [] {
        const char *e = getenv("SKELLY_PERSISTENT_WS");
        return e && atoi(e) != 0;
    }();
    return on;
}

static double *persistent_ws(hipStream_t stream, size_t bytes) {
    struct Entry {
        void *ptr = nullptr;
        size_t cap = 0;
    };
    static std::mutex mu;
    static std::map<hipStream_t, Entry> cache;
    static std::vector<void *> retired;
    std::lock_guard<std::mutex> lk(mu);
    Entry &e = cache[stream];
    if (e.cap < bytes) {
        size_t want = bytes * 2;
        void *p = nullptr;
        if (hipMalloc(&p, want) != hipSuccess)
            return nullptr; /* caller falls back to the async path */
        if (e.ptr)
            retired.push_back(e.ptr);
        e.ptr = p;
        e.cap = want;
    }
    return static_cast<double *>(e.ptr);
}

static inline int pick_tpt(long long n_trg) {
    /* Prefer 4 targets/thread (amortizes the per-source LDS broadcast reads
     * 4x); occupancy for small target counts is recovered by source
     * splitting, not by shrinking TPT. */
    if (n_trg >= BLOCK * 4)
        return 4;
    if (n_trg >= BLOCK * 2)
        return 2;
    return 1;
}

template <typename K>
hipError_t launch_pair(const double *r_src, const double *f_src, const double *r_trg,
                       double *u_trg, long long n_src, long long n_trg,
                       typename K::Params params, hipStream_t stream) {
    if (n_trg <= 0)
        return hipSuccess;
    const int tpt = pick_tpt(n_trg);
    const long long blocks = (n_trg + (long long)BLOCK * tpt - 1) / ((long long)BLOCK * tpt);

    /* Source-split when the target grid alone underfills the chip (256 CUs;
     * aim >= SKELLY_SPLIT_TARGET workgroups, default 1024 — measured +11%
     * over 512 at 1e5 x 1e5) and there are enough sources to slice. */
    static const long long split_target = [] {
        const char *e = getenv("SKELLY_SPLIT_TARGET");
        long long v = e ? atoll(e) : 1024;
        return v > 0 ? v : 1024;
    }();
    int n_slices = 1;
    if (blocks < split_target) {
        long long s = (split_target + blocks - 1) / blocks;
        long long max_by_src = (n_src + 2047) / 2048; /* keep >= ~2048 src/slice */
        n_slices = (int)(s < max_by_src ? s : max_by_src);
        if (n_slices < 1)
            n_slices = 1;
    }

    dim3 block(BLOCK);
    if (n_slices == 1) {
        dim3 grid((unsigned)blocks);
        switch (tpt) {
        case 4:
            hipLaunchKernelGGL((pair_driver<K, 4>), grid, block, 0, stream, r_src, f_src,
                               r_trg, u_trg, n_src, n_trg, params);
            break;
        case 2:
            hipLaunchKernelGGL((pair_driver<K, 2>), grid, block, 0, stream, r_src, f_src,
                               r_trg, u_trg, n_src, n_trg, params);
            break;
        default:
            hipLaunchKernelGGL((pair_driver<K, 1>), grid, block, 0, stream, r_src, f_src,
                               r_trg, u_trg, n_src, n_trg, params);
            break;
        }
        return hipGetLastError();
    }

    /* split path: same TPT, gridDim.y source slices into a partial buffer */
    double *workspace = nullptr;
    bool pooled = false;
    const size_t ws_bytes = (size_t)n_slices * 3 * n_trg * sizeof(double);
    if (persistent_ws_enabled()) {
        workspace = persistent_ws(stream, ws_bytes);
        pooled = workspace != nullptr;
    }
    if (!pooled) {
        hipError_t err = hipMallocAsync((void **)&workspace, ws_bytes, stream);
        if (err != hipSuccess)
            return err;
    }
    dim3 grid((unsigned)blocks, (unsigned)n_slices);
    switch (tpt) {
    case 4:
        hipLaunchKernelGGL((pair_driver<K, 4, true>), grid, block, 0, stream, r_src, f_src,
                           r_trg, workspace, n_src, n_trg, params);
        break;
    case 2:
        hipLaunchKernelGGL((pair_driver<K, 2, true>), grid, block, 0, stream, r_src, f_src,
                           r_trg, workspace, n_src, n_trg, params);
        break;
    default:
        hipLaunchKernelGGL((pair_driver<K, 1, true>), grid, block, 0, stream, r_src, f_src,
                           r_trg, workspace, n_src, n_trg, params);
        break;
    }
    const long long rblocks = (3 * n_trg + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL((reduce_partials<K>), dim3((unsigned)rblocks), block, 0, stream,
                       workspace, u_trg, n_trg, n_slices, params);
    hipError_t err = hipGetLastError();
    if (pooled)
        return err;
    hipError_t err2 = hipFreeAsync(workspace, stream);
    return err != hipSuccess ? err : err2;
}

hipError_t launch_stokeslet(const double *r_src, const double *f_src, const double *r_trg,
                            double *u_trg, long long n_src, long long n_trg, double scale,
                            hipStream_t stream) {
    Stokeslet::Params p{scale};
    return launch_pair<Stokeslet>(r_src, f_src, r_trg, u_trg, n_src, n_trg, p, stream);
}

hipError_t launch_stresslet(const double *r_src, const double *f_src, const double *r_trg,
                            double *u_trg, long long n_src, long long n_trg, double scale,
                            hipStream_t stream) {
    Stresslet::Params p{scale};
    return launch_pair<Stresslet>(r_src, f_src, r_trg, u_trg, n_src, n_trg, p, stream);
}

hipError_t launch_oseen(const double *r_src, const double *density, const double *r_trg,
                        double *u_trg, long long n_src, long long n_trg, double factor,
                        double reg, double eps, hipStream_t stream) {
    OseenContract::Params p{factor, reg * reg, eps * eps};
    return launch_pair<OseenContract>(r_src, density, r_trg, u_trg, n_src, n_trg, p, stream);
}

hipError_t launch_rotlet(const double *r_src, const double *density, const double *r_trg,
                         double *u_trg, long long n_src, long long n_trg, double factor,
                         double reg, double eps, hipStream_t stream) {
    Rotlet::Params p{factor, reg * reg, eps * eps};
    return launch_pair<Rotlet>(r_src, density, r_trg, u_trg, n_src, n_trg, p, stream);
}

hipError_t launch_stresslet_normal_density(const double *r_src, const double *nd,
                                           const double *r_trg, double *u_trg, long long n_src,
                                           long long n_trg, double reg, double eps,
                                           hipStream_t stream) {
    StressletNormalDensity::Params p{-3.0 / (4.0 * M_PI), reg * reg, eps * eps};
    return launch_pair<StressletNormalDensity>(r_src, nd, r_trg, u_trg, n_src, n_trg, p, stream);
}

/* ---- batched oseen_tensor_direct dense builder (kernels.cpp:146-195) ---
 * points: (nf, n, 3); G: (nf, 3n, 3n) row-major (G is symmetric, so this is
 * bit-identical to the reference's col-major). One thread per (fiber, trg,
 * src) pair writes its 3x3 block. Setup-time op (per-fiber self-stokeslet,
 * fiber_finite_difference.cpp:56), not GMRES-hot. */
__global__ __launch_bounds__(BLOCK) void oseen_tensor_kernel(const double *__restrict__ pts,
                                                             double *__restrict__ G,
                                                             long long nf, long long n,
                                                             double factor, double reg2,
                                                             double eps2) {
    const long long idx = (long long)blockIdx.x * BLOCK + threadIdx.x;
    const long long total = nf * n * n;
    if (idx >= total)
        return;
    const long long f = idx / (n * n);
    const long long r = idx % (n * n);
    const long long t = r / n, s = r % n;
    const double *p = pts + f * n * 3;
    const double dx = p[3 * s + 0] - p[3 * t + 0];
    const double dy = p[3 * s + 1] - p[3 * t + 1];
    const double dz = p[3 * s + 2] - p[3 * t + 2];
    double dr2 = dx * dx;
    dr2 = __builtin_fma(dy, dy, dr2);
    dr2 = __builtin_fma(dz, dz, dr2);
    const double denom2 = (dr2 > eps2) ? dr2 : dr2 + reg2;
    double y = rsq_refined(denom2);
    y = (dr2 == 0.0) ? 0.0 : y; /* dr2==0 -> whole block zero (cpp:166-167) */
    const double fr = factor * y;
    const double gr = fr * (y * y);
    const long long ld = 3 * n;
    double *blk = G + f * ld * ld + (3 * t) * ld + 3 * s;
    const double d[3] = {dx, dy, dz};
#pragma unroll
    for (int a = 0; a < 3; ++a)
#pragma unroll
        for (int b = 0; b < 3; ++b)
            blk[a * ld + b] = (a == b ? fr : 0.0) + gr * d[a] * d[b];
}

/* stresslet_times_normal dense builder (kernels.cpp:264-287):
 * Snormal block (i, j) = -3/(4*pi) * (d.n_j)/r^5 * d d^T, d = r_i - r_j;
 * i == j blocks zero (kernels.cpp:273-274); r < eps regularized
 * (kernels.cpp:278-279). Output is (3n, 3n) C row-major with
 * out[3i+a][3j+b] = Snormal(3i+a, 3j+b) — same logical indexing as the
 * reference (its Eigen buffer is col-major; the MATRIX is identical). */
__global__ __launch_bounds__(BLOCK) void stresslet_normal_kernel(
    const double *__restrict__ pts, const double *__restrict__ normals, double *__restrict__ G,
    long long n, double factor, double reg2, double eps2) {
    const long long idx = (long long)blockIdx.x * BLOCK + threadIdx.x;
    if (idx >= n * n)
        return;
    const long long i = idx / n, j = idx % n;
    const long long ld = 3 * n;
    double *blk = G + (3 * i) * ld + 3 * j;
    if (i == j) {
#pragma unroll
        for (int a = 0; a < 3; ++a)
#pragma unroll
            for (int b = 0; b < 3; ++b)
                blk[a * ld + b] = 0.0;
        return;
    }
    const double dx = pts[3 * i + 0] - pts[3 * j + 0];
    const double dy = pts[3 * i + 1] - pts[3 * j + 1];
    const double dz = pts[3 * i + 2] - pts[3 * j + 2];
    double dr2 = dx * dx;
    dr2 = __builtin_fma(dy, dy, dr2);
    dr2 = __builtin_fma(dz, dz, dr2);
    const double denom2 = (dr2 < eps2) ? dr2 + reg2 : dr2;
    const double y = rsq_refined(denom2);
    const double y2 = y * y;
    const double rinv5 = y * y2 * y2;
    double ddn = dx * normals[3 * j + 0];
    ddn = __builtin_fma(dy, normals[3 * j + 1], ddn);
    ddn = __builtin_fma(dz, normals[3 * j + 2], ddn);
    const double c = factor * ddn * rinv5;
    const double d[3] = {dx, dy, dz};
#pragma unroll
    for (int a = 0; a < 3; ++a)
#pragma unroll
        for (int b = 0; b < 3; ++b)
            blk[a * ld + b] = c * d[a] * d[b];
}

hipError_t launch_stresslet_times_normal(const double *pts, const double *normals, double *G,
                                         long long n, double reg, double eps,
                                         hipStream_t stream) {
    if (n <= 0)
        return hipSuccess;
    const long long blocks = (n * n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(stresslet_normal_kernel, dim3((unsigned)blocks), dim3(BLOCK), 0, stream,
                       pts, normals, G, n, -3.0 / (4.0 * M_PI), reg * reg, eps * eps);
    return hipGetLastError();
}

hipError_t launch_oseen_tensor_batched(const double *pts, double *G, long long nf, long long n,
                                       double eta, double reg, double eps, hipStream_t stream) {
    if (nf <= 0 || n <= 0)
        return hipSuccess;
    const double factor = 1.0 / (8.0 * M_PI * eta);
    const long long total = nf * n * n;
    const long long blocks = (total + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(oseen_tensor_kernel, dim3((unsigned)blocks), dim3(BLOCK), 0, stream, pts,
                       G, nf, n, factor, reg * reg, eps * eps);
    return hipGetLastError();
}

/* ---- fp64 FMA peak microbenchmark (roofline denominator) -------------- */

__global__ __launch_bounds__(256) void fp64_fma_peak_kernel(double *out, int iters) {
    /* 16 independent accumulators: with 8, the chain is fp64-FMA
     * dependent-latency-bound (~20% below issue rate) and under-reports the
     * sustained peak. */
    double a[16];
#pragma unroll
    for (int k = 0; k < 16; ++k)
        a[k] = 1.0 + 1e-9 * threadIdx.x + 0.1 * k;
    const double b = 1.0 + 1e-12, c = 1e-12;
    for (int i = 0; i < iters; ++i) {
#pragma unroll
        for (int k = 0; k < 16; ++k)
            a[k] = __builtin_fma(a[k], b, c);
    }
    double s = 0.0;
#pragma unroll
    for (int k = 0; k < 16; ++k)
        s += a[k];
    if (s == -1.0) /* never true; defeats DCE without a store per thread */
        out[blockIdx.x] = s;
}

hipError_t run_fp64_peak(double *out_tflops) {
    const int iters = 100000, blocks = 2048, threads = 256;
    double *d;
    hipError_t err = hipMalloc(&d, blocks * sizeof(double));
    if (err != hipSuccess)
        return err;
    hipEvent_t t0, t1;
    (void)hipEventCreate(&t0);
    (void)hipEventCreate(&t1);
    /* warmup */
    hipLaunchKernelGGL(fp64_fma_peak_kernel, dim3(blocks), dim3(threads), 0, 0, d, iters / 10);
    (void)hipEventRecord(t0);
    hipLaunchKernelGGL(fp64_fma_peak_kernel, dim3(blocks), dim3(threads), 0, 0, d, iters);
    (void)hipEventRecord(t1);
    err = hipEventSynchronize(t1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, t0, t1);
    (void)hipEventDestroy(t0);
    (void)hipEventDestroy(t1);
    (void)hipFree(d);
    if (err != hipSuccess)
        return err;
    const double flops = 2.0 * 16.0 * (double)iters * (double)blocks * threads;
    *out_tflops = flops / (ms * 1e-3) / 1e12;
    return hipSuccess;
}

} // namespace skelly

/* Runtime override of the persistent split-K workspace (see
 * persistent_ws_enabled above): hipGraph capture of the GMRES iteration
 * (gmres.py use_graph) requires it — hipMallocAsync nodes cannot be
 * recorded reliably, and the grow-only hipMalloc happens during the
 * UNCAPTURED warmup pass, so capture itself allocates nothing. */
extern "C" void skelly_set_persistent_ws(int on) {
    skelly::g_persistent_ws_override = on;
}
