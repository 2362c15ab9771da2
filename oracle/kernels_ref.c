/* oracle/kernels_ref.c — CPU restatement of SkellySim's hot-path pair kernels.
 *
 * TEST INFRASTRUCTURE ONLY. This file is the parity oracle and the bench.py
 * `cpu_baseline` leg. It must never be linked into, imported by, or dispatched
 * from the product path (skellysim_amd/). Only tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg may call it.
 *
 * Each function restates, formula by formula, the reference implementation in
 * flatironinstitute/SkellySim v0.10.0 (read-only at /root/reference):
 *
 *   stokeslet : src/core/kernels.cu:62-76 (StokesCuda<double>::uKernel; identical
 *               math to the pvfmm::stokes_vel path used by
 *               kernels::stokeslet_direct_cpu, src/core/kernels.cpp:54-67),
 *               scale 1/(8*pi) per target (kernels.cu:59,122), then /eta
 *               (kernels.cpp:66,365).
 *   stresslet : src/core/kernels.cu:29-54 (StokesDoubleLayerCuda<double>::uKernel
 *               == the SIMD functor stokes_doublevel, kernels.cpp:11-40),
 *               scale 1/(8*pi) then /eta (kernels.cpp:82,358).
 *   oseen_contract (regularized Stokeslet):
 *               src/core/kernels.cpp:85-131 (kernels::oseen_tensor_contract_direct),
 *               defaults reg=5e-3, epsilon_distance=1e-5 (include/kernels.hpp:34-35).
 *               dr==0 pairs are skipped (kernels.cpp:105-106).
 *   rotlet    : src/core/kernels.cpp:206-242 (kernels::rotlet), defaults as above
 *               (include/kernels.hpp:44-45). Note: no dr==0 skip — dr2<eps2 is
 *               regularized to sqrt(reg2+dr2) (kernels.cpp:225), finite at dr=0.
 *
 * The OpenMP parallelization mirrors the reference's static target-chunking
 * (get_chunk_start_and_size, src/core/kernels.cpp:42-51, used at 58-65): the
 * target range is split into contiguous per-thread chunks; accumulation over
 * sources is in source order per target, so results are independent of the
 * thread count (bit-reproducible across n_threads).
 *
 * Array layout matches the reference Evaluator (include/kernels.hpp:14-15):
 * col-major 3 x n (per-point xyz contiguous): r[3*i+0..2] = point i.
 * Stresslet strengths are 9 x n: f[9*i+k], k = i_row*3 + j_col, the 9-component
 * double-layer tensor S (see kernels.cu:41-43).
 */

#include <math.h>
#include <stddef.h>
#ifdef _OPENMP
#include <omp.h>
#endif

#ifndef M_PI
#define M_PI 3.14159265358979323846
#endif

/* get_chunk_start_and_size — restates src/core/kernels.cpp:42-51 */
static void chunk_start_size(int i_thr, int n_thr, int prob_size, int *start, int *size) {
    const int chunk_small = prob_size / n_thr;
    const int chunk_big = chunk_small + 1;
    const int remainder = prob_size % n_thr;
    if (i_thr < remainder) {
        *start = chunk_big * i_thr;
        *size = chunk_big;
    } else {
        *start = remainder * chunk_big + (i_thr - remainder) * chunk_small;
        *size = chunk_small;
    }
}

/* ---- Stokeslet: u(t) = (1/(8*pi*eta)) * sum_s (1/r) (f + rhat (f . rhat)),
 *      r = t - s; r == 0 contributes 0.   kernels.cu:62-76 */
static void stokeslet_chunk(const double *r_src, const double *f_src, long n_src,
                            const double *r_trg, double *u_trg, long t0, long t1, double eta) {
    const double scale = 1.0 / 8.0 / M_PI; /* kernels.cu:59 */
    for (long t = t0; t < t1; ++t) {
        const double tx = r_trg[3 * t + 0], ty = r_trg[3 * t + 1], tz = r_trg[3 * t + 2];
        double ux = 0.0, uy = 0.0, uz = 0.0;
        for (long s = 0; s < n_src; ++s) {
            const double dx = tx - r_src[3 * s + 0]; /* dr = rj - ri = trg - src, kernels.cu:65-67 */
            const double dy = ty - r_src[3 * s + 1];
            const double dz = tz - r_src[3 * s + 2];
            const double r2 = dx * dx + dy * dy + dz * dz;
            const double rinv = r2 == 0.0 ? 0.0 : 1.0 / sqrt(r2); /* kernels.cu:70 */
            const double rinv2 = rinv * rinv;
            const double fx = f_src[3 * s + 0], fy = f_src[3 * s + 1], fz = f_src[3 * s + 2];
            const double inner = (fx * dx + fy * dy + fz * dz) * rinv2;
            ux += rinv * (fx + dx * inner);
            uy += rinv * (fy + dy * inner);
            uz += rinv * (fz + dz * inner);
        }
        /* per-target scale (kernels.cu:121-122), then /eta (kernels.cpp:66,365) */
        u_trg[3 * t + 0] = ux * scale / eta;
        u_trg[3 * t + 1] = uy * scale / eta;
        u_trg[3 * t + 2] = uz * scale / eta;
    }
}

void oracle_stokeslet(const double *r_src, const double *f_src, long n_src,
                      const double *r_trg, double *u_trg, long n_trg, double eta) {
#ifdef _OPENMP
#pragma omp parallel
    {
        int start, size;
        chunk_start_size(omp_get_thread_num(), omp_get_num_threads(), (int)n_trg, &start, &size);
        stokeslet_chunk(r_src, f_src, n_src, r_trg, u_trg, start, start + size, eta);
    }
#else
    stokeslet_chunk(r_src, f_src, n_src, r_trg, u_trg, 0, n_trg, eta);
#endif
}

/* ---- Stresslet (double-layer): u(t) = (1/(8*pi*eta)) * sum_s -3 (d^T S d)/r^5 d,
 *      d = t - s; r == 0 contributes 0.   kernels.cu:29-54 */
static void stresslet_chunk(const double *r_src, const double *f_src, long n_src,
                            const double *r_trg, double *u_trg, long t0, long t1, double eta) {
    const double scale = 1.0 / 8.0 / M_PI; /* kernels.cu:26 */
    for (long t = t0; t < t1; ++t) {
        const double tx = r_trg[3 * t + 0], ty = r_trg[3 * t + 1], tz = r_trg[3 * t + 2];
        double ux = 0.0, uy = 0.0, uz = 0.0;
        for (long s = 0; s < n_src; ++s) {
            const double dx = tx - r_src[3 * s + 0]; /* dr = rj - ri, kernels.cu:32-34 */
            const double dy = ty - r_src[3 * s + 1];
            const double dz = tz - r_src[3 * s + 2];
            const double dr2 = dx * dx + dy * dy + dz * dz;
            const double rinv = 1.0 / sqrt(dr2); /* kernels.cu:37 (inf at 0, masked below) */
            const double rinv2 = rinv * rinv;
            const double rinv5 = dr2 ? rinv * rinv2 * rinv2 : 0.0; /* kernels.cu:39 */
            const double *f = &f_src[9 * s];
            const double sxx = f[0], sxy = f[1], sxz = f[2];
            const double syx = f[3], syy = f[4], syz = f[5];
            const double szx = f[6], szy = f[7], szz = f[8];
            double coeff = sxx * dx * dx + syy * dy * dy + szz * dz * dz;
            coeff += (sxy + syx) * dx * dy;
            coeff += (sxz + szx) * dx * dz;
            coeff += (syz + szy) * dy * dz;
            coeff *= -3.0 * rinv5;
            ux += dx * coeff;
            uy += dy * coeff;
            uz += dz * coeff;
        }
        u_trg[3 * t + 0] = ux * scale / eta;
        u_trg[3 * t + 1] = uy * scale / eta;
        u_trg[3 * t + 2] = uz * scale / eta;
    }
}

void oracle_stresslet(const double *r_src, const double *f_src, long n_src,
                      const double *r_trg, double *u_trg, long n_trg, double eta) {
#ifdef _OPENMP
#pragma omp parallel
    {
        int start, size;
        chunk_start_size(omp_get_thread_num(), omp_get_num_threads(), (int)n_trg, &start, &size);
        stresslet_chunk(r_src, f_src, n_src, r_trg, u_trg, start, start + size, eta);
    }
#else
    stresslet_chunk(r_src, f_src, n_src, r_trg, u_trg, 0, n_trg, eta);
#endif
}

/* ---- Regularized Oseen contraction: kernels.cpp:85-131.
 *      dx = src - trg (kernels.cpp:99-101; symmetric kernel, sign immaterial).
 *      dr == 0 -> skipped (kernels.cpp:105-106).
 *      dr > epsilon_distance: fr = factor/dr, gr = factor/dr^3 (kernels.cpp:108-110)
 *      else: denom = sqrt(dr^2 + reg^2), fr = factor/denom, gr = factor/denom^3
 *      (kernels.cpp:111-115). factor = 1/(8*pi*eta) (kernels.cpp:94). */
static void oseen_chunk(const double *r_src, const double *r_trg, const double *density,
                        double *u_trg, long n_src, long t0, long t1,
                        double eta, double reg, double eps) {
    const double factor = 1.0 / (8.0 * M_PI * eta);
    const double reg2 = reg * reg;
    for (long t = t0; t < t1; ++t) {
        const double tx = r_trg[3 * t + 0], ty = r_trg[3 * t + 1], tz = r_trg[3 * t + 2];
        double ux = 0.0, uy = 0.0, uz = 0.0;
        for (long s = 0; s < n_src; ++s) {
            const double dx = r_src[3 * s + 0] - tx;
            const double dy = r_src[3 * s + 1] - ty;
            const double dz = r_src[3 * s + 2] - tz;
            const double dr2 = dx * dx + dy * dy + dz * dz;
            const double dr = sqrt(dr2);
            if (dr == 0.0)
                continue;
            double fr, gr;
            if (dr > eps) {
                fr = factor / dr;
                gr = factor / (dr * dr * dr);
            } else {
                const double denom_inv = 1.0 / sqrt(dr2 + reg2);
                fr = factor * denom_inv;
                gr = factor * denom_inv * denom_inv * denom_inv;
            }
            const double rx = density[3 * s + 0], ry = density[3 * s + 1], rz = density[3 * s + 2];
            /* res += M . rho with M = fr I + gr d d^T (kernels.cpp:117-126) */
            const double ddotrho = dx * rx + dy * ry + dz * rz;
            ux += fr * rx + gr * dx * ddotrho;
            uy += fr * ry + gr * dy * ddotrho;
            uz += fr * rz + gr * dz * ddotrho;
        }
        u_trg[3 * t + 0] = ux;
        u_trg[3 * t + 1] = uy;
        u_trg[3 * t + 2] = uz;
    }
}

void oracle_oseen_contract(const double *r_src, const double *r_trg, const double *density,
                           double *u_trg, long n_src, long n_trg,
                           double eta, double reg, double eps) {
#ifdef _OPENMP
#pragma omp parallel
    {
        int start, size;
        chunk_start_size(omp_get_thread_num(), omp_get_num_threads(), (int)n_trg, &start, &size);
        oseen_chunk(r_src, r_trg, density, u_trg, n_src, start, start + size, eta, reg, eps);
    }
#else
    oseen_chunk(r_src, r_trg, density, u_trg, n_src, 0, n_trg, eta, reg, eps);
#endif
}

/* ---- Rotlet: kernels.cpp:206-242. dx = trg - src (kernels.cpp:220-222).
 *      dr = dr2 < eps^2 ? sqrt(reg2 + dr2) : sqrt(dr2) (kernels.cpp:225) — no
 *      dr==0 skip. fr = 1/dr^3; u_x += fr*(dz*rho_y - dy*rho_z) etc.
 *      (kernels.cpp:227-236); u *= factor at the end (kernels.cpp:239). */
static void rotlet_chunk(const double *r_src, const double *r_trg, const double *density,
                         double *u_trg, long n_src, long t0, long t1,
                         double eta, double reg, double eps) {
    const double factor = 1.0 / (8.0 * M_PI * eta);
    const double eps2 = eps * eps;
    const double reg2 = reg * reg;
    for (long t = t0; t < t1; ++t) {
        const double tx = r_trg[3 * t + 0], ty = r_trg[3 * t + 1], tz = r_trg[3 * t + 2];
        double ux = 0.0, uy = 0.0, uz = 0.0;
        for (long s = 0; s < n_src; ++s) {
            const double dx = tx - r_src[3 * s + 0];
            const double dy = ty - r_src[3 * s + 1];
            const double dz = tz - r_src[3 * s + 2];
            const double dr2 = dx * dx + dy * dy + dz * dz;
            const double dr = dr2 < eps2 ? sqrt(reg2 + dr2) : sqrt(dr2);
            const double fr = 1.0 / (dr * dr * dr);
            const double rx = density[3 * s + 0], ry = density[3 * s + 1], rz = density[3 * s + 2];
            ux += fr * dz * ry - fr * dy * rz;
            uy += -fr * dz * rx + fr * dx * rz;
            uz += fr * dy * rx - fr * dx * ry;
        }
        u_trg[3 * t + 0] = ux * factor;
        u_trg[3 * t + 1] = uy * factor;
        u_trg[3 * t + 2] = uz * factor;
    }
}

void oracle_rotlet(const double *r_src, const double *r_trg, const double *density,
                   double *u_trg, long n_src, long n_trg,
                   double eta, double reg, double eps) {
#ifdef _OPENMP
#pragma omp parallel
    {
        int start, size;
        chunk_start_size(omp_get_thread_num(), omp_get_num_threads(), (int)n_trg, &start, &size);
        rotlet_chunk(r_src, r_trg, density, u_trg, n_src, start, start + size, eta, reg, eps);
    }
#else
    rotlet_chunk(r_src, r_trg, density, u_trg, n_src, 0, n_trg, eta, reg, eps);
#endif
}

/* ---- stresslet_times_normal_times_density: kernels.cpp:307-334.
 *      Sdn_i = -3/(4 pi) * sum_{j != i} (d.rho_j)(d.n_j)/|d|^5 d,
 *      d = r_i - r_j; r_norm < eps -> r_norm = sqrt(r_norm^2 + reg^2)
 *      (kernels.cpp:320-323); no eta dependence (factor kernels.cpp:311).
 *      The i == j skip (kernels.cpp:316-317) is literal here; the HIP kernel
 *      realizes it as a d==0 mask, identical because a coincident pair's
 *      numerator (d.rho)(d.n) d is identically zero. */
static void sndd_chunk(const double *r_src, const double *normals, const double *density,
                       double *out, long n, long t0, long t1, double reg, double eps) {
    const double factor = -3.0 / (4.0 * M_PI);
    const double reg2 = reg * reg;
    for (long i = t0; i < t1; ++i) {
        const double xi = r_src[3 * i + 0], yi = r_src[3 * i + 1], zi = r_src[3 * i + 2];
        double ax = 0.0, ay = 0.0, az = 0.0;
        for (long j = 0; j < n; ++j) {
            if (i == j)
                continue; /* kernels.cpp:316-317 */
            const double dx = xi - r_src[3 * j + 0];
            const double dy = yi - r_src[3 * j + 1];
            const double dz = zi - r_src[3 * j + 2];
            const double dr2 = dx * dx + dy * dy + dz * dz;
            double rn = sqrt(dr2);
            if (rn < eps)
                rn = sqrt(dr2 + reg2);
            const double rinv5 = 1.0 / (rn * rn * rn * rn * rn);
            const double ddrho = dx * density[3 * j + 0] + dy * density[3 * j + 1] +
                                 dz * density[3 * j + 2];
            const double ddn = dx * normals[3 * j + 0] + dy * normals[3 * j + 1] +
                               dz * normals[3 * j + 2];
            const double f0 = ddrho * ddn * rinv5;
            ax += f0 * dx;
            ay += f0 * dy;
            az += f0 * dz;
        }
        out[3 * i + 0] = ax * factor;
        out[3 * i + 1] = ay * factor;
        out[3 * i + 2] = az * factor;
    }
}

void oracle_stresslet_times_normal_times_density(const double *r_src, const double *normals,
                                                 const double *density, double *out, long n,
                                                 double reg, double eps) {
#ifdef _OPENMP
#pragma omp parallel
    {
        int start, size;
        chunk_start_size(omp_get_thread_num(), omp_get_num_threads(), (int)n, &start, &size);
        sndd_chunk(r_src, normals, density, out, n, start, start + size, reg, eps);
    }
#else
    sndd_chunk(r_src, normals, density, out, n, 0, n, reg, eps);
#endif
}

/* ---- oseen_tensor_direct dense builder: kernels.cpp:146-195 (square,
 *      r_src == r_trg layout — its only production use, the per-fiber
 *      self-stokeslet, fiber_finite_difference.cpp:56). G is (3n, 3n),
 *      block (t, s) = fr I + gr d d^T with d = r_s - r_t; dr2==0 blocks
 *      left zero (kernels.cpp:166-167). Row-major output (G is symmetric,
 *      so the layout matches Eigen's col-major bit for bit). */
void oracle_oseen_tensor(const double *pts, double *G, long n, double eta, double reg,
                         double eps) {
    const double factor = 1.0 / (8.0 * M_PI * eta);
    const double reg2 = reg * reg;
    const long ld = 3 * n;
    for (long t = 0; t < n; ++t)
        for (long s = 0; s < n; ++s) {
            const double dx = pts[3 * s + 0] - pts[3 * t + 0];
            const double dy = pts[3 * s + 1] - pts[3 * t + 1];
            const double dz = pts[3 * s + 2] - pts[3 * t + 2];
            const double dr2 = dx * dx + dy * dy + dz * dz;
            double *blk = G + (3 * t) * ld + 3 * s;
            if (dr2 == 0.0) {
                for (int a = 0; a < 3; ++a)
                    for (int b = 0; b < 3; ++b)
                        blk[a * ld + b] = 0.0;
                continue;
            }
            const double dr = sqrt(dr2);
            double fr, gr;
            if (dr > eps) {
                fr = factor / dr;
                gr = factor / (dr * dr * dr);
            } else {
                const double di = 1.0 / sqrt(dr2 + reg2);
                fr = factor * di;
                gr = factor * di * di * di;
            }
            const double d[3] = {dx, dy, dz};
            for (int a = 0; a < 3; ++a)
                for (int b = 0; b < 3; ++b)
                    blk[a * ld + b] = (a == b ? fr : 0.0) + gr * d[a] * d[b];
        }
}

int oracle_num_threads(void) {
#ifdef _OPENMP
    int n = 0;
#pragma omp parallel
    {
#pragma omp single
        n = omp_get_num_threads();
    }
    return n;
#else
    return 1;
#endif
}
