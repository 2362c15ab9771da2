/* skelly_hip.cpp — C-ABI host layer over the CDNA4 pair kernels.
 *
 * Implements include/skelly_hip.h: the reference drop-in entry points
 * (SkellySim include/kernels.hpp:17-20) plus extended host/device forms.
 * Replaces the reference's per-call cudaMalloc/copy/free round trip
 * (src/core/kernels.cu:149-178) with persistent grow-only device buffers and
 * a dedicated HIP stream.
 */

#include "skelly_hip.h"

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstring>
#include <vector>
#include <mutex>
#include <string>

namespace skelly {
hipError_t launch_stokeslet(const double *, const double *, const double *, double *, long long,
                            long long, double, hipStream_t);
hipError_t launch_stresslet(const double *, const double *, const double *, double *, long long,
                            long long, double, hipStream_t);
hipError_t launch_oseen(const double *, const double *, const double *, double *, long long,
                        long long, double, double, double, hipStream_t);
hipError_t launch_rotlet(const double *, const double *, const double *, double *, long long,
                         long long, double, double, double, hipStream_t);
hipError_t launch_stresslet_normal_density(const double *, const double *, const double *,
                                           double *, long long, long long, double, double,
                                           hipStream_t);
hipError_t launch_oseen_tensor_batched(const double *, double *, long long, long long, double,
                                       double, double, hipStream_t);
hipError_t launch_stresslet_times_normal(const double *, const double *, double *, long long,
                                         double, double, hipStream_t);
hipError_t run_fp64_peak(double *);
} // namespace skelly

namespace {

constexpr double kOneOver8Pi = 1.0 / 8.0 / M_PI; /* kernels.cu:59 */

thread_local std::string g_last_error;

void set_error(const char *where, hipError_t err) {
    g_last_error = std::string(where) + ": " + hipGetErrorString(err);
    std::fprintf(stderr, "[skelly-hip] %s\n", g_last_error.c_str());
}

/* Persistent per-process device state. Calls are serialized (the reference is
 * MPI_THREAD_FUNNELED, skelly_sim.cpp:14). */
struct Context {
    std::mutex mu;
    int device = 0;
    bool stream_ready = false;
    hipStream_t stream = nullptr;

    enum Role { R_SRC = 0, F_SRC, R_TRG, U_TRG, N_ROLES };
    void *buf[N_ROLES] = {};
    size_t cap[N_ROLES] = {};

    hipError_t ensure_stream() {
        if (!stream_ready) {
            hipError_t err = hipSetDevice(device);
            if (err != hipSuccess)
                return err;
            err = hipStreamCreate(&stream);
            if (err != hipSuccess)
                return err;
            stream_ready = true;
        }
        return hipSuccess;
    }

    /* grow-only buffer per role (the caching the reference's GPUEvaluator
     * declared but never implemented; kernels.hpp:176-178) */
    hipError_t ensure(Role r, size_t bytes, void **out) {
        if (cap[r] < bytes) {
            if (buf[r]) {
                (void)hipFree(buf[r]);
                buf[r] = nullptr;
                cap[r] = 0;
            }
            hipError_t err = hipMalloc(&buf[r], bytes);
            if (err != hipSuccess)
                return err;
            cap[r] = bytes;
        }
        *out = buf[r];
        return hipSuccess;
    }

    void release() {
        for (auto &b : buf) {
            if (b)
                (void)hipFree(b);
            b = nullptr;
        }
        for (auto &c : cap)
            c = 0;
        if (stream_ready) {
            (void)hipStreamDestroy(stream);
            stream_ready = false;
        }
    }
};

Context &ctx() {
    static Context c;
    return c;
}

#define CHK(where, call)                                                                           \
    do {                                                                                           \
        hipError_t _e = (call);                                                                    \
        if (_e != hipSuccess) {                                                                    \
            set_error(where, _e);                                                                  \
            return -1;                                                                             \
        }                                                                                          \
    } while (0)

/* Shared host-pointer round trip: upload, launch via `fn`, download, sync. */
template <typename LaunchFn>
int host_eval(const char *where, const double *r_src, const double *f_src, int srcdim,
              long long n_src, const double *r_trg, double *u_trg, long long n_trg,
              LaunchFn &&fn) {
    if (n_src < 0 || n_trg < 0) {
        g_last_error = std::string(where) + ": negative size";
        return -1;
    }
    if (n_trg == 0)
        return 0;
    Context &c = ctx();
    std::lock_guard<std::mutex> lock(c.mu);
    CHK(where, c.ensure_stream());
    double *d_rs = nullptr, *d_fs = nullptr, *d_rt = nullptr, *d_u = nullptr;
    /* hipMalloc(0) quirks avoided: allocate at least 8 bytes */
    CHK(where, c.ensure(Context::R_SRC, (size_t)(n_src ? n_src : 1) * 3 * 8, (void **)&d_rs));
    CHK(where, c.ensure(Context::F_SRC, (size_t)(n_src ? n_src : 1) * srcdim * 8, (void **)&d_fs));
    CHK(where, c.ensure(Context::R_TRG, (size_t)n_trg * 3 * 8, (void **)&d_rt));
    CHK(where, c.ensure(Context::U_TRG, (size_t)n_trg * 3 * 8, (void **)&d_u));
    if (n_src) {
        CHK(where, hipMemcpyAsync(d_rs, r_src, (size_t)n_src * 3 * 8, hipMemcpyHostToDevice,
                                  c.stream));
        CHK(where, hipMemcpyAsync(d_fs, f_src, (size_t)n_src * srcdim * 8, hipMemcpyHostToDevice,
                                  c.stream));
    }
    CHK(where, hipMemcpyAsync(d_rt, r_trg, (size_t)n_trg * 3 * 8, hipMemcpyHostToDevice, c.stream));
    CHK(where, fn(d_rs, d_fs, d_rt, d_u, c.stream));
    CHK(where, hipMemcpyAsync(u_trg, d_u, (size_t)n_trg * 3 * 8, hipMemcpyDeviceToHost, c.stream));
    CHK(where, hipStreamSynchronize(c.stream));
    return 0;
}

} // namespace

extern "C" {

const char *skelly_hip_version(void) { return "skelly-hip 0.1.0 gfx950"; }

const char *skelly_hip_last_error(void) { return g_last_error.c_str(); }

int skelly_hip_device_count(void) {
    int n = 0;
    hipError_t err = hipGetDeviceCount(&n);
    if (err != hipSuccess) {
        set_error("device_count", err);
        return -1;
    }
    return n;
}

int skelly_hip_set_device(int device) {
    Context &c = ctx();
    std::lock_guard<std::mutex> lock(c.mu);
    if (device != c.device) {
        c.release();
        c.device = device;
    }
    CHK("set_device", hipSetDevice(device));
    return 0;
}

int skelly_hip_shutdown(void) {
    Context &c = ctx();
    std::lock_guard<std::mutex> lock(c.mu);
    c.release();
    return 0;
}

/* ---- reference drop-in entry points (scale 1/(8*pi), no eta) ---- */

void stokeslet_direct_gpu_impl(const double *r_src, const double *f_src, int n_src,
                               const double *r_trg, double *u_trg, int n_trg) {
    host_eval("stokeslet_direct_gpu_impl", r_src, f_src, 3, n_src, r_trg, u_trg, n_trg,
              [&](const double *rs, const double *fs, const double *rt, double *u,
                  hipStream_t s) {
                  return skelly::launch_stokeslet(rs, fs, rt, u, n_src, n_trg, kOneOver8Pi, s);
              });
}

void stresslet_direct_gpu_impl(const double *r_src, const double *f_src, int n_src,
                               const double *r_trg, double *u_trg, int n_trg) {
    host_eval("stresslet_direct_gpu_impl", r_src, f_src, 9, n_src, r_trg, u_trg, n_trg,
              [&](const double *rs, const double *fs, const double *rt, double *u,
                  hipStream_t s) {
                  return skelly::launch_stresslet(rs, fs, rt, u, n_src, n_trg, kOneOver8Pi, s);
              });
}

/* ---- extended host-pointer API (fully scaled) ---- */

int skelly_stokeslet_host(const double *r_src, const double *f_src, long long n_src,
                          const double *r_trg, double *u_trg, long long n_trg, double eta) {
    return host_eval("skelly_stokeslet_host", r_src, f_src, 3, n_src, r_trg, u_trg, n_trg,
                     [&](const double *rs, const double *fs, const double *rt, double *u,
                         hipStream_t s) {
                         return skelly::launch_stokeslet(rs, fs, rt, u, n_src, n_trg,
                                                         kOneOver8Pi / eta, s);
                     });
}

int skelly_stresslet_host(const double *r_src, const double *f_src, long long n_src,
                          const double *r_trg, double *u_trg, long long n_trg, double eta) {
    return host_eval("skelly_stresslet_host", r_src, f_src, 9, n_src, r_trg, u_trg, n_trg,
                     [&](const double *rs, const double *fs, const double *rt, double *u,
                         hipStream_t s) {
                         return skelly::launch_stresslet(rs, fs, rt, u, n_src, n_trg,
                                                         kOneOver8Pi / eta, s);
                     });
}

int skelly_oseen_contract_host(const double *r_src, const double *r_trg, const double *density,
                               double *u_trg, long long n_src, long long n_trg, double eta,
                               double reg, double epsilon_distance) {
    const double factor = 1.0 / (8.0 * M_PI * eta); /* kernels.cpp:94 */
    return host_eval("skelly_oseen_contract_host", r_src, density, 3, n_src, r_trg, u_trg, n_trg,
                     [&](const double *rs, const double *ds, const double *rt, double *u,
                         hipStream_t s) {
                         return skelly::launch_oseen(rs, ds, rt, u, n_src, n_trg, factor, reg,
                                                     epsilon_distance, s);
                     });
}

int skelly_rotlet_host(const double *r_src, const double *r_trg, const double *density,
                       double *u_trg, long long n_src, long long n_trg, double eta, double reg,
                       double epsilon_distance) {
    const double factor = 1.0 / (8.0 * M_PI * eta); /* kernels.cpp:213 */
    return host_eval("skelly_rotlet_host", r_src, density, 3, n_src, r_trg, u_trg, n_trg,
                     [&](const double *rs, const double *ds, const double *rt, double *u,
                         hipStream_t s) {
                         return skelly::launch_rotlet(rs, ds, rt, u, n_src, n_trg, factor, reg,
                                                      epsilon_distance, s);
                     });
}

/* ---- device-pointer API (async on caller's stream) ---- */

int skelly_stokeslet_device(const double *d_r_src, const double *d_f_src, long long n_src,
                            const double *d_r_trg, double *d_u_trg, long long n_trg, double eta,
                            void *stream) {
    CHK("skelly_stokeslet_device",
        skelly::launch_stokeslet(d_r_src, d_f_src, d_r_trg, d_u_trg, n_src, n_trg,
                                 kOneOver8Pi / eta, (hipStream_t)stream));
    return 0;
}

int skelly_stresslet_device(const double *d_r_src, const double *d_f_src, long long n_src,
                            const double *d_r_trg, double *d_u_trg, long long n_trg, double eta,
                            void *stream) {
    CHK("skelly_stresslet_device",
        skelly::launch_stresslet(d_r_src, d_f_src, d_r_trg, d_u_trg, n_src, n_trg,
                                 kOneOver8Pi / eta, (hipStream_t)stream));
    return 0;
}

int skelly_oseen_contract_device(const double *d_r_src, const double *d_r_trg,
                                 const double *d_density, double *d_u_trg, long long n_src,
                                 long long n_trg, double eta, double reg, double epsilon_distance,
                                 void *stream) {
    const double factor = 1.0 / (8.0 * M_PI * eta);
    CHK("skelly_oseen_contract_device",
        skelly::launch_oseen(d_r_src, d_density, d_r_trg, d_u_trg, n_src, n_trg, factor, reg,
                             epsilon_distance, (hipStream_t)stream));
    return 0;
}

int skelly_rotlet_device(const double *d_r_src, const double *d_r_trg, const double *d_density,
                         double *d_u_trg, long long n_src, long long n_trg, double eta, double reg,
                         double epsilon_distance, void *stream) {
    const double factor = 1.0 / (8.0 * M_PI * eta);
    CHK("skelly_rotlet_device",
        skelly::launch_rotlet(d_r_src, d_density, d_r_trg, d_u_trg, n_src, n_trg, factor, reg,
                              epsilon_distance, (hipStream_t)stream));
    return 0;
}

int skelly_stresslet_normal_density_host(const double *r_src, const double *normals,
                                         const double *density, double *out, long long n,
                                         double reg, double epsilon_distance) {
    /* interleave [normal | density] into the (n, 6) source-strength array */
    std::vector<double> nd((size_t)n * 6);
    for (long long i = 0; i < n; ++i) {
        for (int k = 0; k < 3; ++k) {
            nd[6 * i + k] = normals[3 * i + k];
            nd[6 * i + 3 + k] = density[3 * i + k];
        }
    }
    return host_eval("skelly_stresslet_normal_density_host", r_src, nd.data(), 6, n, r_src, out,
                     n,
                     [&](const double *rs, const double *fs, const double *rt, double *u,
                         hipStream_t s) {
                         return skelly::launch_stresslet_normal_density(
                             rs, fs, rt, u, n, n, reg, epsilon_distance, s);
                     });
}

int skelly_stresslet_normal_density_device(const double *d_r_src, const double *d_nd,
                                           const double *d_r_trg, double *d_out, long long n_src,
                                           long long n_trg, double reg, double epsilon_distance,
                                           void *stream) {
    CHK("skelly_stresslet_normal_density_device",
        skelly::launch_stresslet_normal_density(d_r_src, d_nd, d_r_trg, d_out, n_src, n_trg, reg,
                                                epsilon_distance, (hipStream_t)stream));
    return 0;
}

int skelly_stresslet_times_normal_device(const double *d_pts, const double *d_normals,
                                         double *d_out, long long n, double reg,
                                         double epsilon_distance, void *stream) {
    CHK("skelly_stresslet_times_normal_device",
        skelly::launch_stresslet_times_normal(d_pts, d_normals, d_out, n, reg,
                                              epsilon_distance, (hipStream_t)stream));
    return 0;
}

int skelly_oseen_tensor_batched_device(const double *d_pts, double *d_G, long long nf,
                                       long long n, double eta, double reg,
                                       double epsilon_distance, void *stream) {
    CHK("skelly_oseen_tensor_batched_device",
        skelly::launch_oseen_tensor_batched(d_pts, d_G, nf, n, eta, reg, epsilon_distance,
                                            (hipStream_t)stream));
    return 0;
}

int skelly_fp64_peak_tflops(double *out_tflops) {
    Context &c = ctx();
    std::lock_guard<std::mutex> lock(c.mu);
    CHK("fp64_peak", c.ensure_stream());
    CHK("fp64_peak", skelly::run_fp64_peak(out_tflops));
    return 0;
}

} /* extern "C" */
