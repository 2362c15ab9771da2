/* skelly_hip.h — C-ABI boundary of the MI355X-native SkellySim hot-path engine.
 *
 * This is the drop-in seam for SkellySim's pair-kernel evaluators
 * (flatironinstitute/SkellySim, read-only reference at /root/reference):
 * the reference selects an Evaluator backend by string ("CPU"/"GPU"/"FMM") at
 * src/core/fiber_container_base.cpp:20-33, src/core/body_container.cpp:552-573
 * and src/core/periphery.cpp:337-352; the "GPU" branch bottoms out in the two
 * C-linkage-style entry points declared at include/kernels.hpp:17-20. This
 * library exports those two entry points (same names, same argument meaning)
 * plus extended device-pointer forms for multi-GPU sharding. See
 * INTEGRATION.md for the reference-side binding a maintainer would add.
 *
 * Layout contract (include/kernels.hpp:14-15): all matrices are fp64,
 * col-major 3 x n — i.e. per-point xyz contiguous. Stresslet strengths are
 * 9 x n (the 9-component double-layer tensor, kernels.cu:41-43).
 *
 * Scaling contract: the *_direct_gpu_impl results INCLUDE the 1/(8*pi) kernel
 * prefactor (applied per target, kernels.cu:59,122) and EXCLUDE the 1/eta
 * division (the reference host wrapper divides, kernels.cpp:358,365). The
 * skelly_*_device / skelly_*_host forms take eta and return fully scaled
 * velocities (1/(8*pi*eta)).
 *
 * Error contract: the reference exits the process on CUDA failure
 * (kernels.cu:9-15); this library instead returns nonzero error codes
 * (int-returning functions) and records a message retrievable via
 * skelly_hip_last_error(). The void reference-signature wrappers record the
 * error and leave u_trg untouched.
 *
 * Threading: calls are serialized by an internal mutex per the reference's
 * MPI_THREAD_FUNNELED usage (src/skelly_sim.cpp:14) — re-entrant per
 * evaluation, not concurrently invoked.
 *
 * Determinism: accumulation order per target is fixed (source-tile order), so
 * results are bit-reproducible for a given device count and shard layout.
 */

#ifndef SKELLY_HIP_H
#define SKELLY_HIP_H

#ifdef __cplusplus
extern "C" {
#endif

/* ---- library / device management ---- */

/* Force the persistent (grow-only hipMalloc) split-K workspace on/off at
 * runtime, overriding the SKELLY_PERSISTENT_WS env var (on=1/0; -1 not
 * accepted — call once). Required before hipGraph capture of launches
 * that take the split path: the async-mempool workspace cannot be
 * recorded, while the persistent one allocates only during uncaptured
 * warmup. */
void skelly_set_persistent_ws(int on);

/* Human-readable build id ("skelly-hip <ver> gfx950"). */
const char *skelly_hip_version(void);

/* Last error message for this thread; empty string if none. */
const char *skelly_hip_last_error(void);

/* Number of visible HIP devices; <0 on error. */
int skelly_hip_device_count(void);

/* Select the device used by subsequent calls (default 0). 0 on success. */
int skelly_hip_set_device(int device);

/* Release all persistent device buffers and streams. 0 on success. */
int skelly_hip_shutdown(void);

/* ---- reference drop-in entry points (include/kernels.hpp:17-20) ----
 *
 * Host pointers in/out; this library owns all device management. Unlike the
 * reference CUDA path (cudaMalloc/copy/free on every call, kernels.cu:149-178)
 * device buffers are persistent and grow-only across calls.
 * f_src is 3 doubles/point for the stokeslet, 9 for the stresslet.
 * u_trg (3 x n_trg) is overwritten (not accumulated), scaled by 1/(8*pi),
 * NOT divided by eta. On failure u_trg is untouched and
 * skelly_hip_last_error() is set. */
void stokeslet_direct_gpu_impl(const double *r_src, const double *f_src, int n_src,
                               const double *r_trg, double *u_trg, int n_trg);
void stresslet_direct_gpu_impl(const double *r_src, const double *f_src, int n_src,
                               const double *r_trg, double *u_trg, int n_trg);

/* ---- extended host-pointer API (fully scaled, returns error codes) ---- */

int skelly_stokeslet_host(const double *r_src, const double *f_src, long long n_src,
                          const double *r_trg, double *u_trg, long long n_trg, double eta);
int skelly_stresslet_host(const double *r_src, const double *f_src, long long n_src,
                          const double *r_trg, double *u_trg, long long n_trg, double eta);
/* Regularized Oseen contraction (kernels.cpp:85-131; defaults reg=5e-3,
 * epsilon_distance=1e-5 per kernels.hpp:34-35). */
int skelly_oseen_contract_host(const double *r_src, const double *r_trg, const double *density,
                               double *u_trg, long long n_src, long long n_trg,
                               double eta, double reg, double epsilon_distance);
/* Rotlet (kernels.cpp:206-242). */
int skelly_rotlet_host(const double *r_src, const double *r_trg, const double *density,
                       double *u_trg, long long n_src, long long n_trg,
                       double eta, double reg, double epsilon_distance);

/* ---- device-pointer API (for torch/RCCL plumbing; async on `stream`) ----
 *
 * All pointers are DEVICE pointers on the current device; `stream` is a
 * hipStream_t (pass e.g. torch.cuda.current_stream().cuda_stream), or NULL
 * for the null stream. No synchronization is performed. Results are fully
 * scaled (1/(8*pi*eta); oseen/rotlet regularization per the reference). */
int skelly_stokeslet_device(const double *d_r_src, const double *d_f_src, long long n_src,
                            const double *d_r_trg, double *d_u_trg, long long n_trg,
                            double eta, void *stream);
int skelly_stresslet_device(const double *d_r_src, const double *d_f_src, long long n_src,
                            const double *d_r_trg, double *d_u_trg, long long n_trg,
                            double eta, void *stream);
int skelly_oseen_contract_device(const double *d_r_src, const double *d_r_trg,
                                 const double *d_density, double *d_u_trg,
                                 long long n_src, long long n_trg,
                                 double eta, double reg, double epsilon_distance, void *stream);
int skelly_rotlet_device(const double *d_r_src, const double *d_r_trg,
                         const double *d_density, double *d_u_trg,
                         long long n_src, long long n_trg,
                         double eta, double reg, double epsilon_distance, void *stream);

/* Contraction of the stresslet with a normal and a density field over one
 * point set (kernels::stresslet_times_normal_times_density,
 * src/core/kernels.cpp:307-334; no eta dependence, factor -3/(4*pi)):
 * out_i = factor * sum_{j != i} (d.rho_j)(d.n_j)/|d|^5 d, d = r_i - r_j,
 * |d| < epsilon_distance regularized to sqrt(|d|^2 + reg^2).
 * Used by the body/fiber operator assembly. Defaults reg=5e-3, eps=1e-5
 * (include/kernels.hpp:50-51). */
int skelly_stresslet_normal_density_host(const double *r_src, const double *normals,
                                         const double *density, double *out, long long n,
                                         double reg, double epsilon_distance);
/* device form: nd is the interleaved (n, 6) [normal | density] array */
int skelly_stresslet_normal_density_device(const double *d_r_src, const double *d_nd,
                                           const double *d_r_trg, double *d_out,
                                           long long n_src, long long n_trg,
                                           double reg, double epsilon_distance, void *stream);

/* Dense stresslet_times_normal builder (kernels::stresslet_times_normal,
 * src/core/kernels.cpp:264-287; body operator assembly): pts/normals (n, 3)
 * -> Snormal (3n, 3n) C row-major, diagonal blocks zero. Device pointers. */
int skelly_stresslet_times_normal_device(const double *d_pts, const double *d_normals,
                                         double *d_out, long long n, double reg,
                                         double epsilon_distance, void *stream);

/* Batched oseen_tensor_direct dense builder (kernels::oseen_tensor_direct,
 * src/core/kernels.cpp:146-195, square/self form — the per-fiber
 * self-stokeslet build, src/core/fiber_finite_difference.cpp:56):
 * pts (nf, n, 3) -> G (nf, 3n, 3n) row-major; coincident blocks zero. */
int skelly_oseen_tensor_batched_device(const double *d_pts, double *d_G, long long nf,
                                       long long n, double eta, double reg,
                                       double epsilon_distance, void *stream);

/* ---- measurement helpers ---- */

/* Measured fp64 FMA throughput (TFLOP/s) of the current device via a pure
 * register FMA chain — used to pin the roofline denominator. 0 on success. */
int skelly_fp64_peak_tflops(double *out_tflops);

#ifdef __cplusplus
}
#endif

#endif /* SKELLY_HIP_H */
