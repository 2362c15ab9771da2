/* skelly_evaluator.hpp — C++ host-side mirror of SkellySim's evaluator API.
 *
 * Header-only. Mirrors the kernels:: evaluator functions the reference host
 * code calls (flatironinstitute/SkellySim include/kernels.hpp:22-51) on top
 * of the C-ABI in skelly_hip.h, so a C++ harness (or the reference itself,
 * see INTEGRATION.md) can call the MI355X engine with the same call shapes.
 *
 * The reference types these signatures with Eigen::MatrixXd / CMatrixRef
 * (col-major fp64, 3 x n, per-point xyz contiguous). Eigen is not vendored
 * here, so the mirror uses a minimal col-major view/owning matrix with the
 * identical memory layout; an Eigen-typed caller binds by pointer+shape
 * (see INTEGRATION.md for the two-line adapters).
 *
 * Semantics mirrored exactly:
 *  - stokeslet consumers pass sources in (r_sl, f_sl) with r_dl/f_dl empty,
 *    stresslet consumers the reverse (kernel_test.cpp:40,65);
 *  - results are 3 x n_trg, already divided by eta (kernels.cpp:358,365);
 *  - oseen/rotlet defaults reg=5e-3, epsilon_distance=1e-5
 *    (kernels.hpp:34-35,44-45).
 */

#ifndef SKELLY_EVALUATOR_HPP
#define SKELLY_EVALUATOR_HPP

#include "skelly_hip.h"

#include <functional>
#include <stdexcept>
#include <string>
#include <vector>

namespace skelly {

/* Minimal col-major fp64 matrix (rows x cols), layout-compatible with the
 * reference's Eigen::MatrixXd. */
class MatrixXd {
  public:
    MatrixXd() = default;
    MatrixXd(long rows, long cols) : rows_(rows), cols_(cols), data_(rows * cols, 0.0) {}
    static MatrixXd Zero(long rows, long cols) { return MatrixXd(rows, cols); }

    long rows() const { return rows_; }
    long cols() const { return cols_; }
    long size() const { return rows_ * cols_; }
    double *data() { return data_.data(); }
    const double *data() const { return data_.data(); }
    double &operator()(long i, long j) { return data_[j * rows_ + i]; }
    double operator()(long i, long j) const { return data_[j * rows_ + i]; }

  private:
    long rows_ = 0, cols_ = 0;
    std::vector<double> data_;
};

/* Read-only view (what the reference calls CMatrixRef). Binds a MatrixXd or
 * any col-major buffer (e.g. an Eigen matrix's data()). */
struct CMatrixRef {
    const double *ptr = nullptr;
    long rows = 0, cols = 0;
    CMatrixRef() = default;
    CMatrixRef(const MatrixXd &m) : ptr(m.data()), rows(m.rows()), cols(m.cols()) {}
    CMatrixRef(const double *p, long r, long c) : ptr(p), rows(r), cols(c) {}
    long size() const { return rows * cols; }
};

/* kernels::Evaluator (reference include/kernels.hpp:14-15). */
using Evaluator = std::function<MatrixXd(const CMatrixRef &r_sl, const CMatrixRef &r_dl,
                                         const CMatrixRef &r_trg, const CMatrixRef &f_sl,
                                         const CMatrixRef &f_dl, double eta)>;

inline void check_rc(int rc, const char *what) {
    if (rc != 0)
        throw std::runtime_error(std::string(what) + ": " + skelly_hip_last_error());
}

/* Mirror of kernels::stokeslet_direct_gpu (kernels.cpp:361-366). */
inline MatrixXd stokeslet_direct_gpu(const CMatrixRef &r_sl, const CMatrixRef & /*r_dl*/,
                                     const CMatrixRef &r_trg, const CMatrixRef &f_sl,
                                     const CMatrixRef & /*f_dl*/, double eta) {
    MatrixXd u(3, r_trg.cols);
    check_rc(skelly_stokeslet_host(r_sl.ptr, f_sl.ptr, r_sl.cols, r_trg.ptr, u.data(),
                                   r_trg.cols, eta),
             "stokeslet_direct_gpu");
    return u;
}

/* Mirror of kernels::stresslet_direct_gpu (kernels.cpp:354-359). */
inline MatrixXd stresslet_direct_gpu(const CMatrixRef & /*r_sl*/, const CMatrixRef &r_dl,
                                     const CMatrixRef &r_trg, const CMatrixRef & /*f_sl*/,
                                     const CMatrixRef &f_dl, double eta) {
    MatrixXd u(3, r_trg.cols);
    check_rc(skelly_stresslet_host(r_dl.ptr, f_dl.ptr, r_dl.cols, r_trg.ptr, u.data(),
                                   r_trg.cols, eta),
             "stresslet_direct_gpu");
    return u;
}

/* Mirror of kernels::oseen_tensor_contract_direct (kernels.cpp:85-131). */
inline MatrixXd oseen_tensor_contract_direct(const CMatrixRef &r_src, const CMatrixRef &r_trg,
                                             const CMatrixRef &density, double eta,
                                             double reg = 5e-3,
                                             double epsilon_distance = 1e-5) {
    MatrixXd u(3, r_trg.size() / 3);
    check_rc(skelly_oseen_contract_host(r_src.ptr, r_trg.ptr, density.ptr, u.data(),
                                        r_src.size() / 3, r_trg.size() / 3, eta, reg,
                                        epsilon_distance),
             "oseen_tensor_contract_direct");
    return u;
}

/* Mirror of kernels::rotlet (kernels.cpp:206-242). */
inline MatrixXd rotlet(const CMatrixRef &r_src, const CMatrixRef &r_trg,
                       const CMatrixRef &density, double eta, double reg = 5e-3,
                       double epsilon_distance = 1e-5) {
    MatrixXd u(3, r_trg.size() / 3);
    check_rc(skelly_rotlet_host(r_src.ptr, r_trg.ptr, density.ptr, u.data(), r_src.size() / 3,
                                r_trg.size() / 3, eta, reg, epsilon_distance),
             "rotlet");
    return u;
}

/* Mirror of the reference's string-keyed backend selection
 * (fiber_container_base.cpp:20-33, periphery.cpp:337-352): "HIP" is this
 * engine; the reference's "CPU"/"FMM" backends are out of scope. */
inline Evaluator make_stokeslet_evaluator(const std::string &name) {
    if (name == "HIP" || name == "GPU")
        return stokeslet_direct_gpu;
    throw std::runtime_error("evaluator \"" + name + "\": only \"HIP\" is provided");
}

inline Evaluator make_stresslet_evaluator(const std::string &name) {
    if (name == "HIP" || name == "GPU")
        return stresslet_direct_gpu;
    throw std::runtime_error("evaluator \"" + name + "\": only \"HIP\" is provided");
}

} // namespace skelly

#endif /* SKELLY_EVALUATOR_HPP */
