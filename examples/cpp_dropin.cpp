/* cpp_dropin.cpp — C++ harness exercising the drop-in boundary exactly the
 * way the reference's kernel parity test does (tests/core/kernel_test.cpp:
 * n_src=1229, n_trg=743, eta=1.3, random clouds), through the C++ evaluator
 * mirror (include/skelly_evaluator.hpp) over the C-ABI.
 *
 * Build: make -C examples            (links skellysim_amd/libskellyhip.so)
 * Run:   ./examples/cpp_dropin      (needs a GPU; run under gpurun)
 *
 * Prints the Frobenius error of GPU-vs-CPU-restatement and exits nonzero if
 * it exceeds the reference gate 5e-9 (kernel_test.cpp:92) — here against a
 * host restatement compiled into this file for independence from oracle/.
 */

#include "skelly_evaluator.hpp"

#include <cmath>
#include <cstdio>
#include <random>

using skelly::CMatrixRef;
using skelly::MatrixXd;

/* host Stokeslet restatement (kernels.cu:62-76 + scale + /eta) for the check */
static MatrixXd stokeslet_host_ref(const MatrixXd &r_src, const MatrixXd &f_src,
                                   const MatrixXd &r_trg, double eta) {
    const double scale = 1.0 / 8.0 / M_PI;
    MatrixXd u(3, r_trg.cols());
    for (long t = 0; t < r_trg.cols(); ++t) {
        double acc[3] = {0, 0, 0};
        for (long s = 0; s < r_src.cols(); ++s) {
            const double dx = r_trg(0, t) - r_src(0, s);
            const double dy = r_trg(1, t) - r_src(1, s);
            const double dz = r_trg(2, t) - r_src(2, s);
            const double r2 = dx * dx + dy * dy + dz * dz;
            const double rinv = r2 == 0.0 ? 0.0 : 1.0 / std::sqrt(r2);
            const double inner =
                (f_src(0, s) * dx + f_src(1, s) * dy + f_src(2, s) * dz) * rinv * rinv;
            acc[0] += rinv * (f_src(0, s) + dx * inner);
            acc[1] += rinv * (f_src(1, s) + dy * inner);
            acc[2] += rinv * (f_src(2, s) + dz * inner);
        }
        for (int i = 0; i < 3; ++i)
            u(i, t) = acc[i] * scale / eta;
    }
    return u;
}

int main() {
    const int n_src = 1229, n_trg = 743; /* kernel_test.cpp:25-26 */
    const double eta = 1.3;              /* kernel_test.cpp:27 */

    std::mt19937_64 rng(100);
    std::uniform_real_distribution<double> dist(-1.0, 1.0);
    MatrixXd r_src(3, n_src), f_src(3, n_src), r_trg(3, n_trg);
    for (long i = 0; i < r_src.size(); ++i)
        r_src.data()[i] = dist(rng);
    for (long i = 0; i < f_src.size(); ++i)
        f_src.data()[i] = dist(rng);
    for (long i = 0; i < r_trg.size(); ++i)
        r_trg.data()[i] = dist(rng);

    MatrixXd nullmat;
    auto evaluator = skelly::make_stokeslet_evaluator("HIP");
    MatrixXd u = evaluator(r_src, nullmat, r_trg, f_src, nullmat, eta);
    MatrixXd ref = stokeslet_host_ref(r_src, f_src, r_trg, eta);

    double err2 = 0.0, norm2 = 0.0;
    for (long i = 0; i < u.size(); ++i) {
        const double d = u.data()[i] - ref.data()[i];
        err2 += d * d;
        norm2 += ref.data()[i] * ref.data()[i];
    }
    const double frob = std::sqrt(err2), rel = frob / std::sqrt(norm2);
    std::printf("cpp_dropin: frobenius err=%.3e (gate 5e-9), rel=%.3e\n", frob, rel);
    return frob > 5e-9; /* kernel_test.cpp:92 */
}
