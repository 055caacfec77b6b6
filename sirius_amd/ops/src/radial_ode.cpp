// Radial ODE integrator for the FP-LAPW branch (CPU, fp64).
//
// Native implementation of the reference Radial_solver behavior
// (src/radial/radial_solver.hpp:326-1290): outward integration of the
// coupled first-order radial system for the Schroedinger /
// Koelling-Harmon / ZORA / IORA / Dirac equations on an exponential
// muffin-tin grid, with
//   - fixed-energy solutions + energy derivatives (dme) via
//     inhomogeneous chi sources               (radial_solver.hpp:786)
//   - bound-state search by node-count bisection (Bound_state,
//     radial_solver.hpp:966)
//   - linearization-energy finder (Enu_finder, radial_solver.hpp:1159)
//
// The integrator is plain RK4 with fixed substeps per grid interval
// (the reference uses adaptive RK8 at eps=1e-3; the basis functions
// only need to span the right space — H/O matrix elements are computed
// exactly from the stored functions, so the variational result is
// insensitive to small integration differences).

#include <torch/extension.h>

#include <cmath>
#include <vector>
#include <array>
#include <stdexcept>

namespace {

constexpr double SPEED_OF_LIGHT = 137.035999139;  // reference constants.hpp:22
constexpr double ALPHA = 1.0 / SPEED_OF_LIGHT;
constexpr double REST_ENERGY = SPEED_OF_LIGHT * SPEED_OF_LIGHT;
constexpr double SQ_ALPHA_HALF = 0.5 / REST_ENERGY;

enum class Rel { none = 0, kh = 1, zora = 2, iora = 3, dirac = 4 };

// ---------------------------------------------------------------- spline
// Cubic spline with NOT-A-KNOT boundary conditions on a non-uniform
// grid (the reference Spline uses not-a-knot, spline.hpp:21; scipy's
// CubicSpline default, used on the Python side, is also not-a-knot —
// keeping all interpolations consistent matters at the 1e-5 Ha level
// for deep-core integrals).
struct Spline {
    std::vector<double> x, a, b, c, d;  // f(t) = a + b*dt + c*dt^2 + d*dt^3

    Spline() = default;

    Spline(const std::vector<double>& xs, const std::vector<double>& ys) { build(xs, ys); }

    void build(const std::vector<double>& xs, const std::vector<double>& ys) {
        int n = (int)xs.size();
        x = xs;
        a = ys;
        b.assign(n, 0.0);
        c.assign(n, 0.0);
        d.assign(n, 0.0);
        if (n < 4) return;
        std::vector<double> h(n - 1);
        for (int i = 0; i < n - 1; i++) h[i] = xs[i + 1] - xs[i];
        // solve for interior c_1..c_{n-2}; c_0, c_{n-1} eliminated via
        // not-a-knot: d3 continuity at x_1 and x_{n-2}:
        //   c0 = c1 + h0*(c1 - c2)/h1
        //   c_{n-1} = c_{n-2} + h_{n-2}*(c_{n-2} - c_{n-3})/h_{n-3}
        int m = n - 2;
        std::vector<double> dl(m, 0.0), dm(m, 0.0), du(m, 0.0), rhs(m, 0.0);
        for (int i = 1; i <= n - 2; i++) {
            int j = i - 1;
            dl[j] = h[i - 1];
            dm[j] = 2.0 * (h[i - 1] + h[i]);
            du[j] = h[i];
            rhs[j] = 3.0 * ((ys[i + 1] - ys[i]) / h[i] - (ys[i] - ys[i - 1]) / h[i - 1]);
        }
        // fold the eliminated boundary c's into the first/last rows
        // row j=0 couples c0: dm += h0*(1 + h0/h1), du += -h0*h0/h1
        dm[0] += h[0] * (1.0 + h[0] / h[1]);
        du[0] += -h[0] * h[0] / h[1];
        dm[m - 1] += h[n - 2] * (1.0 + h[n - 2] / h[n - 3]);
        dl[m - 1] += -h[n - 2] * h[n - 2] / h[n - 3];
        // Thomas solve
        for (int i = 1; i < m; i++) {
            double w = dl[i] / dm[i - 1];
            dm[i] -= w * du[i - 1];
            rhs[i] -= w * rhs[i - 1];
        }
        c[n - 2] = rhs[m - 1] / dm[m - 1];
        for (int i = m - 2; i >= 0; i--) {
            c[i + 1] = (rhs[i] - du[i] * c[i + 2]) / dm[i];
        }
        c[0] = c[1] + h[0] * (c[1] - c[2]) / h[1];
        c[n - 1] = c[n - 2] + h[n - 2] * (c[n - 2] - c[n - 3]) / h[n - 3];
        for (int j = 0; j < n - 1; j++) {
            b[j] = (ys[j + 1] - ys[j]) / h[j] - h[j] * (c[j + 1] + 2.0 * c[j]) / 3.0;
            d[j] = (c[j + 1] - c[j]) / (3.0 * h[j]);
        }
        // derivative coefficients at the last point (for deriv())
        b[n - 1] = b[n - 2] + 2.0 * c[n - 2] * h[n - 2] + 3.0 * d[n - 2] * h[n - 2] * h[n - 2];
    }

    // value inside interval i at offset dt
    inline double eval(int i, double dt) const {
        return a[i] + dt * (b[i] + dt * (c[i] + dt * d[i]));
    }

    inline double deriv(int i, double dt) const {
        return b[i] + dt * (2.0 * c[i] + dt * 3.0 * d[i]);
    }

    // exact integral of the spline times r^0 over the whole grid
    double integrate() const {
        double s = 0.0;
        for (size_t i = 0; i + 1 < x.size(); i++) {
            double h = x[i + 1] - x[i];
            s += h * (a[i] + h * (b[i] / 2.0 + h * (c[i] / 3.0 + h * d[i] / 4.0)));
        }
        return s;
    }
};

inline double spline_integrate(const std::vector<double>& xs, const std::vector<double>& ys) {
    Spline s(xs, ys);
    return s.integrate();
}

// relativistic mass
inline double rel_mass(Rel rel, double enu, double V) {
    switch (rel) {
        case Rel::none: return 1.0;
        case Rel::kh:   return 1.0 + SQ_ALPHA_HALF * (enu - V);
        case Rel::zora:
        case Rel::iora: return 1.0 - SQ_ALPHA_HALF * V;
        default:        return 1.0;
    }
}

struct SolveResult {
    std::vector<double> p, q, dpdr, dqdr;
    int nn{0};
};

// Integrate the coupled system forward (reference
// integrate_forward_gsl, radial_solver.hpp:344-660).  ve = spline of
// electronic potential (v + zn/x); chi_p/chi_q = inhomogeneous sources
// (mutable: scaled on renormalization).
class Integrator {
  public:
    Integrator(int zn, const std::vector<double>& r, const std::vector<double>& v)
        : zn_(zn), r_(r) {
        std::vector<double> ve(r.size());
        for (size_t i = 0; i < r.size(); i++) ve[i] = v[i] + zn / r[i];
        ve_.build(r, ve);
    }

    int num_points() const { return (int)r_.size(); }
    const std::vector<double>& grid() const { return r_; }

    SolveResult integrate_forward(Rel rel, double enu, int l, int kappa_in,
                                  std::vector<double>& chi_p, std::vector<double>& chi_q,
                                  bool bound_state, int nsub = 6) const {
        int nr = num_points();
        SolveResult res;
        res.p.assign(nr, 0.0);
        res.q.assign(nr, 0.0);
        res.dpdr.assign(nr, 0.0);
        res.dqdr.assign(nr, 0.0);

        Spline schi_p(r_, chi_p), schi_q(r_, chi_q);
        double kappa = 0.0;
        if (rel == Rel::dirac) {
            if (kappa_in == l) kappa = kappa_in;
            else if (kappa_in == l + 1) kappa = -kappa_in;
            else throw std::runtime_error("wrong k for l in Dirac solver");
        }
        double ll_half = l * (l + 1) / 2.0;

        auto run = [&](int last_point) -> std::vector<int> {
            std::vector<int> ridx;
            double x = r_[0];
            double y0, y1;
            switch (rel) {
                case Rel::none:
                    if (l == 0) { y0 = 2.0 * x * zn_; y1 = -double(zn_) * zn_ * x; }
                    else { y0 = std::pow(x, l + 1) / (2 * l + 1); y1 = std::pow(x, l) / 4.0; }
                    break;
                case Rel::kh:
                case Rel::zora:
                case Rel::iora: {
                    double aa = l * (l + 1) + 1 - std::pow(ALPHA * zn_, 2);
                    double bb = 0.5 * (1 + std::sqrt(1 + 4 * aa));
                    y0 = std::pow(x, bb);
                    y1 = (x * bb * std::pow(x, bb - 1) - y0) / zn_ / (ALPHA * ALPHA);
                    break;
                }
                case Rel::dirac: {
                    double bb = std::sqrt(kappa * kappa - std::pow(zn_ / SPEED_OF_LIGHT, 2));
                    y0 = std::pow(x, bb);
                    y1 = std::pow(x, bb - 1) * (bb + kappa) * x / ALPHA / zn_;
                    break;
                }
            }
            res.p[0] = y0;
            res.q[0] = y1;
            double chi_scale = 1.0;

            // RHS at (interval ir, offset dt)
            auto rhs = [&](int ir, double dt, double yp, double yq, double& fp, double& fq) {
                double xx = r_[ir] + dt;
                double V = ve_.eval(ir, dt) - zn_ / xx;
                double cp = schi_p.eval(ir, dt) * chi_scale;
                double cq = schi_q.eval(ir, dt) * chi_scale;
                if (rel == Rel::dirac) {
                    fp = ALPHA * (enu - V + 2 * REST_ENERGY) * yq - yp * kappa / xx;
                    fq = -ALPHA * (enu - V) * yp + yq * kappa / xx;
                } else {
                    double M = rel_mass(rel, enu, V);
                    fp = 2 * M * yq + yp / xx + cp;
                    fq = (V - enu + ll_half / (M * xx * xx)) * yp - yq / xx + cq;
                }
            };

            for (int ir = 0; ir < last_point; ir++) {
                double h = (r_[ir + 1] - r_[ir]) / nsub;
                double yp = y0, yq = y1;
                for (int s = 0; s < nsub; s++) {
                    double dt = s * h;
                    double k1p, k1q, k2p, k2q, k3p, k3q, k4p, k4q;
                    rhs(ir, dt, yp, yq, k1p, k1q);
                    rhs(ir, dt + 0.5 * h, yp + 0.5 * h * k1p, yq + 0.5 * h * k1q, k2p, k2q);
                    rhs(ir, dt + 0.5 * h, yp + 0.5 * h * k2p, yq + 0.5 * h * k2q, k3p, k3q);
                    rhs(ir, dt + h, yp + h * k3p, yq + h * k3q, k4p, k4q);
                    yp += h / 6.0 * (k1p + 2 * k2p + 2 * k3p + k4p);
                    yq += h / 6.0 * (k1q + 2 * k2q + 2 * k3q + k4q);
                }
                // node passed: reset renormalization record
                if (yp * res.p[ir] < 0) ridx.clear();
                res.p[ir + 1] = yp;
                res.q[ir + 1] = yq;
                y0 = yp; y1 = yq;
                const double max_val = 1e6;
                if (std::abs(yp) > max_val) {
                    ridx.push_back(ir + 1);
                    for (int j = 0; j <= ir + 1; j++) { res.p[j] /= max_val; res.q[j] /= max_val; }
                    y0 /= max_val; y1 /= max_val;
                    chi_scale /= max_val;
                }
            }
            // propagate the chi scaling back to the caller's source arrays
            if (chi_scale != 1.0) {
                for (auto& v : chi_p) v *= chi_scale;
                for (auto& v : chi_q) v *= chi_scale;
                schi_p.build(r_, chi_p);
                schi_q.build(r_, chi_q);
            }
            return ridx;
        };

        int last_point = nr - 1;
        if (!bound_state) {
            run(last_point);
        } else {
            auto ridx = run(last_point);
            if (!ridx.empty()) {
                last_point = ridx.front();
                run(last_point);
            }
            /* go backward from the last point to the outermost minimum or node */
            for (int j = last_point; j >= 1; j--) {
                if ((res.p[j] * res.p[j - 1] < 0) ||
                    ((std::abs(res.p[j]) < std::abs(res.p[j - 1])) && (res.p[j] * res.p[j - 1] > 0))) {
                    last_point = j;
                    break;
                }
            }
            run(last_point);
            for (int i = last_point + 1; i < nr; i++) { res.p[i] = 0.0; res.q[i] = 0.0; }
        }

        int nn = 0;
        for (int i = 0; i < last_point; i++)
            if (res.p[i] * res.p[i + 1] < 0.0) nn++;
        res.nn = nn;

        /* reconstruct derivatives from the ODE */
        for (int i = 0; i < nr; i++) {
            double xx = r_[i];
            double V = ve_.eval(i, 0.0) - zn_ / xx;
            if (rel == Rel::dirac) continue;  // not needed for Dirac (core only)
            double M = rel_mass(rel, enu, V);
            res.dpdr[i] = 2 * M * res.q[i] + res.p[i] / xx + chi_p[i];
            res.dqdr[i] = (V - enu + ll_half / (M * xx * xx)) * res.p[i] - res.q[i] / xx + chi_q[i];
        }

        /* normalize to \int p^2 dr = 1 (+ \int q^2 for Dirac) */
        std::vector<double> p2(nr);
        for (int i = 0; i < nr; i++) p2[i] = res.p[i] * res.p[i];
        double norm = spline_integrate(r_, p2);
        if (rel == Rel::dirac) {
            for (int i = 0; i < nr; i++) p2[i] = res.q[i] * res.q[i];
            norm += spline_integrate(r_, p2);
        }
        norm = 1.0 / std::sqrt(norm);
        for (int i = 0; i < nr; i++) {
            res.p[i] *= norm; res.q[i] *= norm;
            res.dpdr[i] *= norm; res.dqdr[i] *= norm;
        }
        return res;
    }

  private:
    int zn_;
    std::vector<double> r_;
    Spline ve_;
};

std::vector<double> tensor_to_vec(const torch::Tensor& t) {
    auto tc = t.to(torch::kFloat64).contiguous();
    const double* ptr = tc.data_ptr<double>();
    return std::vector<double>(ptr, ptr + tc.numel());
}

torch::Tensor vec_to_tensor(const std::vector<double>& v) {
    auto t = torch::empty({(long)v.size()}, torch::kFloat64);
    std::copy(v.begin(), v.end(), t.data_ptr<double>());
    return t;
}

// solve at fixed energy with dme energy derivatives
// (reference Radial_solver::solve, radial_solver.hpp:786-946)
std::vector<torch::Tensor> rs_solve(int64_t rel_i, int64_t dme, int64_t l, int64_t zn,
                                    double enu, torch::Tensor r_t, torch::Tensor v_t,
                                    int64_t k = 0) {
    Rel rel = (Rel)rel_i;
    auto r = tensor_to_vec(r_t);
    auto v = tensor_to_vec(v_t);
    int nr = (int)r.size();
    Integrator solver((int)zn, r, v);

    std::vector<std::vector<double>> p, q;
    std::vector<double> chi_p(nr, 0.0), chi_q(nr, 0.0);
    SolveResult last;
    double ll_half = l * (l + 1) / 2.0;

    for (int j = 0; j <= dme; j++) {
        if (j) {
            std::fill(chi_p.begin(), chi_p.end(), 0.0);
            std::fill(chi_q.begin(), chi_q.end(), 0.0);
            if (rel == Rel::none || rel == Rel::zora) {
                for (int i = 0; i < nr; i++) chi_q[i] = -j * p[j - 1][i];
            } else if (rel == Rel::kh) {
                for (int i = 0; i < nr; i++) chi_p[i] = j * 2 * SQ_ALPHA_HALF * q[j - 1][i];
                for (int i = 0; i < nr; i++) {
                    double x = r[i];
                    double V = v[i];
                    double M = rel_mass(Rel::kh, enu, V);
                    double c = SQ_ALPHA_HALF * ll_half / std::pow(x * M, 2);
                    if (j == 1) chi_q[i] = -p[0][i] * (1 + c);
                    else if (j == 2) chi_q[i] = -2 * p[1][i] * (1 + c) + 2 * p[0][i] * SQ_ALPHA_HALF * c / M;
                    else if (j == 3) chi_q[i] = -3 * p[2][i] * (1 + c) + 6 * p[1][i] * SQ_ALPHA_HALF * c / M -
                                                6 * p[0][i] * std::pow(SQ_ALPHA_HALF / M, 2) * c;
                    else throw std::runtime_error("dme > 3 not implemented (KH)");
                }
            } else if (rel == Rel::iora) {
                for (int i = 0; i < nr; i++) {
                    double x = r[i];
                    double V = v[i];
                    double M0 = rel_mass(Rel::zora, enu, V);
                    chi_q[i] = -j * p[j - 1][i] * (1 + SQ_ALPHA_HALF * ll_half / std::pow(M0 * x, 2));
                    double U = 1 - SQ_ALPHA_HALF * enu / M0;
                    if (j == 1) chi_p[i] = q[0][i] * 2 * SQ_ALPHA_HALF * std::pow(U, -2);
                    else if (j == 2) chi_p[i] = q[1][i] * 4 * SQ_ALPHA_HALF * std::pow(U, -2) +
                                                q[0][i] * 4 * std::pow(SQ_ALPHA_HALF, 2) * std::pow(U, -3) / M0;
                    else if (j == 3) chi_p[i] = q[2][i] * 6 * SQ_ALPHA_HALF * std::pow(U, -2) +
                                                q[1][i] * 12 * std::pow(SQ_ALPHA_HALF, 2) * std::pow(U, -2) / (M0 * U) +
                                                q[0][i] * 12 * std::pow(SQ_ALPHA_HALF, 3) * std::pow(M0 * U, -2) * std::pow(U, -2);
                    else throw std::runtime_error("dme > 3 not implemented (IORA)");
                }
            } else {
                throw std::runtime_error("dme not supported for Dirac");
            }
        }
        last = solver.integrate_forward(rel, enu, (int)l, (int)k, chi_p, chi_q, false);
        p.push_back(last.p);
        q.push_back(last.q);
    }

    std::vector<double> rdudr(nr);
    for (int i = 0; i < nr; i++) rdudr[i] = last.dpdr[i] - last.p[i] / r[i];

    double R = r.back();
    std::array<double, 3> uderiv;
    uderiv[0] = last.p.back() / R;
    uderiv[1] = (last.dpdr.back() - last.p.back() / R) / R;
    Spline sdpdr(r, last.dpdr);
    double d2p = sdpdr.deriv(nr - 2, r[nr - 1] - r[nr - 2]);
    uderiv[2] = (d2p - 2 * last.dpdr.back() / R + 2 * last.p.back() / (R * R)) / R;

    auto ud = torch::empty({3}, torch::kFloat64);
    ud[0] = uderiv[0]; ud[1] = uderiv[1]; ud[2] = uderiv[2];
    return {vec_to_tensor(last.p), vec_to_tensor(rdudr), ud,
            torch::tensor((int64_t)last.nn)};
}

// bound state: node-count search + bisection (reference Bound_state::solve)
std::vector<torch::Tensor> rs_bound_state(int64_t rel_i, int64_t zn, int64_t n, int64_t l,
                                          int64_t k, double enu_start, torch::Tensor r_t,
                                          torch::Tensor v_t, double alpha0, double alpha1) {
    Rel rel = (Rel)rel_i;
    auto r = tensor_to_vec(r_t);
    auto v = tensor_to_vec(v_t);
    int nr = (int)r.size();
    Integrator solver((int)zn, r, v);
    std::vector<double> chi_p(nr, 0.0), chi_q(nr, 0.0);
    const double enu_tolerance = 1e-12;
    int nodes_target = (int)(n - l - 1);

    auto integrate = [&](double E) {
        return solver.integrate_forward(rel, E, (int)l, (int)k, chi_p, chi_q, true);
    };

    /* 1st pass: bracket */
    double enu = enu_start;
    int s = 1, sp = 1;
    double denu = enu_tolerance;
    SolveResult res;
    for (int iter = 0; iter < 1000; iter++) {
        res = integrate(enu);
        sp = s;
        s = (res.nn > nodes_target) ? -1 : 1;
        if (s != sp && iter > 1) break;
        denu = (s != sp) ? denu * alpha0 : denu * alpha1;
        enu += s * denu;
        if (iter == 999) throw std::runtime_error("bound state: bracketing failed");
    }
    double e1 = enu, e2 = enu - sp * denu;
    if (e1 > e2) std::swap(e1, e2);

    /* 2nd pass: bisection */
    enu = (e1 + e2) / 2;
    for (int iter = 0; iter < 1000; iter++) {
        res = integrate(enu);
        if (res.nn > nodes_target) e2 = enu; else e1 = enu;
        enu = (e1 + e2) / 2.0;
        if (std::abs(e1 - e2) < enu_tolerance) break;
        if (iter == 999) throw std::runtime_error("bound state: bisection failed");
    }
    /* final: bottom of the refined interval */
    enu = e1;
    res = integrate(enu);

    int nn = 0;
    for (int i = 0; i < nr - 1; i++)
        if (res.p[i] * res.p[i + 1] < 0.0) nn++;
    if (nn != nodes_target) throw std::runtime_error("bound state: wrong number of nodes");

    /* charge density: u^2 (+ (q/r)^2 for Dirac) */
    std::vector<double> rho(nr, 0.0);
    for (int i = 0; i < nr; i++) {
        double u = res.p[i] / r[i];
        rho[i] = u * u;
        if (rel == Rel::dirac) rho[i] += std::pow(res.q[i] / r[i], 2);
    }
    return {torch::tensor(enu, torch::kFloat64), vec_to_tensor(res.p), vec_to_tensor(rho)};
}

// linearization energy finder (reference Enu_finder::find_enu)
double rs_enu_finder(int64_t rel_i, int64_t zn, int64_t n, int64_t l, double enu_start,
                     int64_t auto_enu, torch::Tensor r_t, torch::Tensor v_t) {
    Rel rel = (Rel)rel_i;
    auto r = tensor_to_vec(r_t);
    auto v = tensor_to_vec(v_t);
    int nr = (int)r.size();
    Integrator solver((int)zn, r, v);
    std::vector<double> chi_p(nr, 0.0), chi_q(nr, 0.0);
    int nodes_target = (int)(n - l - 1);

    SolveResult res;
    auto integrate = [&](double E) {
        res = solver.integrate_forward(rel, E, (int)l, 0, chi_p, chi_q, false);
        return res.nn;
    };

    /* top of the band: zero at MT boundary with n-l-1 nodes inside */
    double enu = enu_start;
    int s = 1, sp = 1;
    double denu = 1e-8;
    for (int iter = 0; iter < 1000; iter++) {
        int nn = integrate(enu);
        sp = s;
        s = (nn > nodes_target) ? -1 : 1;
        if (s != sp && iter > 0) break;
        denu *= 10;
        enu += s * denu;
        if (iter == 999) throw std::runtime_error("enu finder: top bracketing failed");
    }
    double e1 = enu, e2 = enu - sp * denu;
    if (e1 > e2) std::swap(e1, e2);
    double etop = (e1 + e2) / 2;
    for (int iter = 0; iter < 1000; iter++) {
        int nn = integrate(etop);
        if (nn > nodes_target) e2 = etop; else e1 = etop;
        etop = (e1 + e2) / 2.0;
        if (std::abs(e1 - e2) < 1e-9) break;
        if (iter == 999) throw std::runtime_error("enu finder: top bisection failed");
    }

    auto surface_deriv = [&]() { return res.dpdr.back(); };
    double sd = surface_deriv();

    /* bottom of the band: derivative zero at boundary */
    denu = 1e-8;
    double e0 = etop;
    for (int iter = 0; iter < 1000; iter++) {
        integrate(e0);
        if (surface_deriv() * sd <= 0 || denu > 20) break;
        denu *= 2;
        e0 -= denu;
        if (iter == 999) throw std::runtime_error("enu finder: bottom bracketing failed");
    }
    e1 = e0;
    e2 = e0 + denu;
    double ebot = (e1 + e2) / 2;
    for (int iter = 0; iter < 1000; iter++) {
        integrate(ebot);
        if (surface_deriv() * sd > 0) e2 = ebot; else e1 = ebot;
        ebot = (e1 + e2) / 2.0;
        if (std::abs(surface_deriv()) < 1e-8) break;
        if (iter == 999) break;  // reference tolerates loose bottom
    }

    switch (auto_enu) {
        case 1: return (ebot + etop) / 2.0;
        case 2: return ebot;
        default: throw std::runtime_error("wrong auto_enu type");
    }
}

}  // namespace

// debug: node count + tail of p in bound mode at fixed energy
std::vector<torch::Tensor> rs_nodes_bound(int64_t rel_i, int64_t zn, int64_t l, int64_t k,
                                          double enu, torch::Tensor r_t, torch::Tensor v_t) {
    Rel rel = (Rel)rel_i;
    auto r = tensor_to_vec(r_t);
    auto v = tensor_to_vec(v_t);
    int nr = (int)r.size();
    Integrator solver((int)zn, r, v);
    std::vector<double> chi_p(nr, 0.0), chi_q(nr, 0.0);
    auto res = solver.integrate_forward(rel, enu, (int)l, (int)k, chi_p, chi_q, true);
    return {torch::tensor((int64_t)res.nn), vec_to_tensor(res.p)};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("nodes_bound", &rs_nodes_bound, "debug bound-mode node count");
    m.def("solve", &rs_solve,
          "fixed-energy radial solution with energy derivatives "
          "(rel, dme, l, zn, enu, r, v, k=0) -> (p, rdudr, uderiv[3], nn)",
          py::arg("rel"), py::arg("dme"), py::arg("l"), py::arg("zn"),
          py::arg("enu"), py::arg("r"), py::arg("v"), py::arg("k") = 0);
    m.def("bound_state", &rs_bound_state,
          "(rel, zn, n, l, k, enu_start, r, v, alpha0, alpha1) -> (enu, p, rho)");
    m.def("enu_finder", &rs_enu_finder,
          "(rel, zn, n, l, enu_start, auto_enu, r, v) -> enu");
}
