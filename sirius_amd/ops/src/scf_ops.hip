// CDNA4 (gfx950) kernels for the plane-wave SCF hot path.
//
// Reference GPU twins (behavior, not code): src/core/gpu_kernels/
//   local_operator.cu  (add_to_hphi_pw, mul_by_veff_*)
//   residuals_aux.cu   (compute_residuals, apply_preconditioner, norms)
//   density_rg.cu      (update_density_rg_*)
//   create_beta_gk.cu  (beta phase application)
//
// MI355X-first design notes:
//  - complex128 = double2: one 16-byte naturally-vectorized load per element
//    (guide: Guideline 13 — always vectorize; double2 is the widest natural
//    access for fp64 complex).
//  - All kernels are memory-bound; target is HBM3E streaming rate. Fusions
//    (unpack+kinetic, residual+precondition+norm) remove whole passes over
//    [nbands, nG] arrays rather than micro-optimizing FLOPs.
//  - 256-thread blocks (4 waves of 64); grid-stride loops sized ≫256
//    workgroups to fill 8 XCDs.
//  - No CUDA compatibility paths: this file is HIP-only, compiled
//    --offload-arch=gfx950.

#include <hip/hip_runtime.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_HIP(t) TORCH_CHECK((t).is_cuda(), #t " must be on device")

static inline int grid_1d(long n, int block) {
    long g = (n + block - 1) / block;
    const long cap = 16384;  // ≫256 workgroups, bounded
    return (int)(g < cap ? g : cap);
}

// ---------------------------------------------------------------------------
// pack: dense_grid[b, idx[g]] = coeff[b, g]   (grid must be pre-zeroed)
// ---------------------------------------------------------------------------
__global__ void pack_sphere_kernel(const double2* __restrict__ coeff,
                                   const long* __restrict__ idx,
                                   double2* __restrict__ grid,
                                   long ng, long grid_size, int nb) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < ng * nb;
         i += (long)gridDim.x * blockDim.x) {
        int b = i / ng;
        long g = i - (long)b * ng;
        grid[(long)b * grid_size + idx[g]] = coeff[i];
    }
}

// ---------------------------------------------------------------------------
// unpack + kinetic: out[b,g] = scale*grid[b, idx[g]] + ekin[g]*psi[b,g]
// (ekin = ½|G+k|² precomputed on host side; `scale` folds the 1/N FFT
// normalization in — the unscaled fftn avoids a full-grid scale kernel)
// (fusion of SpFFT unpack with add_to_hphi_pw, local_operator.cu:32-60)
// ---------------------------------------------------------------------------
__global__ void unpack_add_kinetic_kernel(const double2* __restrict__ grid,
                                          const long* __restrict__ idx,
                                          const double* __restrict__ ekin,
                                          const double2* __restrict__ psi,
                                          double2* __restrict__ out,
                                          long ng, long grid_size, int nb,
                                          double scale) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < ng * nb;
         i += (long)gridDim.x * blockDim.x) {
        int b = i / ng;
        long g = i - (long)b * ng;
        double2 v = grid[(long)b * grid_size + idx[g]];
        double2 p = psi[i];
        double t = ekin[g];
        out[i] = make_double2(scale * v.x + t * p.x, scale * v.y + t * p.y);
    }
}

// plain unpack: out[b,g] = scale*grid[b, idx[g]]
__global__ void unpack_sphere_kernel(const double2* __restrict__ grid,
                                     const long* __restrict__ idx,
                                     double2* __restrict__ out,
                                     long ng, long grid_size, int nb,
                                     double scale) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < ng * nb;
         i += (long)gridDim.x * blockDim.x) {
        int b = i / ng;
        long g = i - (long)b * ng;
        double2 v = grid[(long)b * grid_size + idx[g]];
        out[i] = make_double2(scale * v.x, scale * v.y);
    }
}

// ---------------------------------------------------------------------------
// mul_by_veff: grid[b, r] *= veff[r]  (real potential × complex ψ(r);
// reference mul_by_veff_rr/cr, local_operator.cu:107-130)
// ---------------------------------------------------------------------------
__global__ void mul_veff_kernel(double2* __restrict__ grid,
                                const double* __restrict__ veff,
                                long grid_size, int nb) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < grid_size * nb;
         i += (long)gridDim.x * blockDim.x) {
        long r = i % grid_size;
        double v = veff[r];
        double2 z = grid[i];
        grid[i] = make_double2(z.x * v, z.y * v);
    }
}

// ---------------------------------------------------------------------------
// density accumulate: rho[r] += Σ_b w[b] * |psi_r[b, r]|²
// (reference update_density_rg_1_complex_gpu, density_rg.cu:21-60)
// one thread per grid point looping bands: no atomics, rho stays resident.
// ---------------------------------------------------------------------------
__global__ void density_acc_kernel(const double2* __restrict__ psi_r,
                                   const double* __restrict__ w,
                                   double* __restrict__ rho,
                                   long grid_size, int nb) {
    for (long r = blockIdx.x * blockDim.x + threadIdx.x; r < grid_size;
         r += (long)gridDim.x * blockDim.x) {
        double acc = 0.0;
        for (int b = 0; b < nb; b++) {
            double2 z = psi_r[(long)b * grid_size + r];
            acc += w[b] * (z.x * z.x + z.y * z.y);
        }
        rho[r] += acc;
    }
}

// ---------------------------------------------------------------------------
// residuals: res[b,g] = (hpsi[b,g] − e[b]·spsi[b,g]) / P(h_diag, o_diag, e)
// with P = ½(1 + t + sqrt(1 + (t−1)²)), t = h_diag − e·o_diag, plus per-band
// squared norms (fusion of compute_residuals + apply_preconditioner +
// add_square_sum, residuals_aux.cu:21-110 & :300-315).
// norms2 must be zeroed before the call.
// ---------------------------------------------------------------------------
__global__ void residual_precond_kernel(const double2* __restrict__ hpsi,
                                        const double2* __restrict__ spsi,
                                        const double* __restrict__ eval,
                                        const double* __restrict__ h_diag,
                                        const double* __restrict__ o_diag,
                                        double2* __restrict__ res,
                                        double* __restrict__ norms2,
                                        long ng, int nb) {
    __shared__ double red[256 / 64];
    // 2D: y = band, x = G chunk
    int b = blockIdx.y;
    double e = eval[b];
    double local = 0.0;
    for (long g = blockIdx.x * blockDim.x + threadIdx.x; g < ng;
         g += (long)gridDim.x * blockDim.x) {
        double2 h = hpsi[(long)b * ng + g];
        double2 s = spsi[(long)b * ng + g];
        double rx = h.x - e * s.x;
        double ry = h.y - e * s.y;
        local += rx * rx + ry * ry;  // norm of UNpreconditioned residual
        double t = h_diag[g] - e * o_diag[g];
        double p = 0.5 * (1.0 + t + sqrt(1.0 + (t - 1.0) * (t - 1.0)));
        res[(long)b * ng + g] = make_double2(rx / p, ry / p);
    }
    // wave then block reduction, one atomic per block
    for (int off = 32; off > 0; off >>= 1)
        local += __shfl_down(local, off, 64);
    int wave = threadIdx.x / 64;
    if ((threadIdx.x & 63) == 0) red[wave] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
        double s = 0.0;
        for (int wv = 0; wv < blockDim.x / 64; wv++) s += red[wv];
        atomicAdd(&norms2[b], s);
    }
}

// ---------------------------------------------------------------------------
// beta phase: beta[g, off+xi] = beta_t[g, xi] * phase[g] for one atom
// batched over atoms via blockIdx.y (reference create_beta_gk.cu:25-156).
// beta_t is per-type [ng, nbf_t] (column-major over xi as used from torch
// row-major [nbf, ng] transposed views — we take simple row-major here:
// beta_t[xi, g], beta_out[off+xi, g]).
// phase argument: e^{-2πi (G+k)·τ} computed from miller+k dot tau.
// ---------------------------------------------------------------------------
__global__ void beta_phase_kernel(const double2* __restrict__ beta_t,
                                  const double* __restrict__ mk,   // [ng,3] (G+k) frac
                                  const double* __restrict__ tau,  // [na,3]
                                  const int* __restrict__ type_off,   // per atom: offset in beta_t rows
                                  const int* __restrict__ atom_nbf,   // per atom: nbf
                                  const int* __restrict__ atom_off,   // per atom: output row offset
                                  double2* __restrict__ beta_out,
                                  long ng, int na) {
    int ia = blockIdx.y;
    const double twopi = 6.283185307179586476925286766559;
    double tx = tau[ia * 3 + 0], ty = tau[ia * 3 + 1], tz = tau[ia * 3 + 2];
    int nbf = atom_nbf[ia];
    int t_off = type_off[ia];
    int o_off = atom_off[ia];
    for (long g = blockIdx.x * blockDim.x + threadIdx.x; g < ng;
         g += (long)gridDim.x * blockDim.x) {
        double arg = -twopi * (mk[g * 3 + 0] * tx + mk[g * 3 + 1] * ty + mk[g * 3 + 2] * tz);
        double c = cos(arg), s = sin(arg);
        for (int xi = 0; xi < nbf; xi++) {
            double2 z = beta_t[(long)(t_off + xi) * ng + g];
            beta_out[(long)(o_off + xi) * ng + g] =
                make_double2(z.x * c - z.y * s, z.x * s + z.y * c);
        }
    }
}

// ===========================================================================
// C++/torch bindings
// ===========================================================================

static hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

void pack_sphere(torch::Tensor coeff, torch::Tensor idx, torch::Tensor grid) {
    CHECK_HIP(coeff);
    long ng = coeff.size(-1);
    int nb = coeff.numel() / ng;
    long gs = grid.numel() / nb;
    hipLaunchKernelGGL(pack_sphere_kernel, dim3(grid_1d(ng * nb, 256)), dim3(256), 0,
                       cur_stream(),
                       (const double2*)coeff.data_ptr(), idx.data_ptr<long>(),
                       (double2*)grid.data_ptr(), ng, gs, nb);
}

void unpack_sphere(torch::Tensor grid, torch::Tensor idx, torch::Tensor out,
                   double scale) {
    CHECK_HIP(grid);
    long ng = out.size(-1);
    int nb = out.numel() / ng;
    long gs = grid.numel() / nb;
    hipLaunchKernelGGL(unpack_sphere_kernel, dim3(grid_1d(ng * nb, 256)), dim3(256), 0,
                       cur_stream(),
                       (const double2*)grid.data_ptr(), idx.data_ptr<long>(),
                       (double2*)out.data_ptr(), ng, gs, nb, scale);
}

void unpack_add_kinetic(torch::Tensor grid, torch::Tensor idx, torch::Tensor gk2,
                        torch::Tensor psi, torch::Tensor out, double scale) {
    CHECK_HIP(grid);
    long ng = psi.size(-1);
    int nb = psi.numel() / ng;
    long gs = grid.numel() / nb;
    hipLaunchKernelGGL(unpack_add_kinetic_kernel, dim3(grid_1d(ng * nb, 256)),
                       dim3(256), 0, cur_stream(),
                       (const double2*)grid.data_ptr(), idx.data_ptr<long>(),
                       gk2.data_ptr<double>(), (const double2*)psi.data_ptr(),
                       (double2*)out.data_ptr(), ng, gs, nb, scale);
}

void mul_veff(torch::Tensor grid, torch::Tensor veff) {
    CHECK_HIP(grid);
    long gs = veff.numel();
    int nb = grid.numel() / gs;
    hipLaunchKernelGGL(mul_veff_kernel, dim3(grid_1d(gs * nb, 256)), dim3(256), 0,
                       cur_stream(),
                       (double2*)grid.data_ptr(), veff.data_ptr<double>(), gs, nb);
}

void beta_phase(torch::Tensor beta_types, torch::Tensor mk, torch::Tensor tau,
                torch::Tensor type_off, torch::Tensor atom_nbf,
                torch::Tensor atom_off, torch::Tensor beta_out) {
    CHECK_HIP(beta_types);
    long ng = beta_types.size(-1);
    int na = tau.size(0);
    dim3 grid(grid_1d(ng, 256), na);
    hipLaunchKernelGGL(beta_phase_kernel, grid, dim3(256), 0, cur_stream(),
                       (const double2*)beta_types.data_ptr(),
                       mk.data_ptr<double>(), tau.data_ptr<double>(),
                       type_off.data_ptr<int>(), atom_nbf.data_ptr<int>(),
                       atom_off.data_ptr<int>(),
                       (double2*)beta_out.data_ptr(), ng, na);
}

void density_acc(torch::Tensor psi_r, torch::Tensor w, torch::Tensor rho) {
    CHECK_HIP(psi_r);
    long gs = rho.numel();
    int nb = psi_r.numel() / gs;
    hipLaunchKernelGGL(density_acc_kernel, dim3(grid_1d(gs, 256)), dim3(256), 0,
                       cur_stream(),
                       (const double2*)psi_r.data_ptr(), w.data_ptr<double>(),
                       rho.data_ptr<double>(), gs, nb);
}

torch::Tensor residual_precond(torch::Tensor hpsi, torch::Tensor spsi,
                               torch::Tensor eval, torch::Tensor h_diag,
                               torch::Tensor o_diag, torch::Tensor res) {
    CHECK_HIP(hpsi);
    long ng = hpsi.size(-1);
    int nb = hpsi.numel() / ng;
    auto norms2 = torch::zeros({nb}, hpsi.options().dtype(torch::kFloat64));
    dim3 grid(grid_1d(ng, 256) < 64 ? grid_1d(ng, 256) : 64, nb);
    hipLaunchKernelGGL(residual_precond_kernel, grid, dim3(256), 0, cur_stream(),
                       (const double2*)hpsi.data_ptr(), (const double2*)spsi.data_ptr(),
                       eval.data_ptr<double>(), h_diag.data_ptr<double>(),
                       o_diag.data_ptr<double>(), (double2*)res.data_ptr(),
                       norms2.data_ptr<double>(), ng, nb);
    return norms2;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("pack_sphere", &pack_sphere, "scatter sphere coeffs into dense FFT grid");
    m.def("unpack_sphere", &unpack_sphere, "gather sphere coeffs from dense FFT grid",
          py::arg("grid"), py::arg("idx"), py::arg("out"), py::arg("scale") = 1.0);
    m.def("unpack_add_kinetic", &unpack_add_kinetic,
          "gather + add 0.5|G+k|^2 psi (fused local-op epilogue)",
          py::arg("grid"), py::arg("idx"), py::arg("gk2"), py::arg("psi"),
          py::arg("out"), py::arg("scale") = 1.0);
    m.def("mul_veff", &mul_veff, "in-place psi(r) *= V(r)");
    m.def("density_acc", &density_acc, "rho(r) += sum_b w_b |psi_b(r)|^2");
    m.def("beta_phase", &beta_phase,
          "beta(G+k) per atom = type columns x e^{-i(G+k).tau} "
          "(create_beta_gk.cu twin)");
    m.def("residual_precond", &residual_precond,
          "res=(h-e*s)psi/P fused with norm reduction; returns norms^2");
}
