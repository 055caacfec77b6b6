// Hand-written CDNA4 (gfx950) MFMA fp64 kernel for the Davidson subspace
// Gram product:  C[M,N] = Σ_k conj(A[m,k])·B[n,k]
//
// This is the reference's SPLA pgemm_ssb seam (wf::inner,
// wave_functions.hpp:1659-1723) — the ⟨φ|φ⟩ / ⟨φ|Hφ⟩ tall-skinny
// contraction with K = num_gvec (10⁴-10⁵) and M,N = subspace size
// (10²-10³). rocBLAS/Tensile reaches only ~13.5 TF/s fp64 on this shape
// (profiles/r01_si64_1gpu_kernel_stats_v2.csv); this kernel uses
// v_mfma_f64_16x16x4f64 tiles with LDS staging and split-K over the long
// dimension, combining with global fp64 atomic adds.
//
// Geometry: 256 threads = 4 waves; each wave owns a 16×16 complex output
// tile; the workgroup computes a 32×32 complex tile of C. blockIdx.z
// split-K chunks stream disjoint K ranges. A and B rows are contiguous
// in k (torch row-major [rows, K]), so LDS stages are fully coalesced.

#include <hip/hip_runtime.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef double d4 __attribute__((ext_vector_type(4)));

#define TILE 32              // block tile (M and N)
#define KCH 16               // k-chunk staged in LDS per iteration (32 tried: slower end-to-end)
#define NSLOT ((TILE * KCH) / 256)   // staging slots per thread

__launch_bounds__(256, 2)
__global__ void zgram_splitk_kernel(const double* __restrict__ A,  // [M, 2K] interleaved
                                    const double* __restrict__ B,  // [N, 2K]
                                    double* __restrict__ C,        // [M, 2N] interleaved
                                    int M, int N, long K, long chunk) {
    const long k_begin = (long)blockIdx.z * chunk;
    const long k_end = min(K, k_begin + chunk);
    // LDS: A tile [TILE rows][KCH][2], B tile same: 2 * 32*16*2*8 = 16 KiB
    __shared__ double lA[TILE][2 * KCH + 2];
    __shared__ double lB[TILE][2 * KCH + 2];

    const int m0 = blockIdx.x * TILE;
    const int n0 = blockIdx.y * TILE;
    if (m0 >= M || n0 >= N) {
        return;
    }

    const int tid = threadIdx.x;
    const int wave = tid / 64;          // 0..3
    const int lane = tid % 64;
    // wave grid 2x2 over the 32x32 tile
    const int wm = (wave / 2) * 16;     // wave row offset inside tile
    const int wn = (wave % 2) * 16;     // wave col offset

    // accumulators: C_re, C_im (16x16 f64 tile each, 4 values per lane)
    d4 acc_re = {0, 0, 0, 0};
    d4 acc_im = {0, 0, 0, 0};

    // mfma_f64_16x16x4: A operand: lane l holds A[row = l%16, k = l/16]
    // B operand: lane l holds B[k = l/16, col = l%16]
    const int a_row = lane % 16;
    const int a_k = lane / 16;          // 0..3

    // software pipeline: each thread owns NSLOT fixed (row, k) staging
    // slots; the NEXT chunk is prefetched into registers while MFMAs
    // consume the LDS-resident one.
    int rr[NSLOT], kk[NSLOT];
    #pragma unroll
    for (int j = 0; j < NSLOT; j++) {
        rr[j] = (tid + 256 * j) / KCH;
        kk[j] = (tid + 256 * j) % KCH;
    }
    double pa[NSLOT][2], pb[NSLOT][2];

    auto prefetch = [&](long kb) {
        const int kc = (int)min((long)KCH, k_end - kb);
        #pragma unroll
        for (int j = 0; j < NSLOT; j++) {
            pa[j][0] = pa[j][1] = pb[j][0] = pb[j][1] = 0.0;
            if (kk[j] < kc) {
                if (m0 + rr[j] < M) {
                    const double* src = A + ((long)(m0 + rr[j])) * 2 * K + 2 * (kb + kk[j]);
                    pa[j][0] = src[0];
                    pa[j][1] = src[1];
                }
                if (n0 + rr[j] < N) {
                    const double* src = B + ((long)(n0 + rr[j])) * 2 * K + 2 * (kb + kk[j]);
                    pb[j][0] = src[0];
                    pb[j][1] = src[1];
                }
            }
        }
    };
    auto commit = [&]() {
        #pragma unroll
        for (int j = 0; j < NSLOT; j++) {
            lA[rr[j]][2 * kk[j]] = pa[j][0];
            lA[rr[j]][2 * kk[j] + 1] = pa[j][1];
            lB[rr[j]][2 * kk[j]] = pb[j][0];
            lB[rr[j]][2 * kk[j] + 1] = pb[j][1];
        }
    };

    prefetch(k_begin);
    for (long kb = k_begin; kb < k_end; kb += KCH) {
        commit();
        __syncthreads();
        if (kb + KCH < k_end) {
            prefetch(kb + KCH);
        }
        // 4 MFMA k-steps of 4 over the 16-wide chunk
        #pragma unroll
        for (int ks = 0; ks < KCH; ks += 4) {
            double ar = lA[wm + a_row][2 * (ks + a_k)];
            double ai = lA[wm + a_row][2 * (ks + a_k) + 1];
            double br = lB[wn + a_row][2 * (ks + a_k)];
            double bi = lB[wn + a_row][2 * (ks + a_k) + 1];
            // C = conj(a)·b: re += ar·br + ai·bi ; im += ar·bi − ai·br
            acc_re = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, br, acc_re, 0, 0, 0);
            acc_re = __builtin_amdgcn_mfma_f64_16x16x4f64(ai, bi, acc_re, 0, 0, 0);
            acc_im = __builtin_amdgcn_mfma_f64_16x16x4f64(ar, bi, acc_im, 0, 0, 0);
            acc_im = __builtin_amdgcn_mfma_f64_16x16x4f64(-ai, br, acc_im, 0, 0, 0);
        }
        __syncthreads();
    }

    // accumulator layout of mfma_f64_16x16x4 (probed on gfx950 hardware):
    // lane l, vreg v holds C[row = 4*v + l/16, col = l%16]
    const int c_col = lane % 16;
    const int c_rowg = lane / 16;
    #pragma unroll
    for (int v = 0; v < 4; v++) {
        int m = m0 + wm + 4 * v + c_rowg;
        int n = n0 + wn + c_col;
        if (m < M && n < N) {
            unsafeAtomicAdd(&C[(long)m * 2 * N + 2 * n], acc_re[v]);
            unsafeAtomicAdd(&C[(long)m * 2 * N + 2 * n + 1], acc_im[v]);
        }
    }
}

// Subspace transform:  C[m,g] = alpha * Σ_k op(T[k,m]) · X[k,g]  (+ C[m,g])
//
// The other half of the Davidson seam (wf::transform, SPLA pgemm_sbs,
// wave_functions.hpp:1441-1520): applying the small subspace matrix Z
// ([K,M], K=subspace N ≈ 150-600) to the wavefunction stack X ([K, nG],
// nG ≈ 10⁴-10⁵). rocBLAS runs this at ~2.2 TF/s (Cijk_Alik_* tiles,
// profiles/r01 prof3). Here: each workgroup owns a 32m×32g output tile
// (2×2 waves of 16×16 MFMA), k-loop over the short dimension, direct
// coalesced writeback (no atomics; G/32 tiles fill the 256 CUs).
__launch_bounds__(256, 2)
__global__ void ztrans_kernel(const double* __restrict__ T,  // [K, 2M] interleaved
                              const double* __restrict__ X,  // [K, 2G]
                              double* __restrict__ C,        // [M, 2G]
                              int M, long G, int K, int conj_t,
                              double alpha, int accumulate) {
    __shared__ double lT[KCH][2 * TILE + 2];
    __shared__ double lX[KCH][2 * TILE + 2];

    const int m0 = blockIdx.x * TILE;
    const long g0 = (long)blockIdx.y * TILE;
    if (m0 >= M || g0 >= G) {
        return;
    }
    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int wm = (wave / 2) * 16;
    const int wn = (wave % 2) * 16;

    d4 acc_re = {0, 0, 0, 0};
    d4 acc_im = {0, 0, 0, 0};
    const int a_row = lane % 16;     // m within wave tile
    const int a_k = lane / 16;       // k within 4-step
    const double tsgn = conj_t ? -1.0 : 1.0;

    // software pipeline (same scheme as zgram): NSLOT staging slots per
    // thread, next chunk prefetched into registers during the MFMAs.
    int kks[NSLOT], ccs[NSLOT];
    #pragma unroll
    for (int j = 0; j < NSLOT; j++) {
        kks[j] = (tid + 256 * j) / TILE;
        ccs[j] = (tid + 256 * j) % TILE;
    }
    double pt[NSLOT][2], px[NSLOT][2];

    auto prefetch = [&](int kb) {
        const int kc = min(KCH, K - kb);
        #pragma unroll
        for (int j = 0; j < NSLOT; j++) {
            pt[j][0] = pt[j][1] = px[j][0] = px[j][1] = 0.0;
            if (kks[j] < kc) {
                if (m0 + ccs[j] < M) {
                    const double* src = T + ((long)(kb + kks[j])) * 2 * M + 2 * (m0 + ccs[j]);
                    pt[j][0] = src[0];
                    pt[j][1] = src[1];
                }
                if (g0 + ccs[j] < G) {
                    const double* src = X + ((long)(kb + kks[j])) * 2 * G + 2 * (g0 + ccs[j]);
                    px[j][0] = src[0];
                    px[j][1] = src[1];
                }
            }
        }
    };
    auto commit = [&]() {
        #pragma unroll
        for (int j = 0; j < NSLOT; j++) {
            lT[kks[j]][2 * ccs[j]] = pt[j][0];
            lT[kks[j]][2 * ccs[j] + 1] = pt[j][1];
            lX[kks[j]][2 * ccs[j]] = px[j][0];
            lX[kks[j]][2 * ccs[j] + 1] = px[j][1];
        }
    };

    prefetch(0);
    for (int kb = 0; kb < K; kb += KCH) {
        commit();
        __syncthreads();
        if (kb + KCH < K) {
            prefetch(kb + KCH);
        }
        #pragma unroll
        for (int ks = 0; ks < KCH; ks += 4) {
            double tr = lT[ks + a_k][2 * (wm + a_row)];
            double ti = tsgn * lT[ks + a_k][2 * (wm + a_row) + 1];
            double xr = lX[ks + a_k][2 * (wn + a_row)];
            double xi = lX[ks + a_k][2 * (wn + a_row) + 1];
            // C += (tr + i ti)(xr + i xi)
            acc_re = __builtin_amdgcn_mfma_f64_16x16x4f64(tr, xr, acc_re, 0, 0, 0);
            acc_re = __builtin_amdgcn_mfma_f64_16x16x4f64(-ti, xi, acc_re, 0, 0, 0);
            acc_im = __builtin_amdgcn_mfma_f64_16x16x4f64(tr, xi, acc_im, 0, 0, 0);
            acc_im = __builtin_amdgcn_mfma_f64_16x16x4f64(ti, xr, acc_im, 0, 0, 0);
        }
        __syncthreads();
    }

    const int c_col = lane % 16;
    const int c_rowg = lane / 16;
    #pragma unroll
    for (int v = 0; v < 4; v++) {
        int m = m0 + wm + 4 * v + c_rowg;
        long g = g0 + wn + c_col;
        if (m < M && g < G) {
            double* dst = C + (long)m * 2 * G + 2 * g;
            double re = alpha * acc_re[v];
            double im = alpha * acc_im[v];
            if (accumulate) {
                dst[0] += re;
                dst[1] += im;
            } else {
                dst[0] = re;
                dst[1] = im;
            }
        }
    }
}

void ztrans(torch::Tensor T, torch::Tensor X, torch::Tensor C,
            bool conj_t, double alpha, bool accumulate) {
    TORCH_CHECK(T.is_cuda() && X.is_cuda() && C.is_cuda(), "device tensors required");
    TORCH_CHECK(T.is_contiguous() && X.is_contiguous() && C.is_contiguous(),
                "contiguous required");
    const int K = T.size(0);
    const int M = T.size(1);
    const long G = X.size(1);
    TORCH_CHECK(X.size(0) == K, "K mismatch");
    TORCH_CHECK(C.size(0) == M && C.size(1) == G, "C shape mismatch");
    if (M == 0 || G == 0) {
        return;
    }
    if (K == 0) {
        if (!accumulate) {
            C.zero_();
        }
        return;
    }
    dim3 grid((M + TILE - 1) / TILE, (G + TILE - 1) / TILE, 1);
    auto stream = at::hip::getCurrentHIPStream().stream();
    hipLaunchKernelGGL(ztrans_kernel, grid, dim3(256), 0, stream,
                       (const double*)T.data_ptr(), (const double*)X.data_ptr(),
                       (double*)C.data_ptr(), M, G, K,
                       conj_t ? 1 : 0, alpha, accumulate ? 1 : 0);
}

torch::Tensor zgram(torch::Tensor A, torch::Tensor B, int split_k) {
    TORCH_CHECK(A.is_cuda() && B.is_cuda(), "device tensors required");
    TORCH_CHECK(A.is_contiguous() && B.is_contiguous(), "contiguous required");
    const int M = A.size(0);
    const int N = B.size(0);
    const long K = A.size(1);
    TORCH_CHECK(B.size(1) == K, "K mismatch");
    auto C = torch::zeros({M, N}, A.options());
    if (split_k <= 0) {
        // heuristic: enough chunks to fill 256 CUs
        long tiles = ((M + TILE - 1) / TILE) * ((N + TILE - 1) / TILE);
        split_k = (int)std::min<long>(64, std::max<long>(1, 512 / std::max<long>(tiles, 1)));
    }
    long chunk = (K + split_k - 1) / split_k;
    chunk = ((chunk + KCH - 1) / KCH) * KCH;   // multiple of the LDS stage
    split_k = (int)((K + chunk - 1) / chunk);

    dim3 grid((M + TILE - 1) / TILE, (N + TILE - 1) / TILE, split_k);
    auto stream = at::hip::getCurrentHIPStream().stream();
    hipLaunchKernelGGL(zgram_splitk_kernel, grid, dim3(256), 0, stream,
                       (const double*)A.data_ptr(), (const double*)B.data_ptr(),
                       (double*)C.data_ptr(), M, N, K, chunk);
    return C;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("zgram", &zgram, "C[m,n] = sum_k conj(A[m,k]) B[n,k] (MFMA f64)",
          py::arg("A"), py::arg("B"), py::arg("split_k") = 0);
    m.def("ztrans", &ztrans,
          "C[m,g] (+)= alpha * sum_k op(T[k,m]) X[k,g] (MFMA f64)",
          py::arg("T"), py::arg("X"), py::arg("C"), py::arg("conj_t") = false,
          py::arg("alpha") = 1.0, py::arg("accumulate") = false);
}
