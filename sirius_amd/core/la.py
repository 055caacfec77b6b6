"""Dense linear algebra helpers.

Reference behavior: src/core/la (Eigensolver hierarchy eigenproblem.hpp:39,
chosen by control.std_evp_solver_name; wf::orthogonalize Cholesky path
wave_functions.hpp:1781).

MI355X policy (measured, profiles/r01_si64_1gpu_kernel_stats.csv):
rocSOLVER zheevd on subspace-sized matrices (N≈300) is launch-bound —
~500 kernel dispatches per solve (sytd2/hemv/her2/larfg storms) eating
~25% of SCF GPU time. For N ≤ EIGH_GPU_MIN we therefore solve on the host
(LAPACK via torch-CPU, ~12 ms at N=306) and keep the GPU queue free for
the GEMM/FFT stream; large N stays on rocSOLVER.
"""

from __future__ import annotations

import torch

EIGH_GPU_MIN = 1024  # below this, host LAPACK beats rocSOLVER's launch storm

# LAPACK on many-core hosts (128 threads on the MI355X boxes) oversubscribes
# badly at subspace sizes (~300): 45 ms/zheevd at 128 threads vs ~12 ms at 8.
_HOST_SOLVE_THREADS = min(16, torch.get_num_threads())


PIN_HOST_THREADS = False  # threaded k-loop: caller pinned the pool already


class _host_threads:
    def __enter__(self):
        if PIN_HOST_THREADS:
            return
        self.saved = torch.get_num_threads()
        torch.set_num_threads(_HOST_SOLVE_THREADS)

    def __exit__(self, *a):
        if PIN_HOST_THREADS:
            return
        torch.set_num_threads(self.saved)


# NOTE on staging: pinned bounce buffers for these host-solve round trips
# were tried and measured SLOWER end-to-end (0.265 -> 0.284 s/SCF-iter):
# the extra host-side copies and the blocking sync on the pinned D2H sit
# on the critical path, while the pageable copies overlap with queued GPU
# work. Plain .cpu()/.to() stays.
def to_host(t: torch.Tensor) -> torch.Tensor:
    return t.cpu() if t.is_cuda else t


def to_dev(t: torch.Tensor, device) -> torch.Tensor:
    return t if str(device) == "cpu" else t.to(device)


def eigh(H: torch.Tensor):
    """Hermitian eigensolve returning (evals, evecs) on H's device."""
    n = H.shape[-1]
    if H.is_cuda and n < EIGH_GPU_MIN:
        with _host_threads():
            w, v = torch.linalg.eigh(to_host(H))
        return to_dev(w, H.device), to_dev(v, H.device)
    return torch.linalg.eigh(H)


def cholesky(S: torch.Tensor):
    n = S.shape[-1]
    if S.is_cuda and n < EIGH_GPU_MIN:
        with _host_threads():
            return to_dev(torch.linalg.cholesky(to_host(S)), S.device)
    return torch.linalg.cholesky(S)


def inv_lower(L: torch.Tensor) -> torch.Tensor:
    """L^{-1} for lower-triangular L (small; host when on GPU)."""
    eye = torch.eye(L.shape[-1], dtype=L.dtype, device=L.device)
    if L.is_cuda and L.shape[-1] < EIGH_GPU_MIN:
        with _host_threads():
            out = torch.linalg.solve_triangular(to_host(L), eye.cpu(),
                                                upper=False)
        return to_dev(out, L.device)
    return torch.linalg.solve_triangular(L, eye, upper=False)


def ortho_factor(gram: torch.Tensor) -> torch.Tensor:
    """conj(L^{-1}) for the Cholesky factor of a Gram matrix, in ONE host
    round trip (the Davidson orthonormalization transform). Raises on a
    non-positive-definite gram like cholesky does."""
    n = gram.shape[-1]
    if gram.is_cuda and n < EIGH_GPU_MIN:
        with _host_threads():
            g = to_host(gram)
            L = torch.linalg.cholesky(g)
            t = torch.linalg.solve_triangular(
                L, torch.eye(n, dtype=g.dtype), upper=False).conj()
        return to_dev(t.resolve_conj(), gram.device)
    L = torch.linalg.cholesky(gram)
    eye = torch.eye(n, dtype=gram.dtype, device=gram.device)
    return torch.linalg.solve_triangular(L, eye, upper=False).conj()


_ZGRAM_MAX_MN = 1536  # above this rocBLAS catches up (36.6 vs 36.8 TF/s @ 1049)


def inner_gamma(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Real Gram matrix at the Γ point: with the reality constraint
    c(-G) = c*(G), every inner product Σ_G conj(a)b is real and equals
    the dot product of the REAL views — a single dgemm at half the
    complex flops (reference Γ-trick, wave_functions.hpp:1589-1696;
    there the PW set is also halved — here storage stays full-sphere,
    the algebra runs real)."""
    ar = torch.view_as_real(a).reshape(a.shape[0], -1)
    br = torch.view_as_real(b).reshape(b.shape[0], -1)
    return ar @ br.T


def transform_gamma(T: torch.Tensor, X: torch.Tensor,
                    out: torch.Tensor | None = None, alpha: float = 1.0,
                    accumulate: bool = False) -> torch.Tensor:
    """Y (+)= α Tᵀ X with REAL T [K, M] on complex X [K, G] via the real
    view (a single dgemm at half the complex flops) — the Γ-trick
    counterpart of `transform`."""
    Xr = torch.view_as_real(X).reshape(X.shape[0], -1)
    Tt = T.transpose(0, 1)
    if out is not None:
        outr = torch.view_as_real(out).reshape(out.shape[0], -1)
        if accumulate:
            outr.addmm_(Tt, Xr, alpha=alpha)
        else:
            torch.mm(Tt, Xr, out=outr)
            if alpha != 1.0:
                outr *= alpha
        return out
    Yr = alpha * (Tt @ Xr)
    return torch.view_as_complex(
        Yr.reshape(T.shape[1], X.shape[1], 2).contiguous())


def inner(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Gram block ⟨a_i|b_j⟩ = Σ_g conj(a[i,g]) b[j,g].

    GPU path: hand-written CDNA4 MFMA fp64 kernel (ops/src/zgemm_gram.hip,
    v_mfma_f64_16x16x4f64 tiles + LDS staging + split-K) — measured
    12.6× rocBLAS at the Davidson subspace shape (M=N≈300, K≈18k:
    0.58 ms / 23.7 TF/s vs 2.03 ms / 6.8 TF/s; rocBLAS/Tensile has no
    good tall-skinny fp64 tiling). This is the reference's SPLA
    pgemm_ssb seam (wf::inner, wave_functions.hpp:1659).

    Fallback (CPU, huge M/N, or non-contiguous views): a @ b.conj().T is
    a single zgemm with op='C'; only the small result gets conjugated.
    """
    if (a.is_cuda and a.dtype == torch.complex128 and a.dim() == 2
            and b.dim() == 2 and a.shape[0] <= _ZGRAM_MAX_MN
            and b.shape[0] <= _ZGRAM_MAX_MN
            and a.is_contiguous() and b.is_contiguous()):
        from .. import ops

        z = ops.get_zgemm()     # raises on GPU if the extension is missing
        if z is not None:
            return z.zgram(a, b, 0)
    return (a @ b.conj().T).conj()


def transform(T: torch.Tensor, X: torch.Tensor,
              out: torch.Tensor | None = None, alpha: float = 1.0,
              accumulate: bool = False) -> torch.Tensor:
    """Subspace transform C (+)= alpha · Tᵀ X, with T [K, M] small and
    X [K, G] the wavefunction stack (G huge).

    GPU path: MFMA fp64 kernel (ops/src/zgemm_gram.hip ztrans) — rocBLAS
    runs this tall-output shape at only ~2.2 TF/s (38.9% of SCF GPU time
    in profiles/r01 prof3 before this kernel). The reference delegates
    the same contraction to SPLA's pgemm_sbs (wf::transform).
    `accumulate` fuses the AXPY into the writeback (used for the
    project-out step new -= ⟨sphi|new⟩ᵀ phi of wf::orthogonalize).
    """
    if (X.is_cuda and X.dtype == torch.complex128 and T.dim() == 2
            and X.dim() == 2 and X.is_contiguous()
            and (out is None or out.is_contiguous())):
        from .. import ops

        z = ops.get_zgemm()     # raises on GPU if the extension is missing
        if z is not None:
            Tc = T.resolve_conj().contiguous()
            if out is None:
                out = torch.empty(T.shape[1], X.shape[1], dtype=X.dtype,
                                  device=X.device)
            z.ztrans(Tc, X, out, False, float(alpha), accumulate)
            return out
    r = alpha * (T.transpose(0, 1) @ X)
    if accumulate:
        out += r
        return out
    if out is not None:
        out.copy_(r)
        return out
    return r
