"""Block conjugate-gradient solver with per-column convergence, and the
Sternheimer linear-response operator.

Reference behavior: src/multi_cg/multi_cg.hpp —
- `multi_cg` (multi_cg.hpp:44-165): block CG on A X = B with a
  preconditioner; residuals are monitored in the P-norm; converged
  columns are compacted away (we keep an active mask instead — same
  arithmetic, simpler bookkeeping on batched tensors);
- `Smoothed_diagonal_preconditioner` (multi_cg.hpp:265): the Teter
  diagonal preconditioner of the Davidson solver applied per column with
  its own eigenvalue;
- `Linear_response_operator` (multi_cg.hpp:292-385):
  A x_j = (H − ε_j S) x_j + α_pv · S Q (Q^H S x_j), Q = occupied
  eigenvectors — the QE-DFPT Sternheimer operator behind the
  `sirius_linear_solver` API seam.
"""

from __future__ import annotations

import numpy as np
import torch

from .core import la


def multi_cg(apply_A, apply_P, X: torch.Tensor, B: torch.Tensor,
             maxiters: int = 100, tol: float = 1e-3,
             initial_guess_is_zero: bool = False):
    """Solve A X^T = B^T column-block-wise (rows of X/B are the vectors).

    apply_A(X_rows) -> A·rows;  apply_P(R_rows) -> P·rows (may depend on
    the row's position — both get the ACTIVE row index array as second
    arg). Returns (X, n_iter, residual_history[list per rhs]).
    """
    n = X.shape[0]
    R = B.clone()
    if not initial_guess_is_zero:
        R -= apply_A(X, torch.arange(n, device=X.device))
    active = torch.arange(n, device=X.device)
    U = torch.zeros_like(X)
    rhos = torch.zeros(n, dtype=torch.float64, device=X.device)
    hist: list[list[float]] = [[] for _ in range(n)]
    niter = 0
    for it in range(maxiters):
        niter = it
        C = apply_P(R[active], active)
        rhos_old = rhos[active].clone()
        rho = torch.einsum("ig,ig->i", C.conj(), R[active]).real
        for k, i in enumerate(active.tolist()):
            hist[i].append(float(abs(rho[k])) ** 0.5)
        keep = rho.abs() > tol * tol
        if not bool(keep.any()):
            break
        # compact to still-active columns
        active = active[keep]
        C = C[keep]
        rho = rho[keep]
        rhos_old = rhos_old[keep]
        if it == 0:
            U[active] = C
        else:
            alph = (rho / rhos_old).to(U.dtype)
            U[active] = C + alph[:, None] * U[active]
        rhos[active] = rho
        AC = apply_A(U[active], active)
        sig = torch.einsum("ig,ig->i", U[active].conj(), AC).real
        a = (rho / sig).to(X.dtype)
        X[active] += a[:, None] * U[active]
        R[active] -= a[:, None] * AC
    return X, niter, hist


class SmoothedDiagonalPreconditioner:
    """Teter preconditioner per column (residuals_aux.cu:300-315 formula,
    same as the Davidson preconditioner)."""

    def __init__(self, h_diag: torch.Tensor, o_diag: torch.Tensor,
                 eigvals: torch.Tensor):
        self.h_diag = h_diag
        self.o_diag = o_diag
        self.eigvals = eigvals

    def __call__(self, R: torch.Tensor, active) -> torch.Tensor:
        e = self.eigvals[active]
        t = self.h_diag[None, :] - e[:, None] * self.o_diag[None, :]
        p = 0.5 * (1.0 + t + torch.sqrt(1.0 + (t - 1.0) ** 2))
        return R / p.to(R.dtype)


class LinearResponseOperator:
    """(H − ε_j S + α_pv·S Q Q^H S) — multi_cg.hpp:292."""

    def __init__(self, hk, eigvals: torch.Tensor, evq: torch.Tensor,
                 alpha_pv: float, ispn: int = 0):
        self.hk = hk
        self.eigvals = eigvals
        self.evq = evq.contiguous()       # occupied eigenvectors [nocc, nG]
        self.sevq = None
        self.alpha_pv = alpha_pv
        self.ispn = ispn

    def _apply_s(self, x):
        hk = self.hk
        if hk.Q is None or hk.bp.num_beta_total == 0:
            return x
        b = hk.bp.inner(x)
        out = x.clone().contiguous()
        la.transform(hk.Q @ b, hk.bp.beta_t, out=out, accumulate=True)
        return out

    def __call__(self, X: torch.Tensor, active) -> torch.Tensor:
        hk = self.hk
        X = X.contiguous()
        h, s = hk.apply_h_s(X, self.ispn)
        if s is None:
            s = X
        e = self.eigvals[active].to(X.dtype)
        out = h - e[:, None] * s
        # projector α_pv · S Q (Q^H S x)
        ov = la.inner(self.evq, s)                 # [nocc, nact]
        proj = la.transform(ov, self.evq)          # [nact, nG]
        sproj = self._apply_s(proj)
        return out + self.alpha_pv * sproj


def linear_solver(hk, eigvals, evq, B, alpha_pv: float = 1.0,
                  ispn: int = 0, maxiters: int = 300, tol: float = 1e-9):
    """Sternheimer solve (H − ε_j S + α_pv·S Q Q^H S) X = B for a block of
    right-hand sides — the `sirius_linear_solver` API seam used by
    QE-DFPT (reference: sirius_api.cpp sirius_linear_solver →
    multi_cg.hpp lr::multi_cg).

    eigvals [n] (per-rhs ε_j), evq [nocc, nG] occupied eigenvectors,
    B [n, nG]. Returns (X, niter, residual_histories).
    """
    A = LinearResponseOperator(hk, eigvals, evq, alpha_pv, ispn)
    P = SmoothedDiagonalPreconditioner(hk.h_diag(ispn), hk.o_diag(), eigvals)
    X = torch.zeros_like(B)
    return multi_cg(A, P, X, B.clone(), maxiters=maxiters, tol=tol,
                    initial_guess_is_zero=True)
