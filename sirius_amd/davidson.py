"""Block Davidson eigensolver with locking and preconditioned residuals.

Reference behavior: src/hamiltonian/davidson.hpp:130-856 (generic block
Davidson for H|ψ⟩ = ε S|ψ⟩ with locking, restart, adaptive per-band
energy tolerance), residuals.hpp (r = Hψ − εSψ, Teter-style diagonal
preconditioner p = ½(1 + t + sqrt(1 + (t−1)²)), t = h_diag − ε o_diag —
residuals_aux.cu:300-315), convergence test |Δε_j| ≤ tol(j)
(davidson.hpp:333-335, diagonalize.hpp:48-52).

All tensors live on the compute device; subspace eigensolve via
torch.linalg.eigh (rocSOLVER on MI355X), Gram/transform GEMMs via zgemm.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch

from .core import la


@dataclass
class DavidsonResult:
    eval: np.ndarray           # [nb]
    psi: torch.Tensor          # [nb, nG]
    niter: int = 0
    converged: bool = True


def _inner(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """⟨a_i|b_j⟩ Gram block: [na, nb] (single zgemm, see core.la.inner)."""
    return la.inner(a, b)


def _ortho_block(new: torch.Tensor, snew, phi, sphi):
    """Project existing subspace out of `new` and S-orthonormalize it.

    Mirrors wf::orthogonalize (wave_functions.hpp:1781-2051): project-out,
    then Gram + Cholesky + triangular solve. Returns orthonormal block (rows
    may be fewer than input when rank-deficient).
    """
    if phi is not None and phi.shape[0]:
        ov = _inner(sphi if sphi is not None else phi, new)   # [N, n]
        new = new - ov.T @ phi
        if snew is not None:
            snew = snew - ov.T @ sphi
    s = snew if snew is not None else new
    gram = _inner(new, s)
    gram = 0.5 * (gram + gram.conj().T)
    n = gram.shape[0]
    eye = torch.eye(n, dtype=gram.dtype, device=gram.device)
    try:
        L = la.cholesky(gram)
        linv = la.inv_lower(L)
        new = linv.conj() @ new
        snew = linv.conj() @ snew if snew is not None else None
        return new, snew
    except Exception:
        # rank-deficient: keep the well-conditioned subspace via eigh filter
        w, v = la.eigh(gram)
        keep = w > 1e-10
        t = (v[:, keep] / torch.sqrt(w[keep])).conj().T     # [nkeep, n]
        new = t @ new
        snew = t @ snew if snew is not None else None
        return new, snew


def davidson(apply_h_s, psi0: torch.Tensor, h_diag: torch.Tensor,
             o_diag: torch.Tensor, occ: np.ndarray | None,
             tol_occ: float, tol_empty: float,
             num_steps: int = 20, subspace_size: int = 2,
             min_occupancy: float = 1e-14,
             extra_ortho: bool = False) -> DavidsonResult:
    """Solve for the `nb` lowest eigenpairs of H (S=I or USPP S).

    apply_h_s(phi [n, nG]) -> (hphi, sphi|None).
    """
    nb, nG = psi0.shape
    num_phi_max = min(subspace_size * nb, nG // 2) if nG // 2 > nb else nb
    num_phi_max = max(num_phi_max, nb + 1) if nG > nb + 1 else nb

    tol = np.full(nb, tol_occ)
    if occ is not None:
        tol = np.where(occ > min_occupancy, tol_occ, tol_occ + tol_empty)

    phi, sphi = _ortho_block(psi0.clone(), None, None, None)
    hphi, sphi_new = apply_h_s(phi)
    sphi = sphi_new  # None => S = I
    H = _inner(phi, hphi)
    H = 0.5 * (H + H.conj().T)
    evals, Z = la.eigh(H)
    eval_old = np.full(nb, 1e10)
    niter = 0
    converged = False

    for it in range(num_steps):
        niter = it + 1
        ev = evals[:nb].real.cpu().numpy()
        unconv = np.nonzero(np.abs(ev - eval_old[: len(ev)]) > tol)[0]
        eval_old = ev.copy()
        if len(unconv) == 0:
            converged = True
            break

        N = phi.shape[0]
        # residuals of unconverged bands: r_j = (H - e_j S) phi Z_j
        idx = torch.from_numpy(unconv).to(psi0.device)
        Zs = Z[:, idx]                                     # [N, n]
        e = evals[idx].real
        hpsi = Zs.T @ hphi                                  # [n, nG]
        spsi = Zs.T @ (sphi if sphi is not None else phi)
        if hpsi.is_cuda:
            from . import ops

            ext = ops.get_ext(required=True)
            res = torch.empty_like(hpsi)
            norms2 = ext.residual_precond(hpsi.contiguous(), spsi.contiguous(),
                                          e.contiguous(), h_diag.contiguous(),
                                          o_diag.contiguous(), res)
            rn = torch.sqrt(norms2)
        else:
            res = hpsi - e[:, None].to(hpsi.dtype) * spsi
            rn = torch.linalg.vector_norm(res, dim=1).real
            # precondition (residuals_aux.cu apply_preconditioner)
            t = h_diag[None, :] - e[:, None] * o_diag[None, :]
            p = 0.5 * (1.0 + t + torch.sqrt(1.0 + (t - 1.0) ** 2))
            res = res / p.to(res.dtype)
        # drop residuals that are already tiny in (unpreconditioned) norm
        keep = rn > 1e-12
        if not bool(keep.any()):
            converged = True
            break
        res = res[keep]
        e = e[keep]
        res = res / torch.linalg.vector_norm(res, dim=1, keepdim=True).to(res.dtype)

        n_new = res.shape[0]
        if N + n_new > num_phi_max:
            # restart: collapse subspace to current Ritz vectors
            psi = Z[:, :nb].T @ phi
            hpsi_f = Z[:, :nb].T @ hphi
            spsi_f = Z[:, :nb].T @ sphi if sphi is not None else None
            phi, hphi, sphi = psi, hpsi_f, spsi_f
            H = torch.diag(evals[:nb].to(H.dtype))
            evals = evals[:nb].clone()
            Z = torch.eye(nb, dtype=H.dtype, device=H.device)
            N = nb
            if N + n_new > num_phi_max:
                n_new = num_phi_max - N
                res = res[:n_new]
                if n_new <= 0:
                    break

        new, snew0 = _ortho_block(res, None, phi, sphi)
        if extra_ortho:
            new, snew0 = _ortho_block(new, snew0, phi, sphi)
        if new.shape[0] == 0:
            converged = True
            break
        hnew, snew = apply_h_s(new)
        if sphi is not None and snew is None:
            snew = new
        # grow subspace
        phi = torch.cat([phi, new], dim=0)
        hphi = torch.cat([hphi, hnew], dim=0)
        if sphi is not None:
            sphi = torch.cat([sphi, snew], dim=0)
        Nn = phi.shape[0]
        Hn = torch.empty(Nn, Nn, dtype=H.dtype, device=H.device)
        Hn[:N, :N] = H
        blk = _inner(phi, hnew)                            # [Nn, nnew]
        Hn[:, N:] = blk
        Hn[N:, :N] = blk[:N].conj().T
        H = 0.5 * (Hn + Hn.conj().T)
        evals, Z = la.eigh(H)

    psi = Z[:, :nb].T @ phi
    return DavidsonResult(eval=evals[:nb].real.cpu().numpy(), psi=psi,
                          niter=niter, converged=converged)
