"""Block Davidson eigensolver with locking and preconditioned residuals.

Reference behavior: src/hamiltonian/davidson.hpp:130-856 (generic block
Davidson for H|ψ⟩ = ε S|ψ⟩ with locking, restart, adaptive per-band
energy tolerance; expansion block gets apply_h_s first, then {phi, hphi,
sphi} are orthogonalized together — davidson.hpp:771-781), residuals.hpp
(r = Hψ − εSψ, Teter-style diagonal preconditioner
p = ½(1 + t + sqrt(1 + (t−1)²)), t = h_diag − ε o_diag —
residuals_aux.cu:300-315), convergence test |Δε_j| ≤ tol(j)
(davidson.hpp:333-335, diagonalize.hpp:48-52).

All tensors live on the compute device; subspace eigensolves go through
core.la (host LAPACK below the rocSOLVER-viability size), Gram/transform
GEMMs are single zgemms; the fused residual+preconditioner+norm HIP
kernel runs on GPU.
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
    evp_work: float = 0.0      # Σ (N_subspace / num_bands)³ over eigensolves


def _inner(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """⟨a_i|b_j⟩ Gram block: [na, nb] (single zgemm, see core.la.inner)."""
    return la.inner(a, b)


def _ortho_joint(new, hnew, snew, phi, hphi, sphi, outs=None,
                 gamma: bool = False):
    if gamma and new.is_cuda:
        gamma = False      # GPU: keep the MFMA complex path (see davidson)
    """Project the existing S-orthonormal subspace out of `new` and
    S-orthonormalize it, applying identical transforms to hnew/snew
    (mirrors wf::orthogonalize, wave_functions.hpp:1781-2051).

    snew is None ⇔ S = I. Returns (new, hnew, snew) with possibly fewer
    rows when rank-deficient. With `outs` = (phi_buf, h_buf, s_buf) row
    slices, the final orthonormalizing transform writes DIRECTLY into the
    caller's subspace buffers (no extra device copy).

    gamma=True runs the Γ-trick real algebra (Gram matrices real by the
    c(-G)=c*(G) constraint; dgemm transforms at half the complex cost,
    wave_functions.hpp:1589-1696).
    """
    inner_f = la.inner_gamma if gamma else _inner
    trans_f = la.transform_gamma if gamma else la.transform
    s_of_new = snew if snew is not None else new
    if phi is not None and phi.shape[0]:
        ov = inner_f(sphi if sphi is not None else phi, new)   # [N, n]
        new = trans_f(ov, phi, out=new.contiguous(), alpha=-1.0,
                      accumulate=True)
        if hnew is not None:
            hnew = trans_f(ov, hphi, out=hnew.contiguous(), alpha=-1.0,
                           accumulate=True)
        if snew is not None:
            snew = trans_f(ov, sphi, out=snew.contiguous(), alpha=-1.0,
                           accumulate=True)
        s_of_new = snew if snew is not None else new
    gram = inner_f(new, s_of_new)
    gram = 0.5 * (gram + gram.conj().T)
    n = gram.shape[0]
    try:
        t = la.ortho_factor(gram)
    except Exception:
        w, v = la.eigh(gram)
        keep = w > 1e-10
        t = (v[:, keep] / torch.sqrt(w[keep])).conj().T     # [nkeep, n]
    tT = t.transpose(0, 1)
    nkeep = t.shape[0]
    if outs is not None:
        o_phi, o_h, o_s = outs
        new = trans_f(tT, new, out=o_phi[:nkeep])
        hnew = trans_f(tT, hnew, out=o_h[:nkeep]) \
            if hnew is not None else None
        snew = trans_f(tT, snew, out=o_s[:nkeep]) \
            if snew is not None else None
        return new, hnew, snew
    new = trans_f(tT, new)
    hnew = trans_f(tT, hnew) if hnew is not None else None
    snew = trans_f(tT, snew) if snew is not None else None
    return new, hnew, snew


def davidson(apply_h_s, psi0: torch.Tensor, h_diag: torch.Tensor,
             o_diag: torch.Tensor, occ: np.ndarray | None,
             tol_occ: float, tol_empty: float,
             num_steps: int = 20, subspace_size: int = 2,
             max_block: int = 0,
             min_occupancy: float = 1e-14,
             extra_ortho: bool = False, locking: bool = True,
             early_restart: float = 0.5, gamma: bool = False,
             gamma_neg: torch.Tensor | None = None) -> DavidsonResult:
    """Solve for the `nb` lowest eigenpairs of H ψ = ε S ψ (S=I or USPP S).

    apply_h_s(phi [n, nG]) -> (hphi, sphi|None).

    Locking (reference davidson.hpp:704-723): at every subspace restart
    the leading consecutively-converged Ritz vectors are frozen; the
    subsequent subspace eigenproblems shrink by the locked count
    (evp_work counts ((N - locked)/nb)^3, davidson.hpp:828-834).
    `early_restart` triggers a restart before the subspace is full when
    few bands remain unconverged and >5 are lockable (davidson.hpp:622).
    """
    nb, nG = psi0.shape
    num_phi_max = min(max(subspace_size * nb, nb + 1), max(nG // 2, nb))

    tol = np.full(nb, tol_occ)
    if occ is not None:
        tol = np.where(occ > min_occupancy, tol_occ, tol_occ + tol_empty)

    # preallocated subspace buffers: blocks are appended in place instead of
    # torch.cat (each cat re-copies the whole [N, nG] subspace — measured as
    # ~5% pure copyBuffer traffic in profiles/r01_si64_1gpu_kernel_stats_v2)
    dev, cdt = psi0.device, psi0.dtype
    phi_buf = torch.empty(num_phi_max, nG, dtype=cdt, device=dev)
    hphi_buf = torch.empty(num_phi_max, nG, dtype=cdt, device=dev)
    sphi_buf = None

    # Γ-trick split: the real-GEMM algebra wins on CPU (dgemm at half
    # the complex flops); on GPU the complex Grams/transforms run on the
    # hand-written MFMA fp64 kernels which beat rocBLAS dgemm at these
    # shapes — there we keep complex GEMMs but still use the REAL
    # subspace eigensolves (H is real at Γ) and the symmetry enforcement.
    gamma_alg = gamma and not psi0.is_cuda
    inner_c = la.inner_gamma if gamma_alg else _inner
    trans_f = la.transform_gamma if gamma_alg else la.transform

    def inner_f(a, b):
        r = inner_c(a, b)
        if gamma and not gamma_alg:
            r = r.real
        return r

    def to_T(Z):
        # transform coefficient dtype expected by trans_f
        if gamma_alg:
            return Z.real if Z.is_complex() else Z
        return Z if Z.is_complex() else Z.to(psi0.dtype)

    def enforce_gamma(t):
        # re-impose c(-G) = c*(G): the real-algebra Γ path assumes it and
        # roundoff drift otherwise grows unchecked over many steps (the
        # reference enforces it structurally by storing half the G set,
        # wave_functions.hpp:1589)
        if gamma and gamma_neg is not None and t is not None:
            t.add_(t[:, gamma_neg].conj()).mul_(0.5)
        return t

    phi = psi0.clone()
    enforce_gamma(phi)
    hphi, sphi = apply_h_s(phi)
    phi, hphi, sphi = _ortho_joint(phi, hphi, sphi, None, None, None,
                                   gamma=gamma)
    n0 = phi.shape[0]
    phi_buf[:n0] = phi
    hphi_buf[:n0] = hphi
    phi = phi_buf[:n0]
    hphi = hphi_buf[:n0]
    if sphi is not None:
        sphi_buf = torch.empty(num_phi_max, nG, dtype=cdt, device=dev)
        sphi_buf[:n0] = sphi
        sphi = sphi_buf[:n0]
    H = inner_f(phi, hphi)
    H = 0.5 * (H + H.conj().T)
    evals, Z = la.eigh(H)
    evp_work = (phi.shape[0] / nb) ** 3
    eval_old = np.full(nb, 1e10)
    niter = 0
    converged = False
    nlock = 0                      # number of locked (frozen) leading bands
    eval_locked = np.zeros(0)

    pending = np.array([], dtype=np.int64)   # capped-out, not yet expanded
    for it in range(num_steps):
        niter = it + 1
        nb_act = nb - nlock        # bands still solved in the active block
        ev = np.concatenate([eval_locked,
                             evals[:nb_act].real.cpu().numpy()])
        unconv = np.nonzero(np.abs(ev - eval_old[: len(ev)]) > tol)[0]
        unconv = np.union1d(unconv, pending).astype(np.int64)
        unconv = unconv[unconv >= nlock]
        eval_old = ev.copy()
        if len(unconv) == 0:
            converged = True
            break

        N = phi.shape[0]           # active subspace size (excl. locked)
        # residuals of unconverged bands: r_j = (H - e_j S) phi Z_j
        idx = torch.from_numpy(unconv - nlock).to(psi0.device)
        Zs = Z[:, idx]                                     # [N, n]
        e = evals[idx].real
        hpsi = trans_f(to_T(Zs), hphi)                      # [n, nG]
        spsi = trans_f(to_T(Zs), sphi if sphi is not None else phi)
        if hpsi.is_cuda and hpsi.dtype == torch.complex128:
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
        # drop residuals already tiny in (unpreconditioned) norm
        keep = rn > 1e-12
        if not bool(keep.any()):
            converged = True
            break
        res = res[keep]
        if max_block and res.shape[0] > max_block:
            # cap the expansion block (lowest bands first): trades more,
            # cheaper steps for smaller subspace transforms.  Bands cut
            # from the block stay marked pending — an unexpanded band's
            # eigenvalue barely moves, which would otherwise fool the
            # |de| convergence test
            cut = unconv[keep.cpu().numpy()][max_block:]
            pending = np.union1d(pending, cut)
            res = res[:max_block]
            taken = unconv[keep.cpu().numpy()][:max_block]
            pending = np.setdiff1d(pending, taken)
        elif len(pending):
            pending = np.setdiff1d(pending, unconv[keep.cpu().numpy()])
        res = res / torch.linalg.vector_norm(res, dim=1, keepdim=True).to(res.dtype)
        res = res.contiguous()
        enforce_gamma(res)

        n_new = res.shape[0]
        # leading consecutively-converged active bands are lockable
        lockable = int(unconv[0] - nlock) if len(unconv) else nb_act
        should_restart = (nlock + N + n_new > num_phi_max) or \
            (locking and lockable > 5
             and len(unconv) < early_restart * lockable)
        if should_restart:
            # restart: collapse the active subspace to its Ritz vectors
            Znb = to_T(Z[:, :nb_act])
            psi = trans_f(Znb, phi)
            hpsi_f = trans_f(Znb, hphi)
            spsi_f = trans_f(Znb, sphi) if sphi is not None else None
            phi_buf[nlock:nlock + nb_act] = psi
            hphi_buf[nlock:nlock + nb_act] = hpsi_f
            if spsi_f is not None:
                sphi_buf[nlock:nlock + nb_act] = spsi_f
            ev_act = evals[:nb_act].clone()
            if locking and lockable > 0:
                # freeze the converged leading Ritz vectors
                eval_locked = np.concatenate(
                    [eval_locked, ev_act[:lockable].real.cpu().numpy()])
                nlock += lockable
                nb_act = nb - nlock
                ev_act = ev_act[lockable:]
            phi = phi_buf[nlock:nlock + nb_act]
            hphi = hphi_buf[nlock:nlock + nb_act]
            if sphi is not None:
                sphi = sphi_buf[nlock:nlock + nb_act]
            H = torch.diag(ev_act.to(torch.float64 if gamma else phi_buf.dtype))
            evals = ev_act
            Z = torch.eye(nb_act, dtype=H.dtype, device=H.device)
            N = nb_act
            if nlock + N + n_new > num_phi_max:
                n_new = num_phi_max - N - nlock
                res = res[:n_new]
                if n_new <= 0:
                    break

        hnew, snew = apply_h_s(res)
        lo = nlock + N
        outs = (phi_buf[lo:], hphi_buf[lo:],
                sphi_buf[lo:] if sphi is not None else None)
        # project out the full basis incl. locked vectors
        full_phi = phi_buf[:lo]
        full_h = hphi_buf[:lo]
        full_s = sphi_buf[:lo] if sphi is not None else None
        if extra_ortho:
            res, hnew, snew = _ortho_joint(res, hnew, snew, full_phi,
                                           full_h, full_s, gamma=gamma)
        res, hnew, snew = _ortho_joint(res, hnew, snew, full_phi, full_h,
                                       full_s, outs=outs, gamma=gamma)
        if res.shape[0] == 0:
            converged = True
            break
        # subspace grown in place by _ortho_joint(outs=...)
        nn = res.shape[0]
        phi = phi_buf[nlock:lo + nn]
        hphi = hphi_buf[nlock:lo + nn]
        if sphi is not None:
            sphi = sphi_buf[nlock:lo + nn]
        Nn = phi.shape[0]
        Hn = torch.empty(Nn, Nn, dtype=H.dtype, device=H.device)
        Hn[:N, :N] = H
        blk = inner_f(phi, hnew)                           # [Nn, nnew]
        Hn[:, N:] = blk
        Hn[N:, :N] = blk[:N].conj().T
        H = 0.5 * (Hn + Hn.conj().T)
        evals, Z = la.eigh(H)
        evp_work += (Nn / nb) ** 3

    nb_act = nb - nlock
    psi_act = trans_f(to_T(Z[:, :nb_act]), phi)
    if nlock:
        psi = torch.cat([phi_buf[:nlock], psi_act], dim=0)
        ev_out = np.concatenate([eval_locked,
                                 evals[:nb_act].real.cpu().numpy()])
    else:
        psi = psi_act
        ev_out = evals[:nb].real.cpu().numpy()
    return DavidsonResult(eval=ev_out, psi=psi,
                          niter=niter, converged=converged,
                          evp_work=evp_work)
