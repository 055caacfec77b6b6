"""Iterative (Davidson) first-variational LAPW solver.

Reference behavior: src/hamiltonian/diagonalize_fp.hpp:271
(diagonalize_fp_fv_davidson) + hamiltonian_k.cpp:757 (apply_fv_h_o) +
hamiltonian_k.cpp:190 (get_h_o_diag_lapw).

Trial vectors live in the combined (PW, lo) basis: x = (c[ngk], d[nlo]).
The local-orbital unit vectors are kept as a FIXED extra block of the
subspace (reference phi_extra), so lo degrees of freedom are exactly
represented from the first step.  The subspace problem is solved as a
small GENERALIZED eigenproblem with canonical orthogonalization (the
reference O-orthogonalizes the basis instead; same variational space).

The interstitial H/O application uses coarse-grid FFTs with the step
function truncated to the coarse G sphere — this reproduces the dense
set_fv_h_o interstitial blocks exactly (theta and V*theta are zero
outside the 2*gk sphere, and |G1-G2| <= 2 gk always lies inside it).
"""

from __future__ import annotations

import math
import sys

import numpy as np
import torch
from scipy.linalg import eigh


class FVOperator:
    """H and O application for one k-point."""

    def __init__(self, engine, ik, kp):
        self.engine = engine
        self.ctx = engine.ctx
        self.ik = ik
        self.kp = kp
        ctx = self.ctx
        uc = ctx.unit_cell
        self.ngk = kp.num_gkvec
        self.nlo = engine._num_lo_total()
        self.N = self.ngk + self.nlo
        # coarse-grid V*theta and theta (real space)
        if not hasattr(engine, "_vtheta_coarse"):
            ic, if_ = ctx.coarse_fine_pairs

            def _to_coarse(f_pw_fine):
                c = torch.zeros(ctx.gvec_coarse.num_gvec, dtype=ctx.dtype,
                                device=ctx.device)
                c[ic] = f_pw_fine[if_]
                return ctx.fft_coarse.to_real(c).real

            engine._vtheta_coarse = _to_coarse(engine.potential.veff_pw)
            engine._theta_coarse = _to_coarse(ctx.theta_pw)
            rel = ctx.valence_relativity
            if rel in ("zora", "iora"):
                engine._kin_coarse = _to_coarse(engine.potential.rm_inv_pw)
            else:
                engine._kin_coarse = engine._theta_coarse
            if rel == "iora":
                engine._okin_coarse = _to_coarse(engine.potential.rm2_inv_pw)
            else:
                engine._okin_coarse = None
        self.vtheta_rg = engine._vtheta_coarse
        self.theta_rg = engine._theta_coarse
        self.kin_rg = engine._kin_coarse
        self.okin_rg = engine._okin_coarse
        self.gk = kp.gkvec.gkvec_t.to(torch.float64)   # [ngk, 3]
        # per-atom C and hmt/omt
        self.C = []
        self.hmt = []
        self.omt = []
        for ia in range(uc.num_atoms):
            self.C.append(engine._basis_c(ik, kp, ia))
            self.hmt.append(engine._hmt_cached(ia))
            self.omt.append(engine._omt_cached(ia))

    def apply(self, x: torch.Tensor):
        """x [nb, N] -> (Hx [nb, N], Ox [nb, N])."""
        kp = self.kp
        ngk, nlo = self.ngk, self.nlo
        c = x[:, :ngk]
        hx = torch.zeros_like(x)
        ox = torch.zeros_like(x)

        # interstitial (V*theta) psi + kinetic (theta- or ZORA-mass-
        # weighted) + theta overlap (+ IORA overlap correction)
        psir = kp.fft.to_real(c)
        hx[:, :ngk] += kp.fft.to_pw(self.vtheta_rg * psir)
        ox[:, :ngk] += kp.fft.to_pw(self.theta_rg * psir)
        sq_alpha_half = 0.5 / 137.035999139 ** 2
        for ax in range(3):
            gax = self.gk[:, ax]
            gpsir = kp.fft.to_real(c * gax)
            hx[:, :ngk] += 0.5 * gax * kp.fft.to_pw(self.kin_rg * gpsir)
            if self.okin_rg is not None:
                ox[:, :ngk] += 0.5 * sq_alpha_half * gax * kp.fft.to_pw(
                    self.okin_rg * gpsir)

        # MT: S = C x ; hx += C^H hmt S ; ox += C^H omt S
        for ia in range(len(self.C)):
            C = self.C[ia]
            S = C @ x.T                       # [mt, nb]
            hx += (C.conj().T @ (self.hmt[ia] @ S)).T
            ox += (C.conj().T @ (self.omt[ia] @ S)).T
        return hx, ox

    def diag(self):
        """Exact H and O diagonals in the (PW, lo) basis."""
        ngk, nlo = self.ngk, self.nlo
        hd = torch.zeros(self.N, dtype=torch.float64)
        od = torch.zeros(self.N, dtype=torch.float64)
        theta0 = float(self.engine.ctx.theta_pw[
            self.engine.ctx.gvec_fine.index_of_zero()].real)
        v0 = float(self.engine.potential.veff_pw[
            self.engine.ctx.gvec_fine.index_of_zero()].real)
        g2 = (self.gk ** 2).sum(-1)
        rel = self.engine.ctx.valence_relativity
        if rel in ("zora", "iora"):
            kin0 = float(self.engine.potential.rm_inv_pw[
                self.engine.ctx.gvec_fine.index_of_zero()].real)
        else:
            kin0 = theta0
        hd[:ngk] = 0.5 * g2 * kin0 + v0
        od[:ngk] = theta0
        if rel == "iora":
            sq_alpha_half = 0.5 / 137.035999139 ** 2
            rm2_0 = float(self.engine.potential.rm2_inv_pw[
                self.engine.ctx.gvec_fine.index_of_zero()].real)
            od[:ngk] += 0.5 * sq_alpha_half * g2 * rm2_0
        for ia in range(len(self.C)):
            C = self.C[ia]
            hd += torch.einsum("an,ab,bn->n", C.conj(), self.hmt[ia], C).real
            od += torch.einsum("an,ab,bn->n", C.conj(), self.omt[ia], C).real
        return hd, od


def davidson_fv(engine, ik, kp, nev: int, tol: float = 1e-9,
                num_steps: int = 40, subspace_size: int = 4,
                verbose: bool = False):
    """Block Davidson for H x = e O x in the combined (PW, lo) basis."""
    op = FVOperator(engine, ik, kp)
    N, ngk, nlo = op.N, op.ngk, op.nlo
    hd, od = op.diag()

    # initial guess: previous eigenvectors, else lowest-kinetic PWs
    prev = getattr(kp, "fv_evec", None)
    if prev is not None and prev.shape == (N, nev):
        X = torch.from_numpy(prev.T.copy()).to(torch.complex128)
    else:
        g2 = (op.gk ** 2).sum(-1)
        idx = torch.argsort(g2)[:nev]
        X = torch.zeros(nev, N, dtype=torch.complex128)
        for i, g in enumerate(idx):
            X[i, g] = 1.0
    # fixed extra block: pure local orbitals
    E = torch.zeros(nlo, N, dtype=torch.complex128)
    for j in range(nlo):
        E[j, ngk + j] = 1.0

    max_basis = subspace_size * nev + nlo

    B = torch.cat([X, E], 0)
    HB, OB = op.apply(B)

    eval_prev = None
    for it in range(num_steps):
        nb = B.shape[0]
        Hs = B.conj() @ HB.T
        Os = B.conj() @ OB.T
        Hs = 0.5 * (Hs + Hs.conj().T)
        Os = 0.5 * (Os + Os.conj().T)
        # canonical orthogonalization of the subspace metric
        w_o, v_o = torch.linalg.eigh(Os)
        keep = w_o > max(1e-10, float(w_o.max()) * 1e-12)
        T = v_o[:, keep] / torch.sqrt(w_o[keep])
        Ht = T.conj().T @ Hs.to(T.dtype) @ T
        w, z = torch.linalg.eigh(Ht)
        w = w[:nev]
        Z = (T @ z[:, :nev])                       # [nb, nev]
        X = (Z.T @ B)
        HX = (Z.T @ HB)
        OX = (Z.T @ OB)
        evals = w.numpy().real

        # residuals
        R = HX - torch.from_numpy(evals)[:, None] * OX
        rnorm = R.abs().square().sum(-1).sqrt().numpy().real
        if verbose:
            print(f"  dav it {it} nb {nb} evals {evals[:3]} rmax {rnorm.max():.2e}",
                  file=sys.stderr)
        unconv = [i for i in range(nev) if rnorm[i] > tol]
        de = np.max(np.abs(evals - eval_prev)) if eval_prev is not None else 1.0
        eval_prev = evals
        if not unconv or de < max(tol * 1e-2, 1e-13):
            break

        # precondition unconverged residuals (reference smooth-clamped
        # diagonal: p = (1 + p + sqrt(1 + (p-1)^2))/2, residuals_aux.cu:311)
        newv = []
        for i in unconv:
            p = hd - evals[i] * od
            p = 0.5 * (1.0 + p + torch.sqrt(1.0 + (p - 1.0) ** 2))
            t = (R[i] / p)
            nrm = t.abs().square().sum().sqrt()
            if float(nrm) > 1e-12:
                newv.append(t / nrm)
        if not newv:
            break
        V = torch.stack(newv)
        if nb + V.shape[0] > max_basis:
            # restart: collapse to current Ritz vectors + extra block
            B = torch.cat([X, E], 0)
            HB, OB = op.apply(E)  # recompute only extra (X parts known)
            HB = torch.cat([HX, HB], 0)
            OB = torch.cat([OX, OB], 0)
        HV, OV = op.apply(V)
        B = torch.cat([B, V], 0)
        HB = torch.cat([HB, HV], 0)
        OB = torch.cat([OB, OV], 0)

    kp.fv_eval = evals
    # full eigenvector in (PW, lo) basis
    kp.fv_evec = X.T.numpy().copy()
    kp.eigvals[0, :len(evals)] = evals
    return evals
