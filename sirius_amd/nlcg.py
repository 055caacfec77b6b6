"""Direct total-energy minimization (orbital CG), the robust alternative
to SCF mixing.

Reference behavior: src/nlcglib/ (adaptor to the external Kokkos-based
nlcglib; call_nlcg.hpp:28) and the pure-python sirius.ot / sirius.edft
modules of the reference's python_module — conjugate-gradient
minimization of the total-energy functional over the orthonormality
manifold of the wavefunctions.

This native implementation covers the fixed-occupation (insulating)
case, with the ultrasoft metric when S ≠ I (orthonormality ⟨ψ|S|ψ⟩ = 1):
- gradient  g_k = f ⊙ (1 − S|ψ⟩⟨ψ|) H[ρ] |ψ⟩  per k-point,
- Teter-preconditioned Polak-Ribière CG directions,
- parabolic line search on E(θ) along the tangent direction with
  Löwdin re-orthonormalization,
- the density/potential rebuilt at every functional evaluation (the
  energy is the full self-consistent functional, not a fixed-H model).
"""

from __future__ import annotations

import numpy as np
import torch

from .core import la


def _lowdin(psi: torch.Tensor, spsi: torch.Tensor | None = None) -> torch.Tensor:
    """S-metric Löwdin orthonormalization (S = I when spsi is None)."""
    s = la.inner(psi, spsi if spsi is not None else psi)
    w, v = la.eigh(0.5 * (s + s.conj().T))
    t = (v / torch.sqrt(w.clamp(min=1e-14))) @ v.conj().T
    return t.conj().T @ psi


class DirectMinimizer:
    """Minimize E[{ψ_k}] at fixed integer occupations (no smearing)."""

    def __init__(self, dft, maxiter: int = 100, tol: float = 1e-9):
        self.dft = dft
        self.ctx = dft.ctx
        self.kset = dft.kset
        self.maxiter = maxiter
        self.tol = tol
        if self.ctx.num_spins != 1 or self.ctx.nc_magnetism:
            raise NotImplementedError("direct minimization: nm only")
        # the functional must be evaluated on the raw rho[psi] — the
        # symmetrized density would make E inconsistent with the orbital
        # gradient f·H|psi> (and IBZ k-sets are incomplete for CG)
        if getattr(self.ctx, "symmetry", None) is not None:
            raise ValueError("direct minimization requires use_symmetry: "
                             "false (full k-mesh, unsymmetrized density)")

    def _nocc(self, kp):
        ne = self.ctx.unit_cell.num_electrons
        return int(np.ceil(ne / self.ctx.max_occupancy - 1e-12))

    def _set_occ(self):
        """Aufbau fill at fixed occupations (a fractional last band is
        allowed for odd electron counts)."""
        ne = self.ctx.unit_cell.num_electrons
        mo = self.ctx.max_occupancy
        for kp in self.kset:
            kp.occ[0][:] = 0.0
            left = ne
            for i in range(self._nocc(kp)):
                kp.occ[0][i] = min(mo, left)
                left -= kp.occ[0][i]

    def _apply_s(self, hk, psi):
        if hk.Q is None or hk.bp.num_beta_total == 0:
            return None
        b = hk.bp.inner(psi)
        out = psi.clone().contiguous()
        la.transform(hk.Q @ b, hk.bp.beta_t, out=out, accumulate=True)
        return out

    def _energy(self, psis) -> float:
        """E at given (orthonormal) occupied orbitals."""
        from .hamiltonian import Hamiltonian0

        dft = self.dft
        for kp, psi in zip(self.kset, psis):
            n = psi.shape[0]
            kp.psi[0][:n] = psi
        dft.density.generate(self.kset)
        dft.potential.generate(dft.density)
        dft.potential.generate_paw(dft.density)
        h0 = Hamiltonian0(self.ctx, dft.potential, dft.density)
        self._h0 = h0
        # eval_sum from explicit expectation values
        es = 0.0
        for kp, psi in zip(self.kset, psis):
            hk = h0(kp)
            hpsi, _ = hk.apply_h_s(psi, 0)
            eps = torch.einsum("ig,ig->i", psi.conj(), hpsi).real
            n = psi.shape[0]
            kp.eigvals[0][:n] = eps.cpu().numpy()
            es += kp.weight * float(
                (torch.from_numpy(kp.occ[0][:n]).to(eps.device) * eps).sum())
        self.kset.sync_band()
        self.kset._all_w = self.kset.weights.copy()
        d = dft.total_energy_components()
        etot = (es - d["vxc"] - d["bxc"] - 0.5 * d["vha"] + d["exc"]
                + d["ewald"])
        return etot

    def _grad(self, psis):
        """Projected gradients f(1−P)Hψ and diag preconditioner data."""
        out = []
        for kp, psi in zip(self.kset, psis):
            hk = self._h0(kp)
            hpsi, spsi = hk.apply_h_s(psi, 0)
            ov = la.inner(psi, hpsi)
            # (1 − S|ψ⟩⟨ψ|) H ψ: project with Sψ when S ≠ I
            g = hpsi - la.transform(ov.transpose(0, 1),
                                    spsi if spsi is not None else psi)
            n = psi.shape[0]
            f = torch.from_numpy(kp.occ[0][:n]).to(g.device)
            g = (kp.weight * f)[:, None].to(g.dtype) * g
            # Teter preconditioner on the kinetic energy
            ekin = hk.ekin
            eps = torch.einsum("ig,ig->i", psi.conj(), hpsi).real
            t = ekin[None, :] / eps.abs().clamp(min=1e-3)[:, None]
            p = 0.5 * (1.0 + t + torch.sqrt(1.0 + (t - 1.0) ** 2))
            out.append((g, 1.0 / p))
        return out

    def _lowdin_k(self, kp, psi):
        """Löwdin in the right metric (uses the CURRENT h0's S operator)."""
        if not self.ctx.has_aug:
            return _lowdin(psi)
        hk = self._h0(kp)
        return _lowdin(psi, self._apply_s(hk, psi))

    def run(self) -> dict:
        from .hamiltonian import Hamiltonian0

        self._set_occ()
        # h0 needed for the S metric of the initial orthonormalization
        self._h0 = Hamiltonian0(self.ctx, self.dft.potential,
                                self.dft.density)
        psis = []
        for kp in self.kset:
            n = self._nocc(kp)
            psis.append(self._lowdin_k(kp, kp.psi[0][:n].clone().contiguous()))
        e = self._energy(psis)
        hist = [e]
        dirs = None
        g_prev = None
        converged = False
        for it in range(self.maxiter):
            grads = self._grad(psis)
            gnorm = sum(float((g.conj() * g).sum().real) for g, _ in grads)
            pg = [pc.to(g.dtype) * g for g, pc in grads]
            num = sum(float((p.conj() * g).sum().real)
                      for p, (g, _) in zip(pg, grads))
            if g_prev is not None:
                # Polak-Ribière
                num_pr = sum(float((p.conj() * (g - go)).sum().real)
                             for p, (g, _), go in zip(pg, grads, g_prev))
                beta = max(0.0, num_pr / max(self._den, 1e-300))
            else:
                beta = 0.0
            self._den = num
            if dirs is None:
                dirs = [-p for p in pg]
            else:
                dirs = [-p + beta * d for p, d in zip(pg, dirs)]
            # project directions onto the tangent space
            dirs = [d - la.transform(la.inner(psi, d).transpose(0, 1), psi)
                    for psi, d in zip(psis, dirs)]
            g_prev = [g for g, _ in grads]

            # parabolic line search: E(θ) ≈ e + b θ + c θ²
            b = 2.0 * sum(float((d.conj() * g).sum().real)
                          for d, (g, _) in zip(dirs, grads))
            theta_t = 0.2 / max(1.0, abs(b))
            psis_t = [self._lowdin_k(kp, psi + theta_t * d)
                      for kp, psi, d in zip(self.kset, psis, dirs)]
            e_t = self._energy(psis_t)
            c = (e_t - e - b * theta_t) / theta_t ** 2
            if c > 0:
                theta = max(min(-b / (2 * c), 5 * theta_t), 0.05 * theta_t)
            else:
                theta = theta_t if e_t < e else 0.1 * theta_t
            psis_n = [self._lowdin_k(kp, psi + theta * d)
                      for kp, psi, d in zip(self.kset, psis, dirs)]
            e_n = self._energy(psis_n)
            if e_n > e and e_t < e:
                psis_n, e_n = psis_t, self._energy(psis_t)
            if e_n <= e:
                psis = psis_n
                de = e - e_n
                e = e_n
            else:
                # reject step, restart CG
                dirs = None
                g_prev = None
                de = 0.0
                e = self._energy(psis)
            hist.append(e)
            if it > 2 and abs(de) < self.tol and gnorm < 1e-8:
                converged = True
                break
        # leave kset in a consistent state
        self._energy(psis)
        return {"etot": e, "history": hist, "converged": converged,
                "num_iter": len(hist) - 1}
