"""Structural relaxation (fixed-cell and variable-cell).

Reference behavior: src/dft/lattice_relaxation.hpp:24 (Lattice_relaxation
driver) built on the stabilized quasi-Newton optimizer of
src/vcsqnm/periodic_optimizer.hpp:21 (vc-SQNM: Gubler, Schaefer,
Goedecker — history-based significant-subspace curvature estimates with
gain-controlled step size; the variable-cell mode augments the atomic
coordinates with three scaled lattice vectors driven by the stress).

This module implements the same algorithm natively:
- `SQNM`: the optimizer on a flat coordinate vector;
- `LatticeRelaxation`: SCF → forces (→ stress) → SQNM step → new cell,
  with the converged density/wavefunctions NOT carried over (each
  geometry starts from the atomic superposition — simple and robust).

Convergence: max|F| ≤ forces_thr (and max|σ| ≤ stress_thr for vc),
mirroring lattice_relaxation.hpp:44-76.
"""

from __future__ import annotations


import numpy as np


class SQNM:
    """Stabilized quasi-Newton minimizer (periodic_optimizer.hpp behavior).

    step(x, f, grad) -> new x. Gain-controlled step size: the step length
    grows 1.05× when the new gradient still points along the previous
    step (cosine > 0.2) and shrinks 2× otherwise; curvature is estimated
    in the significant subspace of the displacement history.
    """

    def __init__(self, n: int, alpha: float = 1.0, nhist_max: int = 10,
                 eps_subsp: float = 1e-4):
        self.alpha = alpha
        self.nhist_max = nhist_max
        self.eps_subsp = eps_subsp
        self.x_hist: list[np.ndarray] = []
        self.g_hist: list[np.ndarray] = []
        self.prev_dir = None

    def step(self, x: np.ndarray, grad: np.ndarray) -> np.ndarray:
        x = np.asarray(x, dtype=np.float64).copy()
        g = np.asarray(grad, dtype=np.float64).copy()
        # gain control
        if self.prev_dir is not None:
            denom = np.linalg.norm(g) * np.linalg.norm(self.prev_dir)
            cosa = float(g @ self.prev_dir) / denom if denom > 0 else 0.0
            # prev_dir is the step we took = mostly -alpha*g_prev; moving
            # downhill means the new gradient should be roughly orthogonal
            # or still opposed to the step. If g·step > 0 we overshot.
            if cosa > 0.2:
                self.alpha /= 2.0
            else:
                self.alpha = min(self.alpha * 1.05, 10.0)
        self.x_hist.append(x.copy())
        self.g_hist.append(g.copy())
        if len(self.x_hist) > self.nhist_max + 1:
            self.x_hist.pop(0)
            self.g_hist.pop(0)

        nh = len(self.x_hist) - 1
        dx_pred = -self.alpha * g
        if nh >= 1:
            # displacement/gradient difference history, normalized
            dX = np.stack([self.x_hist[i + 1] - self.x_hist[i]
                           for i in range(nh)], axis=1)     # [n, nh]
            dG = np.stack([self.g_hist[i + 1] - self.g_hist[i]
                           for i in range(nh)], axis=1)
            norms = np.linalg.norm(dX, axis=0)
            ok = norms > 1e-14
            dX, dG = dX[:, ok] / norms[ok], dG[:, ok] / norms[ok]
            if dX.shape[1]:
                # significant subspace of the displacement overlap
                S = dX.T @ dX
                w, v = np.linalg.eigh(S)
                keep = w > self.eps_subsp * w.max()
                if keep.any():
                    basis = dX @ (v[:, keep] / np.sqrt(w[keep]))  # [n, m]
                    dGs = dG @ (v[:, keep] / np.sqrt(w[keep]))
                    # curvature (symmetrized projected Hessian)
                    H = 0.5 * (basis.T @ dGs + dGs.T @ basis)
                    hw, hv = np.linalg.eigh(H)
                    hw = np.maximum(np.abs(hw), 1e-10)
                    gs = basis.T @ g                          # subspace grad
                    gs_h = hv.T @ gs
                    step_s = basis @ (hv @ (gs_h / hw))
                    g_perp = g - basis @ (basis.T @ g)
                    dx_pred = -(step_s + self.alpha * g_perp)
        # trust-region clip
        mx = np.abs(dx_pred).max()
        if mx > 0.5:
            dx_pred *= 0.5 / mx
        self.prev_dir = dx_pred.copy()
        return x + dx_pred


class LatticeRelaxation:
    """relax / vc-relax driver (Lattice_relaxation::find)."""

    def __init__(self, cfg, base_dir: str = ".", device=None,
                 variable_cell: bool = False, forces_thr: float = 1e-4,
                 stress_thr: float = 1e-5, max_steps: int = 30,
                 num_scf_iter: int | None = None):
        self.cfg = cfg
        self.base_dir = base_dir
        self.device = device
        self.vc = variable_cell
        self.forces_thr = forces_thr
        self.stress_thr = stress_thr
        self.max_steps = max_steps
        self.num_scf_iter = num_scf_iter
        self._cell0 = None

    def _scf(self, lattice, pos_frac):
        from .cell import UnitCell
        from .context import SimulationContext
        from .kpoint import KPointSet
        from .dft import DFTGroundState

        if self._cell0 is None:
            self._cell0 = UnitCell.from_config(self.cfg, self.base_dir)
        c0 = self._cell0
        if lattice is None:
            cell = c0
        else:
            cell = UnitCell(np.asarray(lattice),
                            c0.atom_types,
                            [(c0.atoms[i][0], np.asarray(pos_frac[i]))
                             for i in range(c0.num_atoms)])
            cell.vector_fields = c0.vector_fields.copy()
        ctx = SimulationContext(self.cfg, unit_cell=cell,
                                base_dir=self.base_dir, device=self.device)
        kset = KPointSet(ctx)
        dft = DFTGroundState(kset).initial_state()
        res = dft.find(num_dft_iter=self.num_scf_iter)
        return ctx, dft, res

    def run(self) -> dict:
        ctx, dft, res = self._scf(None, None)
        uc = ctx.unit_cell
        na = uc.num_atoms
        lattice = uc.lattice.copy()
        pos = uc.atom_positions_frac().copy()
        ndof = 3 * na + (9 if self.vc else 0)
        opt = SQNM(ndof)
        history = []
        converged = False
        for it in range(self.max_steps):
            f = dft.forces()
            ftot = f["total"]                      # Cartesian [na, 3]
            fmax = float(np.abs(ftot).max())
            entry = {"step": it, "etot": res["energy"]["total"],
                     "fmax": fmax}
            smax = None
            if self.vc:
                st = dft.stress()["total"]
                smax = float(np.abs(st).max())
                entry["smax"] = smax
            history.append(entry)
            if fmax < self.forces_thr and (not self.vc
                                           or smax < self.stress_thr):
                converged = True
                break
            # gradient w.r.t. Cartesian positions is −F; pack DOFs
            gx = (-ftot).reshape(-1)
            xx = (pos @ lattice).reshape(-1)       # Cartesian positions
            if self.vc:
                # dE/dh = Ω h^{-T} σ (strain chain rule, row-vector lattice)
                omega = abs(np.linalg.det(lattice))
                gh = (omega * (np.linalg.inv(lattice.T) @ st)).reshape(-1)
                x = np.concatenate([xx, lattice.reshape(-1)])
                gr = np.concatenate([gx, gh])
            else:
                x, gr = xx, gx
            xn = opt.step(x, gr)
            cart = xn[:3 * na].reshape(na, 3)
            if self.vc:
                lattice = xn[3 * na:].reshape(3, 3)
            pos = cart @ np.linalg.inv(lattice)
            ctx, dft, res = self._scf(lattice, pos)
        return {
            "converged": converged,
            "history": history,
            "lattice": lattice.tolist(),
            "positions_frac": pos.tolist(),
            "energy": res["energy"],
        }
