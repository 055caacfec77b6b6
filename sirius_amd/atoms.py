"""Self-consistent atomic DFT solver (spherical LDA).

Reference behavior: apps/atoms/atom.cpp — solve the free atom with the
radial Kohn-Sham equations: aufbau occupation of (n, l) shells, radial
Poisson for the Hartree potential, LDA XC, linear density mixing. Used
by the reference to generate free-atom densities and starting guesses;
here it exercises the radial solver and provides the same standalone
capability.

Equations (spherical symmetry, Hartree units):
    V_H(r)  = (1/r)∫_0^r 4π s² ρ(s) ds + ∫_r^∞ 4π s ρ(s) ds
    ρ(r)    = Σ_nl occ_nl |R_nl(r)|² / 4π
    E_tot   = Σ occ ε − ∫(V_H/2 + V_xc)ρ 4π r² dr + E_xc
"""

from __future__ import annotations

import numpy as np

from .core.radial_solver import bound_states


# aufbau order: (n, l) by n+l then n
_AUFBAU = [(1, 0), (2, 0), (2, 1), (3, 0), (3, 1), (4, 0), (3, 2), (4, 1),
           (5, 0), (4, 2), (5, 1), (6, 0), (4, 3), (5, 2), (6, 1), (7, 0),
           (5, 3), (6, 2), (7, 1)]


def aufbau_occupations(zn: int):
    """[(n, l, occ)] filling zn electrons."""
    out = []
    left = zn
    for (n, l) in _AUFBAU:
        if left <= 0:
            break
        cap = 2 * (2 * l + 1)
        occ = min(cap, left)
        out.append((n, l, float(occ)))
        left -= occ
    return out


def _hartree(r, rho):
    """V_H for spherical ρ(r) via the two cumulative integrals."""
    from scipy.integrate import cumulative_trapezoid

    q_in = cumulative_trapezoid(4 * np.pi * rho * r * r, r, initial=0.0)
    q_out = cumulative_trapezoid(4 * np.pi * rho * r, r, initial=0.0)
    return q_in / np.maximum(r, 1e-30) + (q_out[-1] - q_out)


def solve_atom(zn: int, rmax: float = 30.0, nr: int = 1200,
               beta: float = 0.35, maxiter: int = 120, tol: float = 1e-8):
    """Self-consistent spherical LDA (PZ) atom.

    Returns dict with energies ε_nl, total energy, ρ(r), grid.
    """
    import torch

    from . import xc as xc_mod

    r = np.geomspace(1e-6, rmax, nr)
    shells = aufbau_occupations(zn)
    lmax = max(l for _, l, _ in shells)
    rho = zn * np.exp(-2.0 * r * zn ** (1 / 3)) \
        * (zn ** 2 / np.pi)                    # crude start
    nrm = np.trapezoid(4 * np.pi * rho * r * r, r)
    rho *= zn / nrm
    e_old = 0.0
    out = {}
    for it in range(maxiter):
        vh = _hartree(r, rho)
        eps_xc, vxc, _ = xc_mod.evaluate(
            ["XC_LDA_X", "XC_LDA_C_PZ"],
            torch.from_numpy(np.maximum(rho, 1e-30)))
        vxc = vxc.numpy()
        eps_xc = eps_xc.numpy()
        v = -zn / r + vh + vxc
        levels = {}
        rho_new = np.zeros_like(rho)
        e_sum = 0.0
        for l in range(lmax + 1):
            want = [(n, occ) for (n, ll, occ) in shells if ll == l]
            if not want:
                continue
            nst = max(n - l for n, _ in want)
            e, R = bound_states(r, v, l, nstates=nst)
            for (n, occ) in want:
                i = n - l - 1
                levels[(n, l)] = float(e[i])
                e_sum += occ * e[i]
                rho_new += occ * R[i] ** 2 / (4 * np.pi)
        # total energy: eval sum − double counting
        dc = np.trapezoid(4 * np.pi * r * r * rho_new
                          * (0.5 * vh + vxc), r)
        exc = np.trapezoid(4 * np.pi * r * r * rho_new * eps_xc, r)
        # note: vh/vxc belong to the OLD density; at convergence exact
        etot = e_sum - dc + exc
        drho = np.trapezoid(4 * np.pi * r * r * np.abs(rho_new - rho), r)
        rho = (1 - beta) * rho + beta * rho_new
        if abs(etot - e_old) < tol and drho < 1e-6:
            out["converged"] = True
            break
        e_old = etot
    out.update({"etot": etot, "levels": levels, "rho": rho, "r": r,
                "num_iter": it + 1})
    out.setdefault("converged", False)
    return out
