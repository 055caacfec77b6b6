"""Bound states of the radial Schrödinger equation.

Reference behavior: src/radial/radial_solver.hpp:786 (Radial_solver —
bound-state and band-energy solutions of the radial (scalar-)
relativistic equations on the muffin-tin grids; used by the FP-LAPW
branch for radial basis functions and by the atomic solver app).

This native implementation solves the non-relativistic radial equation

    −½ u''(r) + [V(r) + l(l+1)/2r²] u(r) = E u(r),  u = r·R

as a standard symmetric tridiagonal eigenproblem on a uniform r grid
(u(0) = u(r_max) = 0), solved with LAPACK's stebz/stein through
scipy.eigh_tridiagonal for the lowest states. (A log-grid generalized
form r²·B conditions catastrophically over 14 decades of r; the
uniform-grid operator with Richardson-sized steps is accurate to ~1e-6
Ha for valence-like states. The scalar-relativistic variants and the
log-grid Numerov shooting for deep cores are round-2 work for the LAPW
branch.)
"""

from __future__ import annotations

import numpy as np


def bound_states(r: np.ndarray, v: np.ndarray, l: int, nstates: int = 5,
                 npts: int = 12000):
    """Lowest `nstates` bound levels (E < 0 not required) and radial
    functions R_nl(r) for potential v on grid r.

    Returns (energies [nstates], R [nstates, len(r)]), ordered by energy;
    R normalized to ∫R² r² dr = 1.
    """
    from scipy.interpolate import CubicSpline
    from scipy.linalg import eigh_tridiagonal

    r = np.asarray(r, dtype=np.float64)
    v = np.asarray(v, dtype=np.float64)
    rmax = r[-1]
    npts = max(npts, 4000)
    ru = np.linspace(0.0, rmax, npts + 2)[1:-1]   # interior points
    h = ru[1] - ru[0]
    vs = CubicSpline(r, v)(np.clip(ru, r[0], rmax))
    # near the origin extend with the Coulombic form v ~ v(r0)·r0/r
    inner = ru < r[0]
    if inner.any():
        vs[inner] = v[0] * r[0] / ru[inner]
    diag = 1.0 / h**2 + vs + 0.5 * l * (l + 1) / ru**2
    off = np.full(npts - 1, -0.5 / h**2)
    w, vecs = eigh_tridiagonal(diag, off, select="i",
                               select_range=(0, nstates - 1))
    e = w[:nstates]
    out = np.zeros((nstates, len(r)))
    for i in range(nstates):
        u = vecs[:, i]
        nrm = np.trapezoid(u * u, ru)
        u = u / np.sqrt(abs(nrm))
        k = np.argmax(np.abs(u[: npts // 4]))
        if u[k] < 0:
            u = -u
        R = np.zeros_like(ru)
        R = u / ru
        out[i] = CubicSpline(ru, R)(np.clip(r, ru[0], rmax))
    return e, out


def hydrogenic_levels(zn: float, l: int, n_levels: int = 4) -> np.ndarray:
    """Exact Coulomb levels −Z²/2n² for n = l+1, … (test reference)."""
    n = np.arange(l + 1, l + 1 + n_levels)
    return -zn**2 / (2.0 * n.astype(np.float64) ** 2)


SPEED_OF_LIGHT = 137.035999084   # Hartree atomic units


def bound_state_sr(r: np.ndarray, v: np.ndarray, n: int, l: int,
                   enu: float | None = None, rel: bool = True,
                   tol: float = 1e-10):
    """One bound state (n, l) of the scalar-relativistic (Koelling-Harmon)
    radial equation by outward RK4 shooting with node-count bisection
    (reference: Radial_solver::integrate_forward_rk4 + the bound-state
    search of src/radial/radial_solver.hpp:786-990).

        u'' = [l(l+1)/r² + 2M(V−E)] u + (M'/M)(u' − u/r),
        M   = 1 + (E − V)/2c²      (M ≡ 1 for rel=False)

    Returns (E, R on r) with ∫R²r²dr = 1. The target state has
    n − l − 1 radial nodes.
    """
    from scipy.interpolate import CubicSpline

    r = np.asarray(r, dtype=np.float64)
    v = np.asarray(v, dtype=np.float64)
    vs = CubicSpline(r, v)
    c2 = SPEED_OF_LIGHT ** 2 if rel else 1e30
    nodes_target = n - l - 1

    # integration grid: log-spaced, dense
    t = np.linspace(np.log(max(r[0], 1e-8)), np.log(r[-1]), 6000)
    rg = np.exp(t)
    vg = vs(rg)
    dvg = vs(rg, 1)

    def shoot(E):
        """integrate u outward; return (nodes, u, u(rmax) sign function)."""
        M = 1.0 + (E - vg) / (2 * c2)
        dM = -dvg / (2 * c2)
        A = l * (l + 1) / rg**2 + 2.0 * M * (vg - E)
        B = dM / M
        u = np.zeros_like(rg)
        up = np.zeros_like(rg)
        u[0] = rg[0] ** (l + 1)
        up[0] = (l + 1) * rg[0] ** l

        def rhs(i, frac, ui, upi):
            # linear interpolation of coefficients inside the step
            a = A[i] * (1 - frac) + A[min(i + 1, len(rg) - 1)] * frac
            b = B[i] * (1 - frac) + B[min(i + 1, len(rg) - 1)] * frac
            rr = rg[i] * (1 - frac) + rg[min(i + 1, len(rg) - 1)] * frac
            return upi, a * ui + b * (upi - ui / rr)

        nodes = 0
        scale = 0.0
        for i in range(len(rg) - 1):
            h = rg[i + 1] - rg[i]
            k1u, k1p = rhs(i, 0.0, u[i], up[i])
            k2u, k2p = rhs(i, 0.5, u[i] + 0.5 * h * k1u, up[i] + 0.5 * h * k1p)
            k3u, k3p = rhs(i, 0.5, u[i] + 0.5 * h * k2u, up[i] + 0.5 * h * k2p)
            k4u, k4p = rhs(i, 1.0, u[i] + h * k3u, up[i] + h * k3p)
            u[i + 1] = u[i] + h / 6 * (k1u + 2 * k2u + 2 * k3u + k4u)
            up[i + 1] = up[i] + h / 6 * (k1p + 2 * k2p + 2 * k3p + k4p)
            if u[i + 1] * u[i] < 0:
                nodes += 1
            m = abs(u[i + 1])
            if m > 1e20:             # renormalize both to avoid overflow
                u[: i + 2] /= m
                up[: i + 2] /= m
                # (node count unaffected: uniform positive scaling)
        return nodes, u

    # bracket E by node count
    elo = (max(vg.min(), -50.0 * (abs(v[np.searchsorted(r, 0.5)]) + 1.0))
           if enu is None else enu - 5.0)
    ehi = 10.0
    for _ in range(200):
        em = 0.5 * (elo + ehi)
        nodes, u = shoot(em)
        if nodes > nodes_target:
            ehi = em
        elif nodes < nodes_target:
            elo = em
        else:
            # right node count: tail sign decides (decaying solution has
            # u(rmax) -> 0; too-high E adds a node / flips the tail)
            if u[-1] * u[np.argmax(np.abs(u))] > 0:
                elo = em
            else:
                ehi = em
        if ehi - elo < tol:
            break
    E = 0.5 * (elo + ehi)
    _, u = shoot(E)
    R = u / rg
    nrm = np.trapezoid(u * u, rg)
    R = R / np.sqrt(abs(nrm))
    if R[np.argmax(np.abs(R[:2000]))] < 0:
        R = -R
    Rr = CubicSpline(rg, R)(np.clip(r, rg[0], rg[-1]))
    return E, Rr
