"""LAPW muffin-tin radial basis per atom symmetry class.

Reference behavior: src/unit_cell/atom_symmetry_class.cpp —
generate_aw_radial_functions (:63), generate_lo_radial_functions (:161),
find_enu (:463), generate_radial_integrals (:605),
generate_core_charge_density (:765).

Radial functions are stored as two channels, u(r) and r*u'(r); the AW
surface derivatives sd[dm, idxrf] feed the matching-coefficient linear
systems (lapw/matching.py).  The spherical Hamiltonian integrals use the
symmetric-gradient kinetic form

    h_{12} = int [ 1/2 (ru_1')(ru_2') Minv
                   + u_1 u_2 ( l(l+1)/2 Minv + V r^2 ) ] dr / Y00

(exactly the reference's form; Minv = 1 for valence_relativity none,
ZORA mass otherwise).
"""

from __future__ import annotations

import math

import numpy as np
import torch
from scipy.interpolate import CubicSpline

from ..ops import get_radial
from .species import FPAtomType

Y00 = 1.0 / (2.0 * math.sqrt(math.pi))
SPEED_OF_LIGHT = 137.035999139
REL_CODE = {"none": 0, "koelling_harmon": 1, "zora": 2, "iora": 3, "dirac": 4}


def spl_int(r: np.ndarray, f: np.ndarray) -> float | np.ndarray:
    """Exact integral of the cubic-spline interpolant of f over r.
    f may be [..., nr]; integrates the last axis."""
    cs = CubicSpline(r, f, axis=-1)
    return cs.integrate(r[0], r[-1])


class AtomSymmetryClass:
    """Radial basis functions + spherical integrals for one symmetry class."""

    def __init__(self, at: FPAtomType, valence_relativity: str = "none",
                 core_relativity: str = "dirac", auto_enu_tol: float = 0.0):
        self.at = at
        self.rel = valence_relativity
        self.core_rel = core_relativity
        self.auto_enu_tol = auto_enu_tol
        nr = at.nmtp
        nrf = at.num_rf
        self.u = np.zeros((nrf, nr))        # u(r)
        self.rdudr = np.zeros((nrf, nr))    # r u'(r)
        self.sd = np.zeros((3, at.num_aw_rf))  # surface derivs of AW u
        self.h_spherical = np.zeros((nrf, nrf))
        self.o_radial = {}                  # (l, o1, o2) -> float
        self.o1_radial = np.zeros((nrf, nrf)) if self.rel == "iora" else None
        self.vs = None                      # spherical potential (incl. -Z/r)
        self.enu_aw = [[d.enu for d in at.aw_descriptors[l]]
                       for l in range(at.lmax_apw + 1)]
        self.enu_lo = [[b.enu for b in lo.rsd_set] for lo in at.lo_descriptors]
        self.ae_core_density = np.zeros(nr)
        self.core_eval_sum = 0.0
        self.core_leakage = 0.0
        self._rt = torch.from_numpy(np.ascontiguousarray(at.r))

    # -------------------------------------------------------------- potential
    def set_spherical_potential(self, vs: np.ndarray):
        """vs = Y00 * V_00(r): the full spherical potential incl. -Z/r
        (reference set_spherical_potential, potential.cpp:424-437)."""
        self.vs = np.asarray(vs, dtype=np.float64)

    # ------------------------------------------------------------------ enu
    def find_enu(self):
        ext = get_radial()
        rel = REL_CODE[self.rel]
        vt = torch.from_numpy(self.vs)
        for l in range(self.at.lmax_apw + 1):
            for o, d in enumerate(self.at.aw_descriptors[l]):
                if d.auto:
                    new = ext.enu_finder(rel, self.at.zn, d.n, d.l,
                                         self.enu_aw[l][o], d.auto,
                                         self._rt, vt)
                    if abs(new - self.enu_aw[l][o]) > self.auto_enu_tol:
                        self.enu_aw[l][o] = new
        for ilo, lo in enumerate(self.at.lo_descriptors):
            for o, d in enumerate(lo.rsd_set):
                if d.auto:
                    new = ext.enu_finder(rel, self.at.zn, d.n, d.l,
                                         self.enu_lo[ilo][o], d.auto,
                                         self._rt, vt)
                    if abs(new - self.enu_lo[ilo][o]) > self.auto_enu_tol:
                        self.enu_lo[ilo][o] = new

    # ------------------------------------------------------- radial functions
    def generate_radial_functions(self):
        assert self.vs is not None, "set_spherical_potential first"
        self.find_enu()
        self._generate_aw()
        self._generate_lo()
        self._generate_integrals()

    def _solve(self, dme: int, l: int, enu: float):
        ext = get_radial()
        p, rdudr, ud, nn = ext.solve(REL_CODE[self.rel], dme, l, self.at.zn,
                                     enu, self._rt, torch.from_numpy(self.vs))
        return p.numpy(), rdudr.numpy(), ud.numpy()

    def _generate_aw(self):
        at = self.at
        r = at.r
        for l in range(at.lmax_apw + 1):
            ok = False
            for kshift in range(100):
                ps, rduds, sds = [], [], []
                ok = True
                for o, d in enumerate(at.aw_descriptors[l]):
                    p, rdu, ud = self._solve(d.dme, l, self.enu_aw[l][o] + 0.5 * kshift)
                    # orthogonalize to previous orders (in p = r*u space)
                    for o1 in range(o):
                        ovlp = float(spl_int(r, p * ps[o1]))
                        p = p - ps[o1] * ovlp
                        rdu = rdu - rduds[o1] * ovlp
                        ud = ud - sds[o1] * ovlp
                    norm = float(spl_int(r, p * p))
                    if abs(norm) < 1e-8:
                        ok = False
                        break
                    s = 1.0 / math.sqrt(norm)
                    ps.append(p * s)
                    rduds.append(rdu * s)
                    sds.append(ud * s)
                if ok:
                    break
            if not ok:
                raise RuntimeError(f"AW radial functions failed for l={l}")
            for o in range(at.aw_order(l)):
                idxrf = at.rf_index(l, o)
                self.u[idxrf] = ps[o] / r
                self.rdudr[idxrf] = rduds[o]
                self.sd[:, idxrf] = sds[o]

    def _generate_lo(self):
        at = self.at
        r = at.r
        for ilo, lo in enumerate(at.lo_descriptors):
            nrs = len(lo.rsd_set)
            us, rdus, a = [], [], np.zeros((3, 3))
            for o, d in enumerate(lo.rsd_set):
                p, rdu, ud = self._solve(d.dme, lo.l, self.enu_lo[ilo][o])
                us.append(p / r)
                rdus.append(rdu)
                a[:, o] = ud      # a[i, order] = u^{(i)}(R) of solution `order`
            rhs = np.zeros(nrs)
            rhs[nrs - 1] = 1.0
            b = np.linalg.solve(a[:nrs, :nrs], rhs)
            u = sum(b[o] * us[o] for o in range(nrs))
            rdu = sum(b[o] * rdus[o] for o in range(nrs))
            norm = float(spl_int(r, (u * r) ** 2))
            s = 1.0 / math.sqrt(norm)
            idxrf = at.num_aw_rf + ilo
            self.u[idxrf] = u * s
            self.rdudr[idxrf] = rdu * s
            if abs(self.u[idxrf][-1]) > 1e-10:
                import sys
                print(f"[sirius_amd] warning: local orbital {ilo} of "
                      f"{at.label} is not zero at the MT boundary "
                      f"({self.u[idxrf][-1]:.2e})", file=sys.stderr)

    # --------------------------------------------------------------- integrals
    def _generate_integrals(self):
        at = self.at
        r = at.r
        nrf = at.num_rf
        sq_alpha_half = 0.0 if self.rel == "none" else 0.5 / SPEED_OF_LIGHT ** 2
        minv = 1.0 / (1.0 - self.vs * sq_alpha_half)
        vr2 = self.vs * r * r

        self.h_spherical.fill(0.0)
        for i1 in range(nrf):
            l1 = at.indexr[i1][0]
            for i2 in range(nrf):
                if at.indexr[i2][0] != l1:
                    continue
                ll = l1 * (l1 + 1)
                t0 = self.u[i1] * self.u[i2]
                t1 = self.rdudr[i1] * self.rdudr[i2]
                s = 0.5 * t1 * minv + t0 * (0.5 * ll * minv + vr2)
                self.h_spherical[i1, i2] = float(spl_int(r, s)) / Y00

        self.o_radial = {}
        for l, orders in at.max_order_by_l.items():
            for o1 in range(orders):
                i1 = at.rf_index(l, o1)
                for o2 in range(orders):
                    i2 = at.rf_index(l, o2)
                    if o1 == o2:
                        v = 1.0
                    else:
                        v = float(spl_int(r, self.u[i1] * self.u[i2] * r * r))
                    self.o_radial[(l, o1, o2)] = v

        if self.rel == "iora":
            self.o1_radial.fill(0.0)
            minv2 = (1.0 - self.vs * sq_alpha_half) ** -2
            for i1 in range(nrf):
                l1 = at.indexr[i1][0]
                for i2 in range(nrf):
                    if at.indexr[i2][0] != l1:
                        continue
                    ll = l1 * (l1 + 1)
                    t0 = self.u[i1] * self.u[i2]
                    t1 = self.rdudr[i1] * self.rdudr[i2]
                    s = sq_alpha_half * 0.5 * minv2 * (t1 + t0 * ll)
                    self.o1_radial[i1, i2] = float(spl_int(r, s))

    # ------------------------------------------------------------- core states
    def generate_core_charge_density(self):
        """Reference atom_symmetry_class.cpp:765-905: solve atomic-like
        bound states in the spherical MT potential extended with an
        alpha/r + beta tail on an extended grid."""
        at = self.at
        if at.num_core_electrons == 0:
            self.ae_core_density.fill(0.0)
            self.core_eval_sum = 0.0
            self.core_leakage = 0.0
            return
        nmtp = at.nmtp
        grid = list(at.r)
        x = at.r[-1]
        dx = at.r[-1] - at.r[-2]
        while x < 30.0 + at.zn / 4.0:
            x += dx
            grid.append(x)
            dx *= 1.025
        rg = np.array(grid)

        # potential tail: fit alpha/r + beta to the electronic part at R
        svmt = CubicSpline(at.r, self.vs + at.zn / at.r)
        R = at.rmt
        alpha = -(R * R * float(svmt(R, 1)) + at.zn)
        beta = float(svmt(R)) - (at.zn + alpha) / R
        veff = np.empty(len(rg))
        veff[:nmtp] = self.vs
        veff[nmtp:] = alpha / rg[nmtp:] + beta

        ext = get_radial()
        rgt = torch.from_numpy(rg)
        vt = torch.from_numpy(veff)
        rel = REL_CODE[self.core_rel]
        rho = np.zeros(len(rg))
        eval_sum = 0.0
        for (n, l, k, occ, is_core) in at.atomic_levels:
            if not is_core:
                continue
            e_start = -0.5 * at.zn / n ** 2 * 2   # reference: -zn/2/n^2
            enu, p, rho_i = ext.bound_state(rel, at.zn, n, l, k, -at.zn / 2.0 / n ** 2,
                                            rgt, vt, 0.5, 10.0)
            rho += occ * rho_i.numpy() / (4 * math.pi)
            eval_sum += float(enu) * occ
        self.ae_core_density = rho[:nmtp].copy()
        self.core_eval_sum = eval_sum
        self.core_leakage = 4 * math.pi * (float(spl_int(rg, rho * rg * rg)) -
                                           float(spl_int(at.r, self.ae_core_density * at.r ** 2)))
