"""PAW: on-site all-electron / pseudo corrections.

Reference behavior:
- on-site densities (density.cpp:506-573): per PAW atom
  ρ^{ae/ps}_{lm3}(r) = Σ_{ξ1≤ξ2} w·dm(ξ1ξ2,ch)·G^{rrr}_{lm1,lm2,lm3}
                       ·[φ_i φ_j (+ Q^{l3}_ij for ps)]/r²
- XC on radial×angular grids with core added to the l=0 channel
  (paw_potential.cpp:118-158 xc_mt_paw; angular sampling like xc_mt);
- Hartree per lm with free-atom boundary conditions, electron-only
  (potential.hpp:296-383 poisson_vmt<true>):
  V_lm(r) = 4π/(2l+1)[g1(r)/r^{l+1} + (g2(∞)−g2(r))·r^l]
- D_ij corrections (paw_potential.cpp:216-300):
  ΔD_ij = Σ_lm3 G_{lm1,lm2,lm3} ∫[V^{ae}_{lm3}φφ − V^{ps}_{lm3}(φφ+Q)]dr
- energies (potential.hpp:656-706): E_PAW = E^ae_H − E^ps_H +
  ∫exc^ae ρ^ae − ∫exc^ps ρ^ps (valence) + core XC part;
  one-elec = Σ dm·ΔD.

GGA on the sphere uses the divergence-by-parts identity:
  ⟨R_lm|∇·F⟩_Ω = (1/r²)∂_r(r² F_r,lm) − (1/r)∫(∂_θR_lm F_θ + ∂_φR_lm/sinθ F_φ)dΩ
with ∂θ/∂φ tables of real harmonics (high-order finite differences).
"""

from __future__ import annotations

import math

import numpy as np
from scipy.interpolate import CubicSpline

from .core import ylm as ylm_mod
from .core.gaunt import gaunt_rrr

Y00 = 1.0 / math.sqrt(4.0 * math.pi)


class AngularGrid:
    """Gauss-Legendre × uniform-φ quadrature with R_lm and derivative tables."""

    def __init__(self, lmax: int):
        self.lmax = lmax
        self.lmmax = ylm_mod.lmmax(lmax)
        ltot = 3 * lmax + 2
        nth = ltot + 4          # oversampled: XC is nonlinear on the sphere
        nph = 2 * ltot + 4
        x, wx = np.polynomial.legendre.leggauss(nth)
        theta = np.arccos(x)
        phi = np.arange(nph) * 2 * math.pi / nph
        tt, pp = np.meshgrid(theta, phi, indexing="ij")
        self.w = np.broadcast_to(wx[:, None] * (2 * math.pi / nph),
                                 tt.shape).reshape(-1).copy()
        self.theta = tt.reshape(-1)
        self.phi = pp.reshape(-1)
        self.npt = len(self.theta)
        self.R = ylm_mod.rlm(lmax, self.theta, self.phi)          # [npt, lmmax]
        # derivative tables by 5-point finite differences
        h = 1e-5

        def d(dth, dph):
            return ylm_mod.rlm(lmax, self.theta + dth, self.phi + dph)

        self.dR_dth = (8 * (d(h, 0) - d(-h, 0)) - (d(2 * h, 0) - d(-2 * h, 0))) / (12 * h)
        dR_dph = (8 * (d(0, h) - d(0, -h)) - (d(0, 2 * h) - d(0, -2 * h))) / (12 * h)
        st = np.sin(self.theta)
        st = np.where(np.abs(st) < 1e-9, 1e-9, st)
        self.dR_dph_sin = dR_dph / st[:, None]

    def to_points(self, f_lm: np.ndarray) -> np.ndarray:
        """[lmmax, nr] -> [npt, nr]."""
        return self.R @ f_lm

    def to_lm(self, f_pt: np.ndarray) -> np.ndarray:
        """[npt, nr] -> [lmmax, nr] via quadrature (orthonormality)."""
        return (self.R * self.w[:, None]).T @ f_pt


def radial_derivative(r: np.ndarray, f: np.ndarray, axis=-1) -> np.ndarray:
    """df/dr via cubic spline (batched over leading dims)."""
    cs = CubicSpline(r, f, axis=axis)
    return cs(r, 1)


def hartree_vmt_free(r: np.ndarray, rho_lm: np.ndarray, lmax: int) -> np.ndarray:
    """Electron Hartree V_lm(r) with free-atom BC (poisson_vmt<true>).

    rho_lm: [lmmax, nr]. Returns v_lm [lmmax, nr].
    """
    lmmax, nr = rho_lm.shape
    v = np.zeros_like(rho_lm)
    l_of = np.concatenate([[l] * (2 * l + 1) for l in range(lmax + 1)])
    for lm in range(lmmax):
        l = int(l_of[lm])
        f1 = rho_lm[lm] * r ** (l + 2)
        f2 = rho_lm[lm] * r ** (1 - l)
        g1 = CubicSpline(r, f1).antiderivative()(r)
        g2 = CubicSpline(r, f2).antiderivative()(r)
        fact = 4 * math.pi / (2 * l + 1)
        v[lm] = fact * (g1 / r ** (l + 1) + (g2[-1] - g2) * r ** l)
    return v


class PAWAtomData:
    """Per-type PAW tables on the (truncated-to-full) radial grid."""

    def __init__(self, at):
        self.at = at
        nr = len(at.r)
        ncut = min(at.paw_cutoff_index or nr, nr)
        self.ncut = ncut
        nbrf = at.num_beta
        self.ae_wf = np.zeros((nbrf, nr))
        self.ps_wf = np.zeros((nbrf, nr))
        for i in range(nbrf):
            self.ae_wf[i, :len(at.paw_ae_wfs[i][:ncut])] = at.paw_ae_wfs[i][:ncut]
            self.ps_wf[i, :len(at.paw_ps_wfs[i][:ncut])] = at.paw_ps_wfs[i][:ncut]
        self.ae_core = np.asarray(at.paw_ae_core, dtype=np.float64)
        self.ps_core = np.asarray(at.rho_core_r, dtype=np.float64)
        # q radial functions lookup (file stores r^2 Q)
        self.qmap = {}
        for q in at.q_radial:
            i, j = min(q.i, q.j), max(q.i, q.j)
            self.qmap[(i, j, q.l)] = np.asarray(q.f_r)


class PAWModule:
    """PAW on-site machinery for all PAW atoms."""

    def __init__(self, ctx):
        self.ctx = ctx
        uc = ctx.unit_cell
        self.types = {}
        self.lmax_b = {}
        for lab, at in uc.atom_types.items():
            if at.is_paw:
                self.types[lab] = PAWAtomData(at)
                self.lmax_b[lab] = max((b.l for b in at.beta), default=0)
        self.paw_atoms = [ia for ia, (lab, _) in enumerate(uc.atoms)
                          if lab in self.types]
        self._ang = {}
        self._gc = {}
        # per-atom results
        self.dij = {}                  # ia -> [nbf, nbf, ncomp]
        self.hartree_energy = 0.0
        self.xc_energy = 0.0

    def ang(self, lab) -> AngularGrid:
        l2 = 2 * self.lmax_b[lab]
        if l2 not in self._ang:
            self._ang[l2] = AngularGrid(l2)
        return self._ang[l2]

    def gc(self, lab) -> np.ndarray:
        lb = self.lmax_b[lab]
        key = lb
        if key not in self._gc:
            self._gc[key] = gaunt_rrr(lb, lb, 2 * lb)
        return self._gc[key]

    # -- on-site densities -------------------------------------------------

    def onsite_density(self, lab: str, dm_aux: np.ndarray):
        """dm_aux [nqlm, ncomp] (no sym weights) → (ae_lm, ps_lm) each
        [ncomp, lmmax, nr]."""
        at = self.ctx.unit_cell.atom_types[lab]
        pd = self.types[lab]
        r = at.r
        idxb = at.beta_lm_index()
        nbf = len(idxb)
        gc = self.gc(lab)
        lmmax = ylm_mod.lmmax(2 * self.lmax_b[lab])
        l_of3 = np.concatenate([[l] * (2 * l + 1)
                                for l in range(2 * self.lmax_b[lab] + 1)])
        ncomp = dm_aux.shape[1]
        nr = len(r)
        ae = np.zeros((ncomp, lmmax, nr))
        ps = np.zeros((ncomp, lmmax, nr))
        inv_r2 = 1.0 / np.maximum(r, 1e-30) ** 2
        for xi2 in range(nbf):
            rf2, l2, m2 = idxb[xi2]
            lm2 = ylm_mod.lm_index(l2, m2)
            for xi1 in range(xi2 + 1):
                rf1, l1, m1 = idxb[xi1]
                lm1 = ylm_mod.lm_index(l1, m1)
                idx12 = xi2 * (xi2 + 1) // 2 + xi1
                w = 1.0 if xi1 == xi2 else 2.0
                pa, pb = min(rf1, rf2), max(rf1, rf2)
                ae_rad = pd.ae_wf[rf1] * pd.ae_wf[rf2] * inv_r2
                ps_rad0 = pd.ps_wf[rf1] * pd.ps_wf[rf2] * inv_r2
                for lm3 in range(lmmax):
                    g = gc[lm1, lm2, lm3]
                    if abs(g) < 1e-14:
                        continue
                    l3 = int(l_of3[lm3])
                    qf = self.qmap_get(pd, pa, pb, l3)
                    ps_rad = ps_rad0 + (qf * inv_r2 if qf is not None else 0.0)
                    for ic in range(ncomp):
                        c = w * g * dm_aux[idx12, ic]
                        ae[ic, lm3] += c * ae_rad
                        ps[ic, lm3] += c * ps_rad
        return ae, ps

    @staticmethod
    def qmap_get(pd, i, j, l):
        return pd.qmap.get((i, j, l))

    # -- XC on the sphere --------------------------------------------------

    def xc_mt(self, lab: str, rho_lm: np.ndarray, core: np.ndarray):
        """rho_lm [ncomp, lmmax, nr] valence on-site density (+core added to
        l=0). Returns (vxc_lm [ncomp, lmmax, nr], exc_lm [lmmax, nr])."""
        from . import xc as xc_mod
        import torch

        ctx = self.ctx
        at = ctx.unit_cell.atom_types[lab]
        r = at.r
        ang = self.ang(lab)
        ncomp = rho_lm.shape[0]
        rho0 = rho_lm[0].copy()
        rho0[0] += core / Y00

        def to_t(x):
            return torch.from_numpy(np.ascontiguousarray(x))

        rr = np.maximum(r, 1e-12)
        if ctx.is_gga:
            def grads(f_lm):
                fr = ang.to_points(radial_derivative(r, f_lm))        # [npt,nr]
                fth = (ang.dR_dth @ f_lm) / rr[None, :]
                fph = (ang.dR_dph_sin @ f_lm) / rr[None, :]
                return fr, fth, fph
        if ncomp == 1:
            n_pt = ang.to_points(rho0)
            if ctx.is_gga:
                gr, gt, gp = grads(rho0)
                sigma = gr**2 + gt**2 + gp**2
                eps, vrho, vsig = xc_mod.evaluate(
                    ctx.xc_names, to_t(n_pt), to_t(sigma))
                eps, vrho, vsig = eps.numpy(), vrho.numpy(), vsig.numpy()
                v_lm = ang.to_lm(vrho) - self._div_by_parts(
                    ang, r, 2 * vsig * gr, 2 * vsig * gt, 2 * vsig * gp)
            else:
                eps, vrho, _ = xc_mod.evaluate(ctx.xc_names, to_t(n_pt))
                eps, vrho = eps.numpy(), vrho.numpy()
                v_lm = ang.to_lm(vrho)
            return v_lm[None, ...], ang.to_lm(eps)
        else:
            m_lm = rho_lm[1]
            up_lm = 0.5 * (rho0 + m_lm)
            dn_lm = 0.5 * (rho0 - m_lm)
            nu = ang.to_points(up_lm)
            nd = ang.to_points(dn_lm)
            if ctx.is_gga:
                gur, gut, gup = grads(up_lm)
                gdr, gdt, gdp = grads(dn_lm)
                s_uu = gur**2 + gut**2 + gup**2
                s_dd = gdr**2 + gdt**2 + gdp**2
                gtr, gtt, gtp = gur + gdr, gut + gdt, gup + gdp
                s_tot = gtr**2 + gtt**2 + gtp**2
                eps, vu, vd, vs_uu, vs_dd, vs_tot = xc_mod.evaluate_spin(
                    ctx.xc_names, to_t(nu), to_t(nd), to_t(s_uu), to_t(s_dd),
                    to_t(s_tot))
                eps, vu, vd = eps.numpy(), vu.numpy(), vd.numpy()
                vs_uu, vs_dd, vs_tot = (vs_uu.numpy(), vs_dd.numpy(),
                                        vs_tot.numpy())
                vu_lm = ang.to_lm(vu) - self._div_by_parts(
                    ang, r, 2 * vs_uu * gur + 2 * vs_tot * gtr,
                    2 * vs_uu * gut + 2 * vs_tot * gtt,
                    2 * vs_uu * gup + 2 * vs_tot * gtp)
                vd_lm = ang.to_lm(vd) - self._div_by_parts(
                    ang, r, 2 * vs_dd * gdr + 2 * vs_tot * gtr,
                    2 * vs_dd * gdt + 2 * vs_tot * gtt,
                    2 * vs_dd * gdp + 2 * vs_tot * gtp)
            else:
                eps, vu, vd, *_ = xc_mod.evaluate_spin(ctx.xc_names, to_t(nu),
                                                       to_t(nd))
                eps, vu, vd = eps.numpy(), vu.numpy(), vd.numpy()
                vu_lm = ang.to_lm(vu)
                vd_lm = ang.to_lm(vd)
            v_lm = np.stack([0.5 * (vu_lm + vd_lm), 0.5 * (vu_lm - vd_lm)])
            return v_lm, ang.to_lm(eps)

    @staticmethod
    def _div_by_parts(ang: AngularGrid, r, Fr, Fth, Fph):
        """⟨R_lm|∇·F⟩ per (lm, r) via integration by parts on the sphere."""
        rr = np.maximum(r, 1e-12)
        Fr_lm = ang.to_lm(Fr)
        term_r = radial_derivative(r, rr[None, :] ** 2 * Fr_lm) / rr[None, :] ** 2
        ang_term = ((ang.dR_dth * ang.w[:, None]).T @ Fth
                    + (ang.dR_dph_sin * ang.w[:, None]).T @ Fph) / rr[None, :]
        return term_r - ang_term

    # -- main entry: generate PAW potential & Dij --------------------------

    def generate(self, density):
        """Compute per-atom vxc/vha, Dij corrections and PAW energies from
        the (symmetrized) density matrix."""
        ctx = self.ctx
        uc = ctx.unit_cell
        ncomp = ctx.num_mag_dims + 1
        self.hartree_energy = 0.0
        self.xc_energy = 0.0
        self.dij = {}
        if density.density_matrix is None:
            return
        for lab, pd in self.types.items():
            at = uc.atom_types[lab]
            r = at.r
            ia_list = list(uc.atoms_of_type(lab))
            d = density.density_matrix[lab]       # [na, nbf, nbf, nspin]
            idxb = at.beta_lm_index()
            nbf = len(idxb)
            iu1, iu2 = [], []
            for xi2 in range(nbf):
                for xi1 in range(xi2 + 1):
                    iu1.append(xi1)
                    iu2.append(xi2)
            for i, ia in enumerate(ia_list):
                dmc = d[i].cpu().numpy()
                # dm_aux channels (density_matrix_aux)
                if ncomp == 1:
                    dm_aux = dmc[iu2, iu1, 0].real[:, None]
                else:
                    dm_aux = np.stack([
                        (dmc[iu2, iu1, 0] + dmc[iu2, iu1, 1]).real,
                        (dmc[iu2, iu1, 0] - dmc[iu2, iu1, 1]).real], axis=1)
                ae, ps = self.onsite_density(lab, dm_aux)
                # XC
                vxc_ae, exc_ae = self.xc_mt(lab, ae, pd.ae_core)
                vxc_ps, exc_ps = self.xc_mt(lab, ps, pd.ps_core)
                # Hartree (electron only, free-atom BC)
                lmax2 = 2 * self.lmax_b[lab]
                vha_ae = hartree_vmt_free(r, ae[0], lmax2)
                vha_ps = hartree_vmt_free(r, ps[0], lmax2)
                v_ae = vxc_ae.copy()
                v_ae[0] += vha_ae
                v_ps = vxc_ps.copy()
                v_ps[0] += vha_ps
                # energies
                self.hartree_energy += 0.5 * (
                    self._inner_lm(r, ae[0], vha_ae) -
                    self._inner_lm(r, ps[0], vha_ps))
                self.xc_energy += (self._inner_lm(r, ae[0], exc_ae)
                                   - self._inner_lm(r, ps[0], exc_ps))
                core_int = (exc_ae[0] * pd.ae_core - exc_ps[0] * pd.ps_core) \
                    / Y00 * r**2
                self.xc_energy += CubicSpline(r, core_int).integrate(r[0], r[-1])
                # Dij
                self.dij[ia] = self._calc_dij(lab, v_ae, v_ps)

    @staticmethod
    def _inner_lm(r, f_lm, g_lm) -> float:
        """Σ_lm ∫ f_lm g_lm r² dr (both [lmmax, nr])."""
        n = min(f_lm.shape[0], g_lm.shape[0])
        integ = (f_lm[:n] * g_lm[:n]).sum(axis=0) * r**2
        return float(CubicSpline(r, integ).integrate(r[0], r[-1]))

    def _calc_dij(self, lab, v_ae, v_ps):
        ctx = self.ctx
        at = ctx.unit_cell.atom_types[lab]
        pd = self.types[lab]
        r = at.r
        idxb = at.beta_lm_index()
        nbf = len(idxb)
        gc = self.gc(lab)
        lmmax = v_ae.shape[1]
        l_of3 = np.concatenate([[l] * (2 * l + 1)
                                for l in range(2 * self.lmax_b[lab] + 1)])
        ncomp = v_ae.shape[0]
        nbrf = at.num_beta
        # radial integrals per (lm3, rf pair, comp)
        integrals = np.zeros((lmmax, nbrf * (nbrf + 1) // 2, ncomp))
        for rb2 in range(nbrf):
            for rb1 in range(rb2 + 1):
                pidx = rb2 * (rb2 + 1) // 2 + rb1
                ae_part = pd.ae_wf[rb1] * pd.ae_wf[rb2]
                ps_base = pd.ps_wf[rb1] * pd.ps_wf[rb2]
                for lm3 in range(lmmax):
                    l3 = int(l_of3[lm3])
                    qf = self.qmap_get(pd, rb1, rb2, l3)
                    ps_part = ps_base + (qf if qf is not None else 0.0)
                    for ic in range(ncomp):
                        intg = v_ae[ic, lm3] * ae_part - v_ps[ic, lm3] * ps_part
                        integrals[lm3, pidx, ic] = CubicSpline(r, intg).integrate(
                            r[0], r[-1])
        dij = np.zeros((nbf, nbf, ncomp))
        for xi2 in range(nbf):
            rf2, l2, m2 = idxb[xi2]
            lm2 = ylm_mod.lm_index(l2, m2)
            for xi1 in range(xi2 + 1):
                rf1, l1, m1 = idxb[xi1]
                lm1 = ylm_mod.lm_index(l1, m1)
                pidx = max(rf1, rf2) * (max(rf1, rf2) + 1) // 2 + min(rf1, rf2)
                for lm3 in range(lmmax):
                    g = gc[lm1, lm2, lm3]
                    if abs(g) < 1e-14:
                        continue
                    for ic in range(ncomp):
                        dij[xi1, xi2, ic] += g * integrals[lm3, pidx, ic]
                if xi1 != xi2:
                    dij[xi2, xi1, :] = dij[xi1, xi2, :]
        return dij

    # -- energies -----------------------------------------------------------

    def total_energy(self) -> float:
        return self.hartree_energy + self.xc_energy

    def one_elec_energy(self, density) -> float:
        """Σ_a Σ_ij dm_aux(ij,ic)·symw·ΔD_ij(ic) (calc_PAW_one_elec_energy;
        density_matrix_aux already includes the full dm so diag symmetric)."""
        if not self.dij or density.density_matrix is None:
            return 0.0
        uc = self.ctx.unit_cell
        ncomp = self.ctx.num_mag_dims + 1
        e = 0.0
        for lab, pd in self.types.items():
            d = density.density_matrix[lab]
            ia_list = list(uc.atoms_of_type(lab))
            for i, ia in enumerate(ia_list):
                if ia not in self.dij:
                    continue
                dmc = d[i].cpu().numpy()
                dij = self.dij[ia]
                if ncomp == 1:
                    chans = [dmc[..., 0].real]
                else:
                    chans = [(dmc[..., 0] + dmc[..., 1]).real,
                             (dmc[..., 0] - dmc[..., 1]).real]
                for ic in range(ncomp):
                    e += float((chans[ic] * dij[..., ic].T).sum())
        return e
