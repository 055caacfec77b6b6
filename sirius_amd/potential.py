"""Effective potential generation: V_loc, Hartree, XC; energies; Ewald.

Reference behavior: src/potential/potential.cpp:236 (generate),
poisson.cpp:151 (PP branch: V_H(G) = 4π ρ(G)/G², G=0 → 0),
xc.cpp:421 (LDA/GGA on the fine real-space grid),
potential.hpp:250 (generate_local_potential), energy.cpp:19 (Ewald).

All grid math is torch on the compute device; radial form factors are
CPU-side setup (numpy/scipy) cached per geometry.
"""

from __future__ import annotations

import math

import numpy as np
import torch

from . import xc as xc_mod


class Potential:
    def __init__(self, ctx):
        self.ctx = ctx
        g = ctx.gvec_fine
        dev = ctx.device
        self.vloc_g = None            # complex [nG_fine]
        self.vloc_r = None            # real fine grid
        self.vha_g = torch.zeros(g.num_gvec, dtype=ctx.dtype, device=dev)
        self.vxc_r = None
        self.exc_r = None
        self.veff_r = None            # fine grid real values
        self.veff_g = None
        self.bz_r = None              # collinear XC magnetic field B_z(r)
        self.bz_g = None
        self.bvec_r = None            # noncollinear [Bx, By, Bz](r)
        self.bvec_g = None
        self.energy_vha = 0.0
        self.ewald = None
        self._ig0 = g.index_of_zero()
        self.generate_local_potential()
        self.ewald = ewald_energy(ctx)

    # -- setup -------------------------------------------------------------

    def generate_local_potential(self):
        ctx = self.ctx
        uc = ctx.unit_cell
        ff = {}
        q = ctx.gvec_fine.shell_len
        for lab, at in uc.atom_types.items():
            ff[lab] = ctx.ri.vloc(lab)(q)
        self.vloc_g = ctx.make_periodic_function(ff)
        self.vloc_r = ctx.fft_fine.to_real(self.vloc_g).real

    # -- per-iteration generation ------------------------------------------

    def _grad_r(self, f_r):
        """∇f of a real fine-grid field via iG (3 FFTs)."""
        ctx = self.ctx
        fg = ctx.fft_fine.to_pw(f_r.to(ctx.dtype))
        gvec = ctx.gvec_fine.gkvec_t
        return [ctx.fft_fine.to_real(1j * gvec[:, d] * fg).real for d in range(3)]

    def _div_r(self, F):
        """∇·F from 3 real fields via iG."""
        ctx = self.ctx
        gvec = ctx.gvec_fine.gkvec_t
        out = None
        for d in range(3):
            fg = ctx.fft_fine.to_pw(F[d].to(ctx.dtype))
            t = ctx.fft_fine.to_real(1j * gvec[:, d] * fg).real
            out = t if out is None else out + t
        return out

    def generate(self, density):
        """Build V_H, V_xc, V_eff (+B_z for collinear magnetism).

        Mirrors Potential::generate (potential.cpp:236): poisson on valence
        ρ(G) (incl. augmentation), XC on ρ_valence+ρ_core real-space
        (xc_rg_nonmagnetic xc.cpp:26 / xc_rg_magnetic xc.cpp:197).
        """
        ctx = self.ctx
        g = ctx.gvec_fine
        # previous V_eff(G) for the SCF force correction (potential.cpp:240-316)
        veff_old = self.veff_g

        # Hartree
        g2 = g.gk2_t.clamp(min=1e-30)
        self.vha_g = 4 * math.pi * density.rho_g / g2
        if self._ig0 >= 0:
            self.vha_g[self._ig0] = 0.0
        vha_r = ctx.fft_fine.to_real(self.vha_g).real
        self.energy_vha = ctx.integrate_rg_fine(density.rho_r * vha_r)

        rho_xc = density.rho_r + density.rho_core_r
        if ctx.nc_magnetism:
            # noncollinear: locally collinear along m̂ (xc_rg_magnetic with
            # rho_up/dn = (ρ+core ± |m|)/2; B = ½(vu−vd)·m̂)
            mx, my, mz = (density.magv_r[0], density.magv_r[1], density.magv_r[2])
            mlen = torch.sqrt(mx**2 + my**2 + mz**2).clamp(min=1e-30)
            ru = 0.5 * (rho_xc + mlen)
            rd = 0.5 * (rho_xc - mlen)
            if ctx.is_gga:
                gu = self._grad_r(ru)
                gd = self._grad_r(rd)
                s_uu = gu[0]**2 + gu[1]**2 + gu[2]**2
                s_dd = gd[0]**2 + gd[1]**2 + gd[2]**2
                gt = [gu[d] + gd[d] for d in range(3)]
                s_tot = gt[0]**2 + gt[1]**2 + gt[2]**2
                eps, vu, vd, vs_uu, vs_dd, vs_tot = xc_mod.evaluate_spin(
                    ctx.xc_names, ru, rd, s_uu, s_dd, s_tot)
                vu = vu - 2.0 * self._div_r([vs_uu * gu[d] for d in range(3)])                         - 2.0 * self._div_r([vs_tot * gt[d] for d in range(3)])
                vd = vd - 2.0 * self._div_r([vs_dd * gd[d] for d in range(3)])                         - 2.0 * self._div_r([vs_tot * gt[d] for d in range(3)])
            else:
                eps, vu, vd, *_ = xc_mod.evaluate_spin(ctx.xc_names, ru, rd)
            self.vxc_r = 0.5 * (vu + vd)
            b_amp = 0.5 * (vu - vd)
            self.bvec_r = [b_amp * mx / mlen, b_amp * my / mlen,
                           b_amp * mz / mlen]
            self.bvec_g = [ctx.fft_fine.to_pw(b.to(ctx.dtype))
                           for b in self.bvec_r]
            self.bz_r = self.bvec_r[2]
            self.bz_g = self.bvec_g[2]
            self.exc_r = eps
        elif ctx.num_spins == 1:
            if ctx.is_gga:
                grads = self._grad_r(rho_xc)
                sigma = grads[0] ** 2 + grads[1] ** 2 + grads[2] ** 2
                eps, vrho, vsigma = xc_mod.evaluate(ctx.xc_names, rho_xc, sigma)
                div = self._div_r([vsigma * gr for gr in grads])
                self.vxc_r = vrho - 2.0 * div
                self.exc_r = eps
            else:
                eps, vrho, _ = xc_mod.evaluate(ctx.xc_names, rho_xc)
                self.vxc_r = vrho
                self.exc_r = eps
            self.bz_r = None
        else:
            # collinear: rho_up/dn = (rho + core ± m_z)/2 (get_rho_up_dn)
            ru = 0.5 * (rho_xc + density.mag_r)
            rd = 0.5 * (rho_xc - density.mag_r)
            if ctx.is_gga:
                gu = self._grad_r(ru)
                gd = self._grad_r(rd)
                s_uu = gu[0] ** 2 + gu[1] ** 2 + gu[2] ** 2
                s_dd = gd[0] ** 2 + gd[1] ** 2 + gd[2] ** 2
                gt = [gu[d] + gd[d] for d in range(3)]
                s_tot = gt[0] ** 2 + gt[1] ** 2 + gt[2] ** 2
                eps, vu, vd, vs_uu, vs_dd, vs_tot = xc_mod.evaluate_spin(
                    ctx.xc_names, ru, rd, s_uu, s_dd, s_tot)
                vu = vu - 2.0 * self._div_r([vs_uu * gu[d] for d in range(3)])                         - 2.0 * self._div_r([vs_tot * gt[d] for d in range(3)])
                vd = vd - 2.0 * self._div_r([vs_dd * gd[d] for d in range(3)])                         - 2.0 * self._div_r([vs_tot * gt[d] for d in range(3)])
            else:
                eps, vu, vd, *_ = xc_mod.evaluate_spin(ctx.xc_names, ru, rd)
            self.vxc_r = 0.5 * (vu + vd)
            self.bz_r = 0.5 * (vu - vd)
            self.exc_r = eps
            self.bz_g = ctx.fft_fine.to_pw(self.bz_r.to(ctx.dtype))

        self.veff_r = self.vloc_r + vha_r + self.vxc_r
        self.veff_g = ctx.fft_fine.to_pw(self.veff_r.to(ctx.dtype))
        self.dveff_g = (self.veff_g - veff_old if veff_old is not None
                        else self.veff_g.clone())
        if ctx.hubbard is not None:
            ctx.hubbard.generate_potential()
        return self

    def generate_paw(self, density):
        """PAW on-site potentials + Dij (generate_PAW_effective_potential)."""
        if self.ctx.paw is not None:
            self.ctx.paw.generate(density)

    def energy_bxc(self, density) -> float:
        """∫ m·B (reference energy.cpp:85-93)."""
        if self.ctx.nc_magnetism and self.bvec_r is not None:
            return sum(self.ctx.integrate_rg_fine(density.magv_r[i] * self.bvec_r[i])
                       for i in range(3))
        if self.bz_r is None or density.mag_r is None:
            return 0.0
        return self.ctx.integrate_rg_fine(density.mag_r * self.bz_r)

    # -- energies ----------------------------------------------------------

    def energy_vxc(self, density) -> float:
        return self.ctx.integrate_rg_fine(density.rho_r * self.vxc_r)

    def energy_exc(self, density) -> float:
        return self.ctx.integrate_rg_fine(
            (density.rho_r + density.rho_core_r) * self.exc_r)

    def energy_vloc(self, density) -> float:
        return self.ctx.integrate_rg_fine(density.rho_r * self.vloc_r)

    def energy_veff(self, density) -> float:
        return self.ctx.integrate_rg_fine(density.rho_r * self.veff_r)


def ewald_lambda(ctx) -> float:
    """Adaptive Ewald screening parameter (simulation_context.cpp:130-151)."""
    uc = ctx.unit_cell
    lam = 1.0
    gmax = ctx.pw_cutoff
    q = uc.num_electrons
    while True:
        lam += 0.1
        ub = q * q * math.sqrt(2.0 * lam / (2 * math.pi)) * math.erfc(gmax * math.sqrt(1.0 / (4.0 * lam)))
        if ub >= 1e-8:
            break
    return lam


def ewald_energy(ctx) -> float:
    """Ewald energy of point charges Z_a (reference: energy.cpp:18-63)."""
    uc = ctx.unit_cell
    alpha = ewald_lambda(ctx)
    q = uc.num_electrons

    g = ctx.gvec_fine
    zn = np.array([uc.atom_types[lab].zn for lab, _ in uc.atoms], dtype=np.float64)
    tau = uc.atom_positions_frac()
    m = g.miller.astype(np.float64)
    ph = np.exp(2j * math.pi * (tau @ m.T))  # e^{+iG·τ}
    rho_g = zn @ ph  # [nG]
    g2 = g.gk_len ** 2
    nz = g2 > 1e-20
    ewald_g = float(np.sum(np.abs(rho_g[nz]) ** 2 * np.exp(-g2[nz] / (4 * alpha)) / g2[nz]))
    ewald_g -= q * q / alpha / 4.0
    ewald_g *= 2 * math.pi / uc.omega
    for z in zn:
        ewald_g -= math.sqrt(alpha / math.pi) * z * z

    r_cut = max(np.linalg.norm(uc.lattice, axis=1))  # unit_cell.cpp:836-841
    ewald_r = 0.0
    for ia, ja, d in uc.nearest_neighbours(r_cut):
        ewald_r += 0.5 * zn[ia] * zn[ja] * math.erfc(math.sqrt(alpha) * d) / d
    return ewald_g + ewald_r
