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

from .core.radial import RadialIntegrals
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
        self.bxc_r = []               # magnetic fields (collinear: [Bz])
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

    def generate(self, density):
        """Build V_H, V_xc, V_eff from the current density.

        Mirrors Potential::generate (potential.cpp:236): poisson on valence
        ρ(G) (incl. augmentation), XC on ρ_valence+ρ_core real-space.
        """
        ctx = self.ctx
        g = ctx.gvec_fine

        # Hartree
        g2 = g.gk2_t.clamp(min=1e-30)
        self.vha_g = 4 * math.pi * density.rho_g / g2
        if self._ig0 >= 0:
            self.vha_g[self._ig0] = 0.0
        vha_r = ctx.fft_fine.to_real(self.vha_g).real
        self.energy_vha = ctx.integrate_rg_fine(density.rho_r * vha_r)

        # XC on rho_val + rho_core
        rho_xc = density.rho_r + density.rho_core_r
        if ctx.is_gga:
            # grad rho via iG on the fine sphere
            rho_tot_g = ctx.fft_fine.to_pw(rho_xc.to(ctx.dtype))
            gvec = g.gkvec_t  # [nG,3]
            grads = []
            for d in range(3):
                gr = ctx.fft_fine.to_real(1j * gvec[:, d] * rho_tot_g).real
                grads.append(gr)
            sigma = grads[0] ** 2 + grads[1] ** 2 + grads[2] ** 2
            eps, vrho, vsigma = xc_mod.evaluate(ctx.xc_names, rho_xc, sigma)
            # v = vrho - 2 div(vsigma grad rho)
            div = torch.zeros_like(vrho)
            for d in range(3):
                fg = ctx.fft_fine.to_pw((vsigma * grads[d]).to(ctx.dtype))
                div = div + ctx.fft_fine.to_real(1j * gvec[:, d] * fg).real
            self.vxc_r = vrho - 2.0 * div
            self.exc_r = eps
        else:
            eps, vrho, _ = xc_mod.evaluate(ctx.xc_names, rho_xc)
            self.vxc_r = vrho
            self.exc_r = eps

        self.veff_r = self.vloc_r + vha_r + self.vxc_r
        self.veff_g = ctx.fft_fine.to_pw(self.veff_r.to(ctx.dtype))

        # collinear magnetism: Bxc = vxc_up - vxc_dn (added with spin support)
        return self

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


def ewald_energy(ctx) -> float:
    """Ewald energy of point charges Z_a (reference: energy.cpp:18-63)."""
    uc = ctx.unit_cell
    # adaptive lambda (simulation_context.cpp:130-151)
    lam = 1.0
    gmax = ctx.pw_cutoff
    q = uc.num_electrons
    while True:
        lam += 0.1
        ub = q * q * math.sqrt(2.0 * lam / (2 * math.pi)) * math.erfc(gmax * math.sqrt(1.0 / (4.0 * lam)))
        if ub >= 1e-8:
            break
    alpha = lam

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
