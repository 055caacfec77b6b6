"""Stress tensor for the pseudopotential plane-wave branch.

Reference behavior: src/geometry/stress.cpp (Stress class). Total
(calc_stress_total, stress.cpp:76-100):

    σ = σ_kin + σ_har + σ_ewald + σ_vloc + σ_core + σ_xc + σ_us
        + σ_nonloc (+ σ_hubbard)

- kin    (stress.cpp:634-676): −(1/Ω)Σ_kn occ·w·|ψ(G)|²·(G+k)⊗(G+k)
- har    (stress.cpp:598-631): 2π|ρ(G)|²/G²·(2 Ĝ⊗Ĝ − 1)
- ewald  (stress.cpp:477-545): G-space + real-space with rc⊗rc
- vloc   (stress.cpp:691-738): Re[conj(ρ)·dv/dG]·G⊗G − δ·Re[conj(ρ)v]
- core   (stress.cpp:205-268): same shape with ρ_core / V_xc
- xc     (stress.cpp:271-355): δ·(E_xc − ∫Vxc·ρ − ∫Bxc·m)/Ω + GGA
         gradient term Σ 2·vsigma·∇ρ⊗∇ρ
- us     (stress.cpp:358-474): augmentation with dQ(G)/dG_nu
- nonloc (stress.cpp:26-73 + non_local_functor.hpp): strain-derivative
  beta projectors (beta_projectors_strain_deriv.hpp:25-105):
  d⟨G+k|β⟩/dε_{μν} = (−i)^l·4π/√Ω·[f_l(q)(−G_μ·∂R_lm/∂G_ν − ½δ_{μν}R_lm)
                      + (df_l/dq)·R_lm·(−G_μ G_ν/q)]

Each term is symmetrized σ ← (1/N)Σ_ops Rᵀ σ R
(symmetrize_stress_tensor.hpp:21-36). Output convention matches the
reference app: result["stress"][i][j] = σ(j,i) in Ha/bohr³
(apps/mini_app/sirius.scf.cpp:187-191).
"""

from __future__ import annotations

import math

import numpy as np
import torch

from .core import la
from .core import ylm as ylm_mod
from .parallel import get_comm


def _sym_stress(ctx, s: np.ndarray) -> np.ndarray:
    sym = getattr(ctx, "symmetry", None)
    if sym is None or len(sym.ops) <= 1:
        return s
    out = np.zeros_like(s)
    for op in sym.ops:
        out += op.S.T @ s @ op.S
    return out / len(sym.ops)


_rlm_and_grad = ylm_mod.rlm_and_cart_grad


class BetaProjectorsStrain:
    """9-component strain-derivative beta projectors, stored transposed
    [nbf_tot, nGk] like BetaProjectors.beta_t (see
    beta_projectors_strain_deriv.hpp)."""

    def __init__(self, ctx, kp):
        uc = ctx.unit_cell
        g = kp.gkvec
        gc = g.gkvec_cart                                   # [nGk, 3]
        glen = g.gk_len
        lmax = max((b.l for at in uc.atom_types.values() for b in at.beta),
                   default=0)
        rl, rl_dg = _rlm_and_grad(lmax, gc)
        inv_len = np.where(glen > 1e-10, 1.0 / np.maximum(glen, 1e-300), 0.0)

        cols_t = {}   # per type, per component: [nGk, nbf_t]
        for lab, at in uc.atom_types.items():
            if at.num_beta == 0:
                cols_t[lab] = [np.zeros((len(glen), 0), dtype=np.complex128)
                               for _ in range(9)]
                continue
            f0 = ctx.ri.beta(lab)(glen)                     # [nrb, nGk]
            f1 = ctx.ri.beta_djl(lab)(glen)
            comp = []
            for nu in range(3):
                for mu in range(3):
                    p = 0.5 if mu == nu else 0.0
                    cols = []
                    for (irf, l, m) in at.beta_lm_index():
                        lm = ylm_mod.lm_index(l, m)
                        z = (-1j) ** l * (4 * math.pi / math.sqrt(uc.omega))
                        d1 = f0[irf] * (-gc[:, mu] * rl_dg[:, nu, lm]
                                        - p * rl[:, lm])
                        d2 = f1[irf] * rl[:, lm] * (-gc[:, mu] * gc[:, nu]
                                                    * inv_len)
                        cols.append(z * (d1 + d2))
                    comp.append(np.stack(cols, axis=1))
            # reorder into x = mu + 3 nu
            cols_t[lab] = comp  # comp[nu*3 + mu] indexed below as mu+3nu
        mk = (g.miller + g.k_frac).astype(np.float64)
        self.beta_t = []
        for x in range(9):
            blocks = []
            for ia, (lab, tau) in enumerate(uc.atoms):
                if uc.atom_types[lab].num_beta_lm == 0:
                    continue
                phase = np.exp(-2j * math.pi * (mk @ tau))
                blocks.append(cols_t[lab][x] * phase[:, None])
            if blocks:
                bt = np.ascontiguousarray(np.concatenate(blocks, axis=1).T)
            else:
                bt = np.zeros((0, len(glen)), dtype=np.complex128)
            self.beta_t.append(torch.from_numpy(bt).to(ctx.device))


class Stress:
    def __init__(self, ctx, density, potential, kset, h0):
        self.ctx = ctx
        self.density = density
        self.potential = potential
        self.kset = kset
        self.h0 = h0

    # -- helpers -----------------------------------------------------------

    def _g_quantities(self):
        ctx = self.ctx
        g = ctx.gvec_fine
        gc = g.gkvec_cart
        g2 = g.gk_len ** 2
        return g, gc, g2

    def _struct(self, ff_of_lab) -> torch.Tensor:
        """(4π/Ω)·Σ_a ff_t(|G|)·e^{-iG·τ_a}
        (make_periodic_function, make_periodic_function.hpp:23-51)."""
        ctx = self.ctx
        uc = ctx.unit_cell
        out = torch.zeros(ctx.gvec_fine.num_gvec, dtype=ctx.dtype,
                          device=ctx.device)
        for lab in uc.atom_types:
            ff = ff_of_lab(lab)
            if ff is None:
                continue
            ffd = torch.from_numpy(np.ascontiguousarray(ff)).to(ctx.device)
            ph = ctx.phase_pos(lab).conj().sum(dim=0)       # Σ_a e^{-iGτ}
            out += ffd * ph
        return out * (4 * math.pi / uc.omega)

    # -- components --------------------------------------------------------

    def calc_stress_kin(self) -> np.ndarray:
        ctx = self.ctx
        s = np.zeros((3, 3))
        for kp in self.kset:
            gkc = torch.from_numpy(kp.gkvec.gkvec_cart).to(ctx.device)
            ng = kp.num_gkvec
            nss = ctx.num_spin_steps
            for iss in range(nss):
                occ = kp.occ[iss]
                sel = np.nonzero(occ > 1e-14)[0]
                if len(sel) == 0:
                    continue
                w = torch.from_numpy(occ[sel] * kp.weight).to(ctx.device)
                psi = kp.psi[iss][torch.from_numpy(sel).to(ctx.device)]
                if ctx.nc_magnetism:
                    dens = (psi[:, :ng].abs() ** 2
                            + psi[:, ng:].abs() ** 2)       # both spinors
                else:
                    dens = psi.abs() ** 2
                d = torch.einsum("n,ng->g", w.to(torch.float64), dens)
                s += torch.einsum("g,gu,gv->uv", d, gkc, gkc).cpu().numpy()
        comm = get_comm()
        if comm.active:
            t = torch.from_numpy(s)
            comm.allreduce_(t)
            s = t.numpy()
        s *= -1.0 / ctx.unit_cell.omega
        return _sym_stress(ctx, s)

    def calc_stress_har(self) -> np.ndarray:
        ctx = self.ctx
        g, gc, g2 = self._g_quantities()
        rho = self.density.rho_g.cpu().numpy()
        nz = g2 > 1e-20
        d = 2 * math.pi * np.abs(rho[nz]) ** 2 / g2[nz]
        s = 2.0 * np.einsum("g,gu,gv->uv", d / g2[nz], gc[nz], gc[nz])
        s -= np.eye(3) * d.sum()
        return _sym_stress(ctx, s)

    def calc_stress_ewald(self) -> np.ndarray:
        from .potential import ewald_lambda

        ctx = self.ctx
        uc = ctx.unit_cell
        lam = ewald_lambda(ctx)
        g, gc, g2 = self._g_quantities()
        zn = np.array([uc.atom_types[lab].zn for lab, _ in uc.atoms])
        tau = uc.atom_positions_frac()
        m = g.miller.astype(np.float64)
        ph = np.exp(2j * math.pi * (tau @ m.T))
        rho = zn @ ph
        nz = g2 > 1e-20
        g2l = g2[nz] / (4 * lam)
        a1 = 2 * math.pi * (np.abs(rho[nz]) / uc.omega) ** 2 \
            * np.exp(-g2l) / g2[nz]
        s = np.einsum("g,gu,gv->uv", a1 * 2 * (g2l + 1) / g2[nz],
                      gc[nz], gc[nz])
        s -= np.eye(3) * a1.sum()
        s += np.eye(3) * (2 * math.pi * (uc.num_electrons / uc.omega) ** 2
                          / (4 * lam))
        r_cut = max(np.linalg.norm(uc.lattice, axis=1))
        for ia, ja, d, T, rc in uc.nearest_neighbours_full(r_cut):
            a1r = (0.5 * zn[ia] * zn[ja] / uc.omega / d ** 3) * (
                -2 * math.exp(-lam * d * d) * math.sqrt(lam / math.pi) * d
                - math.erfc(math.sqrt(lam) * d))
            s += a1r * np.outer(rc, rc)
        return _sym_stress(ctx, s)

    def calc_stress_vloc(self) -> np.ndarray:
        ctx = self.ctx
        g, gc, g2 = self._g_quantities()
        q = g.gk_len
        v = self._struct(lambda lab: ctx.ri.vloc(lab)(q)).cpu().numpy()
        dv = self._struct(lambda lab: ctx.ri.vloc_djl(lab)(q)).cpu().numpy()
        rho = self.density.rho_g.cpu().numpy()
        nz = g2 > 1e-20
        w = np.real(np.conj(rho[nz]) * dv[nz])
        s = np.einsum("g,gu,gv->uv", w, gc[nz], gc[nz])
        sdiag = float(np.real(np.conj(rho) * v).sum())
        s -= np.eye(3) * sdiag
        return _sym_stress(ctx, s)

    def calc_stress_core(self) -> np.ndarray:
        ctx = self.ctx
        uc = ctx.unit_cell
        has_core = any(at.core_correction and at.rho_core_r.any()
                       for at in uc.atom_types.values())
        if not has_core:
            return np.zeros((3, 3))
        g, gc, g2 = self._g_quantities()
        q = g.gk_len

        def ff(tab):
            def f(lab):
                at = uc.atom_types[lab]
                if not (at.core_correction and at.rho_core_r.any()):
                    return None
                return tab(lab)(q)
            return f
        rhoc = self._struct(ff(ctx.ri.rho_core)).cpu().numpy()
        drhoc = self._struct(ff(ctx.ri.rho_core_djl)).cpu().numpy()
        vxc = ctx.fft_fine.to_pw(
            self.potential.vxc_r.to(ctx.dtype)).cpu().numpy()
        nz = g2 > 1e-20
        glen = np.sqrt(g2[nz])
        w = np.real(np.conj(vxc[nz]) * drhoc[nz]) / glen
        s = -np.einsum("g,gu,gv->uv", w, gc[nz], gc[nz])
        sdiag = float(np.real(np.conj(vxc) * rhoc).sum())
        s -= np.eye(3) * sdiag
        return _sym_stress(ctx, s)

    def calc_stress_xc(self) -> np.ndarray:
        from . import xc as xc_mod

        ctx = self.ctx
        pot = self.potential
        dens = self.density
        e = (pot.energy_exc(dens) - pot.energy_vxc(dens)
             - pot.energy_bxc(dens))
        s = np.eye(3) * (e / ctx.unit_cell.omega)
        if ctx.is_gga:
            if ctx.nc_magnetism:
                raise NotImplementedError("GGA stress: collinear only")
            rho_xc = dens.rho_r + dens.rho_core_r
            t = np.zeros((3, 3))
            if ctx.num_spins == 1:
                grads = pot._grad_r(rho_xc)
                sigma = grads[0] ** 2 + grads[1] ** 2 + grads[2] ** 2
                _, _, vsigma = xc_mod.evaluate(ctx.xc_names, rho_xc, sigma)
                for mu in range(3):
                    for nu in range(3):
                        t[mu, nu] = float((2.0 * grads[mu] * grads[nu]
                                           * vsigma).sum())
            else:
                ru = 0.5 * (rho_xc + dens.mag_r)
                rd = 0.5 * (rho_xc - dens.mag_r)
                gu = pot._grad_r(ru)
                gd = pot._grad_r(rd)
                s_uu = gu[0] ** 2 + gu[1] ** 2 + gu[2] ** 2
                s_dd = gd[0] ** 2 + gd[1] ** 2 + gd[2] ** 2
                gt = [gu[d] + gd[d] for d in range(3)]
                s_tot = gt[0] ** 2 + gt[1] ** 2 + gt[2] ** 2
                _, _, _, vs_uu, vs_dd, vs_tot = xc_mod.evaluate_spin(
                    ctx.xc_names, ru, rd, s_uu, s_dd, s_tot)
                # libxc-convention expansion (stress.cpp:316-345) rewritten
                # in (σ_uu, σ_dd, σ_tot) variables:
                # t = 2[vs_uu ∇ρ↑⊗∇ρ↑ + vs_dd ∇ρ↓⊗∇ρ↓ + vs_tot ∇ρ⊗∇ρ]
                for mu in range(3):
                    for nu in range(3):
                        t[mu, nu] = float(
                            (2.0 * (vs_uu * gu[mu] * gu[nu]
                                    + vs_dd * gd[mu] * gd[nu]
                                    + vs_tot * gt[mu] * gt[nu])).sum())
            t *= -1.0 / (ctx.fft_fine.size)
            s += t
        return _sym_stress(ctx, s)

    def calc_stress_us(self) -> np.ndarray:
        ctx = self.ctx
        uc = ctx.unit_cell
        s = np.zeros((3, 3))
        dm = getattr(self.density, "density_matrix", None)
        if not ctx.has_aug or dm is None:
            return s
        from .augmentation import AugmentationOperator

        g, gc, g2 = self._g_quantities()
        glen = np.maximum(np.sqrt(g2), 1e-300)
        gch = torch.from_numpy(gc / glen[:, None]).to(ctx.device)
        comps = [self.potential.veff_g]
        if ctx.num_spins == 2 and not ctx.nc_magnetism:
            comps += [self.potential.bz_g]
        elif ctx.nc_magnetism:
            comps += [self.potential.bvec_g[0], self.potential.bvec_g[1],
                      self.potential.bvec_g[2]]
        for lab, at in uc.atom_types.items():
            if not (at.augment and at.num_beta):
                continue
            aug = ctx.aug_op(lab)
            ph = ctx.phase_pos(lab)
            d = dm[lab]
            nbf = aug.nbf
            xi1_idx, xi2_idx = [], []
            for xi2 in range(nbf):
                for xi1 in range(xi2 + 1):
                    xi1_idx.append(xi1)
                    xi2_idx.append(xi2)
            dpk = d[:, xi2_idx, xi1_idx, :]
            chans = [dpk[..., 0].real] if dpk.shape[-1] == 1 else \
                [(dpk[..., 0] + dpk[..., 1]).real,
                 (dpk[..., 0] - dpk[..., 1]).real]
            if ctx.nc_magnetism:
                chans = [(dpk[..., 0] + dpk[..., 1]).real,
                         2.0 * dpk[..., 2].real,
                         -2.0 * dpk[..., 2].imag,
                         (dpk[..., 0] - dpk[..., 1]).real]
            sw = aug.sym_weight.to(torch.float64)
            for nu in range(3):
                dq = AugmentationOperator(ctx, at, gvec_deriv=nu)  # dQ/dG_nu
                for s_idx, dm_aux in enumerate(chans):
                    vg = comps[s_idx]
                    for mu in range(3):
                        vx = (-gch[:, mu]).to(ctx.dtype) * vg
                        v_at = ph * vx[None, :]
                        tmp = la.inner(dq.q_pw, v_at).real  # [npack, na]
                        val = torch.einsum(
                            "q,aq,qa->", sw,
                            dm_aux.real.to(torch.float64), tmp)
                        s[mu, nu] += float(val)
        # no 1/Ω here: our q_pw derivative keeps the base 4π/Ω
        # prefactor (the reference's deriv drops /Ω and divides at
        # the end, stress.cpp:469-473)
        return _sym_stress(ctx, s)

    def calc_stress_nonloc(self) -> np.ndarray:
        ctx = self.ctx
        uc = ctx.unit_cell
        if any(at.spin_orbit for at in uc.atom_types.values()):
            raise NotImplementedError(
                "stress with spin-orbit coupling is not supported "
                "(matches reference)")
        s = np.zeros((3, 3))
        from .hamiltonian import HamiltonianK

        for kp in self.kset:
            hk = HamiltonianK(self.h0, kp)
            bp = hk.bp
            if bp.num_beta_total == 0:
                continue
            bps = BetaProjectorsStrain(ctx, kp)
            ng = kp.num_gkvec
            nss = ctx.num_spin_steps
            nspins = 2 if ctx.nc_magnetism else ctx.num_spins
            for ispn in range(nspins):
                iss = 0 if ctx.nc_magnetism else ispn
                occ = kp.occ[iss]
                sel = np.nonzero(occ > 1e-14)[0]
                if len(sel) == 0:
                    continue
                idx = torch.from_numpy(sel).to(ctx.device)
                if ctx.nc_magnetism:
                    psi = kp.psi[0][idx, ispn * ng:(ispn + 1) * ng].contiguous()
                else:
                    psi = kp.psi[ispn][idx].contiguous()
                eps = torch.from_numpy(kp.eigvals[iss][sel]).to(ctx.device)
                w = torch.from_numpy(occ[sel] * kp.weight).to(ctx.device)
                bphi = bp.inner(psi)
                D = hk.D[ispn]
                M = D @ bphi.conj()
                Q = hk.Q_blocks[ispn] if (ctx.nc_magnetism
                                          and hk.Q_blocks) else hk.Q
                N = Q @ bphi.conj() if Q is not None else None
                for x in range(9):
                    mu, nu = x % 3, x // 3
                    bgphi = la.inner(bps.beta_t[x].contiguous(), psi)
                    P = (bgphi * M)
                    val = -2.0 * float(torch.einsum(
                        "n,in->", w.to(torch.float64), P.real))
                    if N is not None:
                        P2 = (bgphi * N)
                        val += 2.0 * float(torch.einsum(
                            "n,in->", (w * eps).to(torch.float64), P2.real))
                    s[mu, nu] -= val
        comm = get_comm()
        if comm.active:
            t = torch.from_numpy(s)
            comm.allreduce_(t)
            s = t.numpy()
        s *= 1.0 / uc.omega
        return _sym_stress(ctx, s)

    def calc_stress_hubbard(self) -> np.ndarray:
        """calc_stress_hubbard (stress.cpp:103-203): σ(μ,ν) −= Re[V·dn]/Ω
        per k, then a global sign flip."""
        ctx = self.ctx
        hub = ctx.hubbard
        uc = ctx.unit_cell
        s = np.zeros((3, 3))
        if hub is None or hub.num_wf == 0:
            return s
        if ctx.nc_magnetism:
            raise NotImplementedError("Hubbard stress: collinear only")
        from .hamiltonian import HamiltonianK
        from .hubbard import compute_occupancies_stress_derivatives

        nsp = ctx.num_spins
        for kp in self.kset:
            hk = HamiltonianK(self.h0, kp)
            dn = compute_occupancies_stress_derivatives(hub, kp, hk)
            for d1 in range(3):
                for d2 in range(3):
                    x = d1 + 3 * d2
                    acc = 0.0
                    for il, (ia1, o) in enumerate(hub.levels):
                        if not hub.level_use[il]:
                            continue
                        off = hub.offsets[il]
                        mm = 2 * o.l + 1
                        for ispn in range(nsp):
                            acc += float(torch.einsum(
                                "ba,abx->x", hub.um[il][..., ispn],
                                dn[off:off + mm, off:off + mm, ispn])[x].real)
                    for i, p in enumerate(hub.nonlocal_pairs):
                        z1 = complex(np.exp(-2j * math.pi
                                            * float(np.dot(p.T, kp.k_frac))))
                        a1 = hub._find_level(p.ia, p.n1, p.il)
                        a2 = hub._find_level(p.ja, p.n2, p.jl)
                        o1, o2 = hub.offsets[a1], hub.offsets[a2]
                        mi, mj = 2 * p.il + 1, 2 * p.jl + 1
                        for ispn in range(nsp):
                            acc += float((z1 * torch.einsum(
                                "ab,bax->x", hub.um_nl[i][..., ispn],
                                dn[o2:o2 + mj, o1:o1 + mi, ispn].conj())[x])
                                .real)
                    s[d1, d2] -= acc / uc.omega
        comm = get_comm()
        if comm.active:
            t = torch.from_numpy(s)
            comm.allreduce_(t)
            s = t.numpy()
        return -1.0 * _sym_stress(ctx, s)

    # -- total -------------------------------------------------------------

    def calc_stress_total(self) -> dict:
        parts = {
            "kin": self.calc_stress_kin(),
            "har": self.calc_stress_har(),
            "ewald": self.calc_stress_ewald(),
            "vloc": self.calc_stress_vloc(),
            "core": self.calc_stress_core(),
            "xc": self.calc_stress_xc(),
            "us": self.calc_stress_us(),
            "nonloc": self.calc_stress_nonloc(),
            "hubbard": self.calc_stress_hubbard(),
        }
        parts["total"] = sum(parts.values())
        return parts
