"""Atomic forces for the pseudopotential plane-wave branch.

Reference behavior: src/geometry/force.cpp (Force class). Total PP force
per atom (calc_forces_total, force.cpp:167-204):

    F = F_vloc + F_us + F_nonloc + F_core + F_ewald (+ F_scf_corr
        + F_hubbard)

with the individual terms:
- vloc     (force.cpp:655-703): −Σ_G Im[G · 4π·ff_vloc(|G|)·conj(ρ(G))·
           conj(e^{iG·τ})]
- core     (force.cpp:531-588): same with ff = ps core density and V_xc(G)
- scf_corr (force.cpp:475-528): same with ff = ps total density and
           ΔV_eff(G) = V_new − V_old across the last potential generation
- ewald    (force.cpp:306-378): G-space sum + screened real-space
           neighbour sum
- us       (force.cpp:381-472): Ω·Σ dm_aux·symw·Re[conj(Q_ij(G))·
           (−iG_x e^{iGτ} V_s(G))]
- nonloc   (non_local_functor.hpp:25-170): −2 Σ_kn occ·w·
           Re[conj(⟨β_j|ψ⟩)(D_ij − ε_n Q_ij)⟨∂β_i/∂x|ψ⟩] with
           ∂β/∂x = −i(G+k)_x β (beta_projectors_gradient.hpp:28-44)

Each term is symmetrized over the space group
(symmetrize_forces.hpp:21-59). Forces with spin-orbit coupling raise, as
in the reference (non_local_functor.hpp:93).

MI355X design: the G-space sums are batched torch tensor contractions on
the device (one [natoms, nG] phase matrix per type, already cached for
the D-matrix/augmentation machinery); the β-gradient inner products ride
the same MFMA Gram kernel as the SCF itself.
"""

from __future__ import annotations

import math

import numpy as np
import torch

from .core import la
from .parallel import get_comm


def _sym_forces(ctx, f: np.ndarray) -> np.ndarray:
    """F[perm[a]] += S·F[a] averaged over ops (symmetrize_forces.hpp)."""
    sym = getattr(ctx, "symmetry", None)
    if sym is None or len(sym.ops) <= 1:
        return f
    out = np.zeros_like(f)
    for op in sym.ops:
        for ia in range(f.shape[0]):
            out[op.perm[ia]] += op.S @ f[ia]
    return out / len(sym.ops)


class Forces:
    def __init__(self, ctx, density, potential, kset, h0):
        self.ctx = ctx
        self.density = density
        self.potential = potential
        self.kset = kset
        self.h0 = h0

    # -- shared helpers ----------------------------------------------------

    def _gcart(self):
        ctx = self.ctx
        g = ctx.gvec_fine
        return torch.from_numpy(g.gkvec_cart).to(ctx.device)   # [nG, 3]

    def _formfactor_force(self, ff_of_lab, field_g) -> np.ndarray:
        """F[ia, x] = −Σ_G Im[G_x · 4π·ff_t(|G|)·conj(field(G)·e^{iG·τ_a})]

        (the shared shape of vloc/core/scf_corr forces, force.cpp:655/531/475;
        fact=1 — our G set is the full ±G sphere, not the reduced half.)"""
        ctx = self.ctx
        uc = ctx.unit_cell
        gc = self._gcart()                                     # [nG, 3]
        q = ctx.gvec_fine.gk_len
        out = np.zeros((uc.num_atoms, 3))
        for lab, at in uc.atom_types.items():
            ff = ff_of_lab(lab)
            if ff is None:
                continue
            ffd = torch.from_numpy(np.ascontiguousarray(ff)).to(ctx.device)
            ph = ctx.phase_pos(lab)                            # [na, nG] e^{+iGτ}
            z = (4 * math.pi) * ffd * torch.conj(field_g)      # [nG]
            zph = torch.conj(ph) * z[None, :]                  # conj(field·phase)·4πff
            fa = -torch.einsum("ag,gx->ax", zph.imag.to(torch.float64), gc)
            fa = fa.cpu().numpy()
            for i, ia in enumerate(uc.atoms_of_type(lab)):
                out[ia] = fa[i]
        return out

    # -- individual contributions -----------------------------------------

    def calc_forces_vloc(self) -> np.ndarray:
        ctx = self.ctx

        def ff(lab):
            return ctx.ri.vloc(lab)(ctx.gvec_fine.gk_len)
        return _sym_forces(ctx, self._formfactor_force(ff, self.density.rho_g))

    def calc_forces_core(self) -> np.ndarray:
        ctx = self.ctx
        vxc_g = ctx.fft_fine.to_pw(self.potential.vxc_r.to(ctx.dtype))

        def ff(lab):
            at = ctx.unit_cell.atom_types[lab]
            if not (at.core_correction and at.rho_core_r.any()):
                return None
            return ctx.ri.rho_core(lab)(ctx.gvec_fine.gk_len)
        return _sym_forces(ctx, self._formfactor_force(ff, vxc_g))

    def calc_forces_scf_corr(self) -> np.ndarray:
        ctx = self.ctx
        dveff = getattr(self.potential, "dveff_g", None)
        if dveff is None:
            return np.zeros((ctx.unit_cell.num_atoms, 3))

        def ff(lab):
            return ctx.ri.rho_pseudo(lab)(ctx.gvec_fine.gk_len)
        return _sym_forces(ctx, self._formfactor_force(ff, dveff))

    def calc_forces_ewald(self) -> np.ndarray:
        """force.cpp:306-378. NOTE the real-space angular factor: the
        reference multiplies the pair scalar by the LATTICE TRANSLATION
        t = A·T (force.cpp:364-371), not the full connecting vector —
        we reproduce that verbatim so forces match its outputs."""
        from .potential import ewald_lambda

        ctx = self.ctx
        uc = ctx.unit_cell
        alpha = ewald_lambda(ctx)
        g = ctx.gvec_fine
        zn = np.array([uc.atom_types[lab].zn for lab, _ in uc.atoms])
        tau = uc.atom_positions_frac()
        m = g.miller.astype(np.float64)
        ph = np.exp(2j * math.pi * (tau @ m.T))                # e^{+iG·τ} [na, nG]
        rho_c = np.conj(zn @ ph)                               # conj Σ Z e^{iGτ}
        g2 = g.gk_len ** 2
        nz = g2 > 1e-20
        pref = 2.0 * (2 * math.pi / uc.omega)                  # non-reduced G set
        scal = np.zeros_like(g2)
        scal[nz] = pref * np.exp(-g2[nz] / (4 * alpha)) / g2[nz]
        # scalar_part[a, G] = scal·Im[rho_c·e^{iGτ_a}]·Z_a
        s = scal[None, :] * (rho_c[None, :] * ph).imag * zn[:, None]
        f = s @ g.gkvec_cart                                   # [na, 3]

        r_cut = max(np.linalg.norm(uc.lattice, axis=1))
        invpi = 1.0 / math.pi
        for ia, ja, d, T, rc in uc.nearest_neighbours_full(r_cut):
            t = T.astype(np.float64) @ uc.lattice
            sp = (zn[ia] * zn[ja] / (d * d)
                  * (math.erfc(math.sqrt(alpha) * d) / d
                     + 2.0 * math.sqrt(alpha * invpi) * math.exp(-d * d * alpha)))
            f[ia] += sp * t
        return _sym_forces(ctx, f)

    def calc_forces_us(self) -> np.ndarray:
        ctx = self.ctx
        uc = ctx.unit_cell
        out = np.zeros((uc.num_atoms, 3))
        dm = getattr(self.density, "density_matrix", None)
        if not ctx.has_aug or dm is None:
            return out
        gc = self._gcart()
        nch = 4 if ctx.nc_magnetism else ctx.num_spins
        # potential spin components matching dm_aux channels (force.cpp:428)
        comps = [self.potential.veff_g]
        if ctx.nc_magnetism:
            comps += [self.potential.bvec_g[0], self.potential.bvec_g[1],
                      self.potential.bvec_g[2]]
        elif ctx.num_spins == 2:
            comps += [self.potential.bz_g]
        for lab, at in uc.atom_types.items():
            if not (at.augment and at.num_beta):
                continue
            aug = ctx.aug_op(lab)
            ph = ctx.phase_pos(lab)                            # [na, nG]
            d = dm[lab]                                        # [na,nbf,nbf,nsp]
            nbf = aug.nbf
            xi1_idx, xi2_idx = [], []
            for xi2 in range(nbf):
                for xi1 in range(xi2 + 1):
                    xi1_idx.append(xi1)
                    xi2_idx.append(xi2)
            dpk = d[:, xi2_idx, xi1_idx, :]                    # [na, npack, nsp]
            chans = [dpk[..., 0].real] if dpk.shape[-1] == 1 else \
                [(dpk[..., 0] + dpk[..., 1]).real,
                 (dpk[..., 0] - dpk[..., 1]).real]
            if ctx.nc_magnetism:
                chans = [(dpk[..., 0] + dpk[..., 1]).real,
                         2.0 * dpk[..., 2].real,
                         -2.0 * dpk[..., 2].imag,
                         (dpk[..., 0] - dpk[..., 1]).real]
            ia_list = uc.atoms_of_type(lab)
            sw = aug.sym_weight.to(torch.float64)
            for s_idx, dm_aux in enumerate(chans):
                vg = comps[s_idx]
                for x in range(3):
                    vx = (-1j) * gc[:, x].to(ctx.dtype) * vg   # [nG]
                    v_at = ph * vx[None, :]                    # [na, nG]
                    # tmp[i, a] = Re Σ_G conj(Q_i(G))·v_at[a, G]
                    tmp = la.inner(aug.q_pw, v_at).real        # [npack, na]
                    val = uc.omega * torch.einsum(
                        "q,aq,qa->a", sw, dm_aux.real.to(torch.float64), tmp)
                    for i, ia in enumerate(ia_list):
                        out[ia, x] += float(val[i])
        return _sym_forces(ctx, out)

    def calc_forces_nonloc(self) -> np.ndarray:
        """−2 Σ Re[conj(⟨β|ψ⟩)(D − εQ)⟨∂β|ψ⟩] (non_local_functor.hpp)."""
        ctx = self.ctx
        uc = ctx.unit_cell
        if any(at.spin_orbit for at in uc.atom_types.values()):
            raise NotImplementedError(
                "forces with spin-orbit coupling are not supported "
                "(matches reference, non_local_functor.hpp:93)")
        if ctx.nc_magnetism:
            raise NotImplementedError("nonlocal forces: collinear only")
        out = np.zeros((uc.num_atoms, 3))
        min_occ = 1e-14
        from .hamiltonian import HamiltonianK

        for kp in self.kset:
            hk = HamiltonianK(self.h0, kp)
            bp = hk.bp
            if bp.num_beta_total == 0:
                continue
            gkc = torch.from_numpy(kp.gkvec.gkvec_cart).to(ctx.device)
            # per-atom block bounds
            offs = bp.atom_offsets + [bp.num_beta_total]
            for ispn in range(ctx.num_spins):
                occ = kp.occ[ispn]
                sel = np.nonzero(occ > min_occ)[0]
                if len(sel) == 0:
                    continue
                psi = kp.psi[ispn][torch.from_numpy(sel).to(ctx.device)].contiguous()
                eps = torch.from_numpy(kp.eigvals[ispn][sel]).to(ctx.device)
                w = torch.from_numpy(occ[sel] * kp.weight).to(ctx.device)
                bphi = bp.inner(psi)                           # [nbf, nocc]
                D = hk.D[ispn]                                 # [nbf, nbf]
                M = D @ bphi.conj()                            # [nbf, nocc]
                if hk.Q is not None:
                    N = hk.Q @ bphi.conj()
                else:
                    N = None
                for x in range(3):
                    dbeta = bp.beta_t * ((-1j) * gkc[:, x].to(ctx.dtype))[None, :]
                    bgphi = la.inner(dbeta.contiguous(), psi)  # [nbf, nocc]
                    P = (bgphi * M)                            # Σ_j D conj(bphi)
                    val = -2.0 * torch.einsum("n,in->i", w.to(torch.float64),
                                              P.real)
                    if N is not None:
                        P2 = (bgphi * N)
                        val = val + 2.0 * torch.einsum(
                            "n,in->i", (w * eps).to(torch.float64), P2.real)
                    v = val.cpu().numpy()
                    for ia in range(uc.num_atoms):
                        out[ia, x] += v[offs[ia]:offs[ia + 1]].sum()
        comm = get_comm()
        if comm.active:
            t = torch.from_numpy(out)
            comm.allreduce_(t)
            out = t.numpy()
        return _sym_forces(ctx, out)

    def calc_forces_hubbard(self) -> np.ndarray:
        """−Re Σ_k [V_loc·dn + Σ_pairs e^{−2πik·T}·conj(dn)·V_nl]
        (calc_forces_hubbard + hubbard_force_add_k_contribution_collinear,
        force.cpp:270-303, :591-653)."""
        ctx = self.ctx
        hub = ctx.hubbard
        uc = ctx.unit_cell
        out = np.zeros((uc.num_atoms, 3))
        if hub is None or hub.num_wf == 0:
            return out
        if ctx.nc_magnetism:
            raise NotImplementedError("Hubbard forces: collinear only "
                                      "(matches reference)")
        from .hamiltonian import HamiltonianK
        from .hubbard import compute_occupancies_derivatives

        nsp = ctx.num_spins
        for kp in self.kset:
            hk = HamiltonianK(self.h0, kp)
            dn = compute_occupancies_derivatives(hub, kp, hk)
            d = torch.zeros(3, uc.num_atoms, dtype=dn.dtype, device=dn.device)
            for il, (ia1, o) in enumerate(hub.levels):
                if not hub.level_use[il]:
                    continue
                off = hub.offsets[il]
                mm = 2 * o.l + 1
                for ispn in range(nsp):
                    d += torch.einsum(
                        "ab,baxj->xj", hub.um[il][..., ispn],
                        dn[off:off + mm, off:off + mm, ispn])
            for i, p in enumerate(hub.nonlocal_pairs):
                z1 = complex(np.exp(-2j * math.pi
                                    * float(np.dot(p.T, kp.k_frac))))
                a1 = hub._find_level(p.ia, p.n1, p.il)
                a2 = hub._find_level(p.ja, p.n2, p.jl)
                o1, o2 = hub.offsets[a1], hub.offsets[a2]
                mi, mj = 2 * p.il + 1, 2 * p.jl + 1
                for ispn in range(nsp):
                    d += (z1 * torch.einsum(
                        "ab,baxj->xj", hub.um_nl[i][..., ispn],
                        dn[o2:o2 + mj, o1:o1 + mi, ispn].conj())).real \
                        .to(d.dtype)
            out -= d.real.cpu().numpy().T
        comm = get_comm()
        if comm.active:
            t = torch.from_numpy(out)
            comm.allreduce_(t)
            out = t.numpy()
        return _sym_forces(ctx, out)

    # -- totals ------------------------------------------------------------

    def calc_forces_total(self, add_scf_corr: bool = True) -> dict:
        parts = {
            "vloc": self.calc_forces_vloc(),
            "us": self.calc_forces_us(),
            "nonloc": self.calc_forces_nonloc(),
            "core": self.calc_forces_core(),
            "ewald": self.calc_forces_ewald(),
        }
        if add_scf_corr:
            parts["scf_corr"] = self.calc_forces_scf_corr()
        parts["hubbard"] = self.calc_forces_hubbard()
        parts["total"] = sum(parts.values())
        return parts
