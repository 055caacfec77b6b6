"""Hamiltonian: local operator, beta projectors, non-local D/Q operators.

Reference behavior:
- Local_operator (src/hamiltonian/local_operator.cpp:23/:275): map V_eff
  from fine to coarse G grid, apply per band as FFT→multiply→FFT plus
  kinetic ½|G+k|² (GPU twins: mul_by_veff_*, add_to_hphi_pw in
  local_operator.cu).
- Beta_projectors (src/beta_projectors/beta_projectors.hpp:26-72):
  β_t(G+k) = (-i)^l (4π/√Ω) R_lm(Ĝ+k) f_l(|G+k|), per atom ×
  e^{-i(G+k)·τ} (GPU twin: create_beta_gk.cu).
- Non_local_operator D/Q apply (src/hamiltonian/non_local_operator_base.hpp:130):
  hphi += β · (D · ⟨β|φ⟩); USPP: sphi += β · (Q · ⟨β|φ⟩).

MI355X design: the whole apply is batched torch on device — the FFT pair
via rocFFT with fused pack/unpack HIP kernels (sirius_amd.ops), β inner
products as zgemm (rocBLAS), the small per-atom D·bp contraction as one
block-diagonal matmul over all atoms at once instead of the reference's
per-atom streamed gemms.
"""

from __future__ import annotations

import math

import numpy as np
import torch

from .core import la
from .core import ylm as ylm_mod


class BetaProjectors:
    """<G+k|β> for all projectors of all atoms: device matrix [nGk, nbf_tot].

    Also holds the block-diagonal D (and Q) matrices over atoms.
    """

    def __init__(self, ctx, kp):
        self.ctx = ctx
        uc = ctx.unit_cell
        g = kp.gkvec
        dev = ctx.device
        # radial form factors per type at the exact |G+k| values
        gk = g.gkvec_cart
        glen = g.gk_len
        r_, theta, phi = ylm_mod.spherical_coords(gk)
        lmax = max((b.l for at in uc.atom_types.values() for b in at.beta), default=0)
        rl = ylm_mod.rlm(lmax, theta, phi)  # [nGk, lmmax]

        cols_t = {}  # per type: [nGk, nbf_t] complex
        for lab, at in uc.atom_types.items():
            if at.num_beta == 0:
                cols_t[lab] = np.zeros((len(glen), 0), dtype=np.complex128)
                continue
            fl = ctx.ri.beta(lab)(glen)  # [n_beta_radial, nGk]
            cols = []
            for (irf, l, m) in at.beta_lm_index():
                z = (-1j) ** l * (4 * math.pi / math.sqrt(uc.omega))
                cols.append(z * rl[:, ylm_mod.lm_index(l, m)] * fl[irf])
            cols_t[lab] = np.stack(cols, axis=1)

        # assemble per atom with phase e^{-i(G+k)·τ}.  The per-TYPE column
        # blocks above are small ([nGk, nbf_type], computed once); the
        # per-ATOM phase expansion is O(nGk x nbf_total) and runs on
        # device through the beta_phase kernel (create_beta_gk.cu twin;
        # was a host numpy loop that dominated setup at many atoms).
        self.atom_offsets = []
        self.atom_nbf = []
        type_row_off = {}
        off_t = 0
        for lab in uc.type_labels:
            type_row_off[lab] = off_t
            off_t += uc.atom_types[lab].num_beta_lm
        off = 0
        for ia, (lab, tau) in enumerate(uc.atoms):
            nb = uc.atom_types[lab].num_beta_lm
            self.atom_offsets.append(off)
            self.atom_nbf.append(nb)
            off += nb
        self.num_beta_total = off
        ngk = len(glen)
        mk = (g.miller + g.k_frac).astype(np.float64)  # (G+k) in frac recip
        # stored TRANSPOSED [nbf_tot, nGk]: both ⟨β|ψ⟩ (Gram) and the
        # β·(D⟨β|ψ⟩) apply (transform) then run on the MFMA zgemm kernels
        # (ops/src/zgemm_gram.hip) with fully coalesced row-major access.
        if off == 0:
            self.beta_t = torch.zeros(0, ngk, dtype=ctx.dtype, device=dev)
        elif str(dev).startswith("cuda"):
            from . import ops

            ext = ops.get_ext(required=True)
            bt_types = np.concatenate(
                [cols_t[lab].T for lab in uc.type_labels], axis=0)
            bt_types_d = torch.from_numpy(
                np.ascontiguousarray(bt_types)).to(dev)
            mk_d = torch.from_numpy(np.ascontiguousarray(mk)).to(dev)
            tau_d = torch.from_numpy(
                uc.atom_positions_frac().copy()).to(dev)
            t_off = torch.tensor([type_row_off[lab] for lab, _ in uc.atoms],
                                 dtype=torch.int32, device=dev)
            a_nbf = torch.tensor(self.atom_nbf, dtype=torch.int32, device=dev)
            a_off = torch.tensor(self.atom_offsets, dtype=torch.int32,
                                 device=dev)
            self.beta_t = torch.empty(off, ngk, dtype=ctx.dtype, device=dev)
            ext.beta_phase(bt_types_d, mk_d, tau_d, t_off, a_nbf, a_off,
                           self.beta_t)
        else:
            blocks = []
            for ia, (lab, tau) in enumerate(uc.atoms):
                if uc.atom_types[lab].num_beta_lm == 0:
                    continue
                phase = np.exp(-2j * math.pi * (mk @ tau))  # [nGk]
                blocks.append(cols_t[lab] * phase[:, None])
            bt = np.ascontiguousarray(np.concatenate(blocks, axis=1).T)
            self.beta_t = torch.from_numpy(bt).to(dev)

    def inner(self, psi: torch.Tensor) -> torch.Tensor:
        """⟨β|ψ⟩ = β^H ψ: [nbf_tot, nb] (MFMA Gram kernel; reference
        non_local_operator_base.hpp:130-199 inner_prod_beta → SPLA)."""
        return la.inner(self.beta_t, psi)  # psi [nb, nGk] -> [nbf, nb]


def block_diag_matrix(ctx, bp: BetaProjectors, per_atom_matrix):
    """Assemble block-diagonal [nbf_tot, nbf_tot] operator.

    per_atom_matrix: callable ia -> [nbf, nbf] numpy (or None for zero)."""
    uc = ctx.unit_cell
    n = bp.num_beta_total
    out = torch.zeros(n, n, dtype=ctx.dtype, device=ctx.device)
    for ia in range(uc.num_atoms):
        o = bp.atom_offsets[ia]
        nb = bp.atom_nbf[ia]
        m = per_atom_matrix(ia)
        if nb and m is not None:
            out[o:o + nb, o:o + nb] = torch.from_numpy(np.ascontiguousarray(m)).to(ctx.device)
    return out


def d_matrix_lm(at) -> np.ndarray:
    """Expand the radial D_ion [nrb, nrb] to the lm-resolved [nbf, nbf]:
    D_{(i l m),(j l' m')} = D_ion[i,j] δ_ll' δ_mm' (NC/USPP without SO)."""
    idx = at.beta_lm_index()
    n = len(idx)
    out = np.zeros((n, n))
    for a, (ia_, la, ma) in enumerate(idx):
        for b, (ib_, lb, mb) in enumerate(idx):
            if la == lb and ma == mb:
                out[a, b] = at.d_ion[ia_, ib_]
    return out


class Hamiltonian0:
    """Potential-dependent, k-independent parts (reference hamiltonian.hpp:62):
    V_eff on the coarse real grid + D-operator matrices.

    D assembly for augmented (USPP) types follows
    generate_d_operator_matrix.cpp: D_a = Ω·Σ_G conj(Q_ij(G))·V_eff(G)·e^{iG·τ_a}
    + D^ion δ_{lm,lm'}; for NC types D = D^ion alone
    (non_local_operator.cpp:208-253)."""

    def __init__(self, ctx, potential, density=None):
        self.ctx = ctx
        self.potential = potential
        nsp = ctx.num_spins

        def to_coarse(fg):
            return ctx.fft_coarse.to_real(fg[ctx.coarse_to_fine]).real

        veff_c = to_coarse(potential.veff_g)
        self.v_ud_coarse = None
        if nsp == 1:
            self.veff_r_coarse = [veff_c]
        elif ctx.nc_magnetism:
            bz_c = to_coarse(potential.bvec_g[2])
            bx_c = to_coarse(potential.bvec_g[0])
            by_c = to_coarse(potential.bvec_g[1])
            self.veff_r_coarse = [veff_c + bz_c, veff_c - bz_c]
            # off-diagonal spin block (Bx − iBy); conj is the dn-up block
            self.v_ud_coarse = (bx_c - 1j * by_c).to(ctx.dtype)
        else:
            bz_c = to_coarse(potential.bz_g)
            # spin 0 = up: V + Bz; spin 1 = dn: V − Bz (non_local_operator.cpp:236-238)
            self.veff_r_coarse = [veff_c + bz_c, veff_c - bz_c]
        uc = ctx.unit_cell
        # per-atom, per-spin(-block) D matrices [nbf, nbf]
        nblk = 4 if ctx.nc_magnetism else nsp
        self.d_atom = [[None] * uc.num_atoms for _ in range(nblk)]
        for lab, at in uc.atom_types.items():
            ions = d_matrix_lm(at)
            ia_list = uc.atoms_of_type(lab)
            if at.augment and at.num_beta:
                aug = ctx.aug_op(lab)
                ph = ctx.phase_pos(lab)                    # [na, nG]

                def dints(fg):
                    va = fg[None, :] * ph
                    return ((aug.q_pw.conj() @ va.T).real * uc.omega).cpu().numpy()

                dt0 = dints(potential.veff_g)               # [nqlm, na]
                if ctx.nc_magnetism:
                    dtz = dints(potential.bvec_g[2])
                    dtx = dints(potential.bvec_g[0])
                    dty = dints(potential.bvec_g[1])
                elif nsp == 2:
                    dt1 = dints(potential.bz_g)
                else:
                    dt1 = None
                nbf = aug.nbf
                paw = ctx.paw
                # packed-triangle -> full unpack, vectorized (was an
                # O(nbf^2) python loop per atom per iteration)
                _x1, _x2 = np.triu_indices(nbf)
                _p12 = _x2 * (_x2 + 1) // 2 + _x1
                for i, ia in enumerate(ia_list):
                    def unpack(dt):
                        m = np.zeros((nbf, nbf))
                        m[_x1, _x2] = dt[_p12, i]
                        m[_x2, _x1] = dt[_p12, i]
                        return m

                    d0 = unpack(dt0)
                    if paw is not None and ia in paw.dij:
                        d0 = d0 + paw.dij[ia][..., 0]
                    if ctx.nc_magnetism:
                        dz = unpack(dtz)
                        dx = unpack(dtx)
                        dy = unpack(dty)
                        if at.spin_orbit:
                            from . import so as so_mod

                            fc = ctx.so_fcoef(lab)
                            blocks = so_mod.so_d_blocks(at, [d0, dz, dx, dy], fc)
                            for ib in range(4):
                                self.d_atom[ib][ia] = blocks[ib]
                            continue
                        self.d_atom[0][ia] = ions + d0 + dz        # up-up
                        self.d_atom[1][ia] = ions + d0 - dz        # dn-dn
                        self.d_atom[2][ia] = dx - 1j * dy          # up-dn
                        self.d_atom[3][ia] = dx + 1j * dy          # dn-up
                    elif nsp == 2:
                        d1 = unpack(dt1)
                        if paw is not None and ia in paw.dij:
                            d1 = d1 + paw.dij[ia][..., 1]
                        self.d_atom[0][ia] = ions + d0 + d1
                        self.d_atom[1][ia] = ions + d0 - d1
                    else:
                        self.d_atom[0][ia] = ions + d0
            else:
                if ctx.nc_magnetism and at.spin_orbit and at.num_beta:
                    from . import so as so_mod

                    fc = ctx.so_fcoef(lab)
                    zero = np.zeros_like(ions)
                    blocks = so_mod.so_d_blocks(at, [zero] * 4, fc)
                    for ia in ia_list:
                        for ib in range(4):
                            self.d_atom[ib][ia] = blocks[ib]
                else:
                    for ia in ia_list:
                        for ib in range(nblk):
                            if ib < 2 or not ctx.nc_magnetism:
                                self.d_atom[ib][ia] = ions
                            else:
                                self.d_atom[ib][ia] = np.zeros_like(ions)

    def __call__(self, kp) -> "HamiltonianK":
        return HamiltonianK(self, kp)


class HamiltonianK:
    """Per-k-point Hamiltonian application (reference hamiltonian.hpp:172,
    apply_h_s :457-515)."""

    def __init__(self, h0: Hamiltonian0, kp):
        self.h0 = h0
        self.ctx = h0.ctx
        self.kp = kp
        if kp.beta is None:
            kp.beta = BetaProjectors(self.ctx, kp)
        self.bp = kp.beta
        ctx = self.ctx
        nblk = 4 if ctx.nc_magnetism else ctx.num_spins
        self.D = [block_diag_matrix(ctx, self.bp,
                                    lambda ia, s=s: h0.d_atom[s][ia])
                  for s in range(nblk)]
        self.Q = None  # USPP overlap augmentation (else S = I)
        self.Q_blocks = None  # nc/SO spin blocks [Q00, Q11, Q01, Q10]
        if ctx.has_aug:
            uc = ctx.unit_cell
            has_so = any(at.spin_orbit and at.augment
                         for at in uc.atom_types.values())

            def qmat(ia):
                lab = uc.atoms[ia][0]
                at = uc.atom_types[lab]
                return ctx.aug_op(lab).q_mtrx if at.augment else None

            if ctx.nc_magnetism and has_so:
                from . import so as so_mod

                def qblk(ia, ib):
                    lab = uc.atoms[ia][0]
                    at = uc.atom_types[lab]
                    if not at.augment:
                        return None
                    q = ctx.aug_op(lab).q_mtrx
                    if at.spin_orbit:
                        return so_mod.so_q_blocks(at, q, ctx.so_fcoef(lab))[ib]
                    return q if ib < 2 else np.zeros_like(q)

                self.Q_blocks = [block_diag_matrix(ctx, self.bp,
                                                   lambda ia, b=b: qblk(ia, b))
                                 for b in range(4)]
                self.Q = self.Q_blocks[0]
            else:
                self.Q = block_diag_matrix(ctx, self.bp, qmat)
        self.ekin = 0.5 * kp.gkvec.gk2_t  # [nGk] float64

    def h_diag(self, ispn: int = 0) -> torch.Tensor:
        """Diagonal of H for the Davidson preconditioner
        (reference get_h_o_diag_pw, hamiltonian_k.cpp:67)."""
        def diag_for(s):
            v0 = float(self.h0.veff_r_coarse[s].mean())
            d = self.ekin + v0
            if self.bp.num_beta_total:
                bt = self.bp.beta_t
                bD = la.transform(self.D[s].transpose(0, 1), bt)  # [nbf, nGk]
                # Σ_i bD[i,g]·conj(bt[i,g]) without materializing conj(bt)
                d = d + (torch.einsum("ig,ig->g", bD.real, bt.real)
                         + torch.einsum("ig,ig->g", bD.imag, bt.imag))
            return d
        if self.ctx.nc_magnetism:
            return torch.cat([diag_for(0), diag_for(1)])
        return diag_for(ispn)

    def o_diag(self) -> torch.Tensor:
        if self.ctx.nc_magnetism and self.Q_blocks is not None:
            outs = []
            for b in (0, 1):
                d = torch.ones_like(self.ekin)
                bt = self.bp.beta_t
                bQ = la.transform(self.Q_blocks[b].transpose(0, 1), bt)
                d = d + (torch.einsum("ig,ig->g", bQ.real, bt.real)
                         + torch.einsum("ig,ig->g", bQ.imag, bt.imag))
                outs.append(d)
            return torch.cat(outs)
        d = torch.ones_like(self.ekin)
        if self.Q is not None:
            bt = self.bp.beta_t
            bQ = la.transform(self.Q.transpose(0, 1), bt)
            d = d + (torch.einsum("ig,ig->g", bQ.real, bt.real)
                     + torch.einsum("ig,ig->g", bQ.imag, bt.imag))
        if self.ctx.nc_magnetism:
            return torch.cat([d, d])
        return d

    def _fp32_tensors(self, ispn: int):
        """complex64/float32 copies of the apply tables for the fp32
        wave-function mode (reference precision_wf,
        dft_ground_state.cpp:269-304); cached per Hamiltonian0."""
        c = getattr(self, "_fp32_cache", None)
        if c is None:
            self._fp32_cache = c = {}
        if ispn not in c:
            c[ispn] = (self.h0.veff_r_coarse[ispn].to(torch.float32),
                       self.ekin.to(torch.float32),
                       self.bp.beta_t.to(torch.complex64)
                       if self.bp.num_beta_total else None,
                       self.D[ispn].to(torch.complex64),
                       self.Q.to(torch.complex64)
                       if self.Q is not None else None)
        return c[ispn]

    def apply_h_s(self, psi: torch.Tensor, ispn: int = 0):
        """psi [nb, nGk] (or [nb, 2·nGk] spinors) -> (hpsi, spsi).

        complex64 input runs the whole chain in fp32 (cFFT/cgemm via
        torch; the fp64 HIP kernels are bypassed)."""
        if self.ctx.nc_magnetism:
            return self._apply_h_s_nc(psi)
        kp = self.kp
        self.ctx.counters["local_operator_num_applied"] += psi.shape[0]
        fp32 = psi.dtype == torch.complex64
        if fp32:
            veff_r, ekin, beta_t, D, Q = self._fp32_tensors(ispn)
        else:
            veff_r, ekin = self.h0.veff_r_coarse[ispn], self.ekin
            beta_t, D, Q = self.bp.beta_t, self.D[ispn], self.Q
        # fused local operator: FFT⁻¹ → ×V_eff(r) → FFT → +½|G+k|²ψ
        hpsi = kp.fft.apply_veff_kinetic(psi, veff_r, ekin)

        spsi = None
        if self.bp.num_beta_total:
            bphi = la.inner(beta_t, psi)        # [nbf, nb]
            la.transform(D @ bphi, beta_t, out=hpsi, accumulate=True)
            if Q is not None:
                spsi = psi.clone()
                la.transform(Q @ bphi, beta_t, out=spsi, accumulate=True)
        if self.ctx.hubbard is not None:
            self.ctx.hubbard.apply(kp, self, psi, hpsi, ispn)
        return hpsi, spsi

    def _apply_h_s_nc(self, psi: torch.Tensor):
        """Spinor apply (hamiltonian.hpp:454-515 noncollinear branch):
        h↑ = (T+V+Bz)ψ↑ + (Bx−iBy)ψ↓ + β[D⁰⁰⟨β|ψ↑⟩ + D⁰¹⟨β|ψ↓⟩], etc."""
        kp = self.kp
        self.ctx.counters["local_operator_num_applied"] += 2 * psi.shape[0]
        ng = kp.num_gkvec
        up, dn = psi[:, :ng], psi[:, ng:]
        h0 = self.h0
        up_r = kp.fft.to_real(up)
        dn_r = kp.fft.to_real(dn)
        vud = h0.v_ud_coarse
        hup = kp.fft.to_pw(up_r * h0.veff_r_coarse[0] + dn_r * vud)
        hdn = kp.fft.to_pw(dn_r * h0.veff_r_coarse[1] + up_r * vud.conj())
        hup = hup + self.ekin * up
        hdn = hdn + self.ekin * dn
        sup = sdn = None
        if self.bp.num_beta_total:
            bu = self.bp.inner(up)
            bd = self.bp.inner(dn)
            bt = self.bp.beta_t
            la.transform(self.D[0] @ bu + self.D[2] @ bd, bt,
                         out=hup, accumulate=True)
            la.transform(self.D[1] @ bd + self.D[3] @ bu, bt,
                         out=hdn, accumulate=True)
            if self.Q_blocks is not None:
                sup = up.clone().contiguous()
                sdn = dn.clone().contiguous()
                la.transform(self.Q_blocks[0] @ bu + self.Q_blocks[2] @ bd,
                             bt, out=sup, accumulate=True)
                la.transform(self.Q_blocks[1] @ bd + self.Q_blocks[3] @ bu,
                             bt, out=sdn, accumulate=True)
            elif self.Q is not None:
                sup = up.clone().contiguous()
                sdn = dn.clone().contiguous()
                la.transform(self.Q @ bu, bt, out=sup, accumulate=True)
                la.transform(self.Q @ bd, bt, out=sdn, accumulate=True)
        hpsi = torch.cat([hup, hdn], dim=1)
        spsi = torch.cat([sup, sdn], dim=1) if sup is not None else None
        return hpsi, spsi
