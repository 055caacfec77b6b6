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

from .core import ylm as ylm_mod
from .core.radial import RadialIntegrals


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

        # assemble per atom with phase e^{-i(G+k)·τ}
        blocks = []
        self.atom_offsets = []
        self.atom_nbf = []
        off = 0
        mk = (g.miller + g.k_frac).astype(np.float64)  # (G+k) in frac recip
        for ia, (lab, tau) in enumerate(uc.atoms):
            nb = uc.atom_types[lab].num_beta_lm
            self.atom_offsets.append(off)
            self.atom_nbf.append(nb)
            off += nb
            if nb == 0:
                continue
            phase = np.exp(-2j * math.pi * (mk @ tau))  # [nGk]
            blocks.append(cols_t[lab] * phase[:, None])
        self.num_beta_total = off
        if blocks:
            self.beta = torch.from_numpy(np.concatenate(blocks, axis=1)).to(dev)
        else:
            self.beta = torch.zeros(len(glen), 0, dtype=ctx.dtype, device=dev)

    def inner(self, psi: torch.Tensor) -> torch.Tensor:
        """⟨β|ψ⟩ = β^H ψ: [nbf_tot, nb] (zgemm; reference
        non_local_operator_base.hpp:130-199 inner_prod_beta)."""
        return self.beta.conj().T @ psi.T  # psi [nb, nGk] -> [nbf, nb]


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
        if nsp == 1:
            self.veff_r_coarse = [veff_c]
        else:
            bz_c = to_coarse(potential.bz_g)
            # spin 0 = up: V + Bz; spin 1 = dn: V − Bz (non_local_operator.cpp:236-238)
            self.veff_r_coarse = [veff_c + bz_c, veff_c - bz_c]
        uc = ctx.unit_cell
        # per-atom, per-spin D matrices [nbf, nbf] (numpy, real)
        self.d_atom = [[None] * uc.num_atoms for _ in range(nsp)]
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
                dt1 = dints(potential.bz_g) if nsp == 2 else None
                nbf = aug.nbf
                paw = ctx.paw
                for i, ia in enumerate(ia_list):
                    # unpack packed Q·V integrals per component
                    comps = [np.zeros((nbf, nbf)) for _ in range(nsp if nsp == 2 else 1)]
                    d0 = np.zeros((nbf, nbf))
                    d1 = np.zeros((nbf, nbf)) if nsp == 2 else None
                    for xi2 in range(nbf):
                        for xi1 in range(xi2 + 1):
                            idx12 = xi2 * (xi2 + 1) // 2 + xi1
                            d0[xi1, xi2] = d0[xi2, xi1] = dt0[idx12, i]
                            if nsp == 2:
                                d1[xi1, xi2] = d1[xi2, xi1] = dt1[idx12, i]
                    if paw is not None and ia in paw.dij:
                        d0 = d0 + paw.dij[ia][..., 0]
                        if nsp == 2:
                            d1 = d1 + paw.dij[ia][..., 1]
                    for ispn in range(nsp):
                        d = np.array(ions)
                        d += d0 if nsp == 1 else (d0 + d1 if ispn == 0 else d0 - d1)
                        self.d_atom[ispn][ia] = d
            else:
                for ia in ia_list:
                    for ispn in range(nsp):
                        self.d_atom[ispn][ia] = ions

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
        self.D = [block_diag_matrix(ctx, self.bp,
                                    lambda ia, s=s: h0.d_atom[s][ia])
                  for s in range(ctx.num_spins)]
        self.Q = None  # USPP overlap augmentation (else S = I)
        if ctx.has_aug:
            uc = ctx.unit_cell

            def qmat(ia):
                lab = uc.atoms[ia][0]
                at = uc.atom_types[lab]
                return ctx.aug_op(lab).q_mtrx if at.augment else None

            self.Q = block_diag_matrix(ctx, self.bp, qmat)
        self.ekin = 0.5 * kp.gkvec.gk2_t  # [nGk] float64

    def h_diag(self, ispn: int = 0) -> torch.Tensor:
        """Diagonal of H for the Davidson preconditioner
        (reference get_h_o_diag_pw, hamiltonian_k.cpp:67)."""
        v0 = float(self.h0.veff_r_coarse[ispn].mean())
        d = self.ekin + v0
        if self.bp.num_beta_total:
            # sum_{ij} conj(β_i(G)) D_ij β_j(G) per G
            bD = self.bp.beta @ self.D[ispn].T  # [nGk, nbf]
            d = d + torch.einsum("gi,gi->g", bD, self.bp.beta.conj()).real
        return d

    def o_diag(self) -> torch.Tensor:
        d = torch.ones_like(self.ekin)
        if self.Q is not None:
            bQ = self.bp.beta @ self.Q.T
            d = d + torch.einsum("gi,gi->g", bQ, self.bp.beta.conj()).real
        return d

    def apply_h_s(self, psi: torch.Tensor, ispn: int = 0):
        """psi [nb, nGk] -> (hpsi, spsi). spsi is None for NC (S = I)."""
        kp = self.kp
        # fused local operator: FFT⁻¹ → ×V_eff(r) → FFT → +½|G+k|²ψ
        hpsi = kp.fft.apply_veff_kinetic(psi, self.h0.veff_r_coarse[ispn],
                                         self.ekin)

        spsi = None
        if self.bp.num_beta_total:
            bphi = self.bp.inner(psi)           # [nbf, nb]
            hpsi += (self.bp.beta @ (self.D[ispn] @ bphi)).T
            if self.Q is not None:
                spsi = psi + (self.bp.beta @ (self.Q @ bphi)).T
        if self.ctx.hubbard is not None:
            self.ctx.hubbard.apply(kp, self, psi, hpsi, ispn)
        return hpsi, spsi
