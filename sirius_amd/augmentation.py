"""Ultrasoft/PAW augmentation operator Q_ij(G).

Reference behavior: src/density/augmentation_operator.{hpp,cpp}
(generate_pw_coeffs, augmentation_operator.cpp:19-120; GPU twin
aug_op_pw_coeffs kernel):

  Q_{ξ1ξ2}(G) = (4π/Ω) Σ_{l3m3} (-i)^{l3} R_{l3m3}(Ĝ)
                ⟨j_{l3}|Q^{l3}_{rf1,rf2}⟩(|G|) · G^{rrr}_{lm1,lm2,l3m3}

with ⟨j|Q⟩ the aug radial integral at rpow=0 (the UPF file stores r²Q),
real-harmonic Gaunts, and q_mtrx = Ω·Q(G=0) the overlap charges for the
S operator. Packed storage: idx12 = ξ2(ξ2+1)/2 + ξ1 for ξ1 ≤ ξ2;
sym_weight 1 (diag) / 2 (offdiag).
"""

from __future__ import annotations

import math

import numpy as np
import torch

from .core import ylm as ylm_mod
from .core.gaunt import gaunt_rrr


def packed_index(xi1: int, xi2: int) -> int:
    return xi2 * (xi2 + 1) // 2 + xi1


class AugmentationOperator:
    """Per-atom-type Q_ij(G) on the fine G sphere (device tensor).

    With gvec_deriv = ν ∈ {0,1,2} builds instead |G|·dQ_ij(G)/dG_ν with
    the SAME 4π/Ω prefactor as the base operator
    (generate_pw_coeffs_gvec_deriv, augmentation_operator.cpp:125-180 —
    there rlm_dq is the r-scaled gradient, divide_by_r=false; the
    reference's deriv drops the 1/Ω and restores it in the us stress)."""

    def __init__(self, ctx, at, gvec_deriv: int | None = None):
        self.ctx = ctx
        self.at = at
        uc = ctx.unit_cell
        g = ctx.gvec_fine
        idxb = at.beta_lm_index()           # [(idxrf, l, m)]
        nbf = len(idxb)
        self.nbf = nbf
        nqlm = nbf * (nbf + 1) // 2
        lmax_beta = max((b.l for b in at.beta), default=0)
        lmax3 = 2 * lmax_beta
        lmmax3 = ylm_mod.lmmax(lmax3)

        # aug radial integrals on the G shells from the interpolation table
        # (reference: Radial_integrals_aug at settings.nprii_aug resolution)
        shells = g.shell_len
        ri = ctx.ri.aug(at.label)(shells)   # [n_rf_pairs, lmax3+1, nshell]

        # R_lm(G-hat) for l<=lmax3, and Gaunt table
        if gvec_deriv is None:
            _, theta, phi = ylm_mod.spherical_coords(g.g_cart)
            rlm3 = ylm_mod.rlm(lmax3, theta, phi)        # [nG, lmmax3]
        else:
            rlm3, rlm3_dg = ylm_mod.rlm_and_cart_grad(lmax3, g.g_cart)
            glen_g = g.gk_len
            ri_dq = ctx.ri.aug_djl(at.label)(shells)
            nu = gvec_deriv
        lmax_b = lmax_beta
        gc = gaunt_rrr(lmax_b, lmax_b, lmax3)            # [lmmax1, lmmax2, lmmax3]

        lm_of = [ylm_mod.lm_index(l, m) for (_, l, m) in idxb]
        rf_of = [irf for (irf, _, _) in idxb]

        # assemble Q packed [nqlm, nG]
        q_pw = np.zeros((nqlm, g.num_gvec), dtype=np.complex128)
        l_by_lm = np.concatenate([[l] * (2 * l + 1) for l in range(lmax3 + 1)])
        pref = 4 * math.pi / uc.omega
        shell_idx = g.shell_of_g
        for xi2 in range(nbf):
            for xi1 in range(xi2 + 1):
                idx12 = packed_index(xi1, xi2)
                rf1, rf2 = rf_of[xi1], rf_of[xi2]
                pair = packed_index(min(rf1, rf2), max(rf1, rf2))
                lm1, lm2 = lm_of[xi1], lm_of[xi2]
                acc = np.zeros(g.num_gvec, dtype=np.complex128)
                for lm3 in range(lmmax3):
                    gcv = gc[lm1, lm2, lm3]
                    if abs(gcv) < 1e-14:
                        continue
                    l3 = int(l_by_lm[lm3])
                    if gvec_deriv is None:
                        acc += ((-1j) ** l3 * gcv) * rlm3[:, lm3] \
                            * ri[pair, l3][shell_idx]
                    else:
                        # v = conj(i^l3)·(|G|·dRlm/dG_ν·f + Rlm·f'·G_ν)
                        acc += ((-1j) ** l3 * gcv) * (
                            glen_g * rlm3_dg[:, nu, lm3] * ri[pair, l3][shell_idx]
                            + rlm3[:, lm3] * ri_dq[pair, l3][shell_idx]
                            * g.g_cart[:, nu])
                q_pw[idx12] = pref * acc
        self.q_pw = torch.from_numpy(q_pw).to(ctx.device)

        ig0 = g.index_of_zero()
        self.q_mtrx = np.zeros((nbf, nbf))
        if ig0 >= 0:
            for xi2 in range(nbf):
                for xi1 in range(xi2 + 1):
                    v = uc.omega * q_pw[packed_index(xi1, xi2), ig0].real
                    self.q_mtrx[xi1, xi2] = self.q_mtrx[xi2, xi1] = v

        sw = np.empty(nqlm)
        for xi2 in range(nbf):
            for xi1 in range(xi2 + 1):
                sw[packed_index(xi1, xi2)] = 1.0 if xi1 == xi2 else 2.0
        self.sym_weight = torch.from_numpy(sw).to(ctx.device)

    @property
    def nqlm(self) -> int:
        return self.nbf * (self.nbf + 1) // 2
