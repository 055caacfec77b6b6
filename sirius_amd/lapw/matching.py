"""APW matching coefficients A_{lm,nu}(G+k).

Reference behavior: src/lapw/matching_coefficients.hpp — matching of
plane waves to the AW radial functions at the MT boundary:

  A_{L nu}(G) = [du^{(j)}_{l nu}/dr^j |_R]^{-1}_{nu j}
                 * d^j j_l(|G+k| r)/dr^j |_R
                 * (4 pi / sqrt(Omega)) i^l e^{i(G+k) tau} Y*_L(G+k)

The per-(l) inverse derivative matrices come from the atom symmetry
class surface derivatives.
"""

from __future__ import annotations

import math

import numpy as np
import torch

from ..core.ylm import ylm as _ylm, lmmax as _lmmax
from ..core.radial import sbessel


def sbessel_deriv_at_R(lmax: int, glen: np.ndarray, R: float) -> np.ndarray:
    """j_l(gR) and first two r-derivatives at R: [3, nG, lmax+1]."""
    ng = len(glen)
    x = np.outer(glen, np.ones(1)) * R  # gR
    jl = np.empty((ng, lmax + 2))
    for l in range(lmax + 2):
        jl[:, l] = sbessel(l, glen * R)
    out = np.zeros((3, ng, lmax + 1))
    for l in range(lmax + 1):
        out[0, :, l] = jl[:, l]
        out[1, :, l] = -jl[:, l + 1] * glen + (l / R) * jl[:, l]
        out[2, :, l] = (2.0 * glen * jl[:, l + 1] / R
                        + ((l - 1) * l - (glen * R) ** 2) * jl[:, l] / R ** 2)
    return out


class MatchingCoefficients:
    """Per-k-point matching coefficient generator."""

    def __init__(self, ctx, gkvec):
        self.ctx = ctx
        self.gkvec = gkvec
        uc = ctx.unit_cell
        lmax_apw = max(at.lmax_apw for at in uc.atom_types.values())
        self.lmax_apw = lmax_apw
        glen = gkvec.gk_len
        gk = gkvec.gkvec_cart
        r = np.linalg.norm(gk, axis=1)
        with np.errstate(invalid="ignore", divide="ignore"):
            theta = np.where(r > 1e-12, np.arccos(np.clip(gk[:, 2] / np.maximum(r, 1e-300), -1, 1)), 0.0)
            phi = np.where(r > 1e-12, np.arctan2(gk[:, 1], gk[:, 0]), 0.0)
        self.gkvec_ylm = _ylm(lmax_apw, theta, phi)          # [nG, lmmax]
        # i^l (4pi/sqrt(omega)) j_l^{(dm)}(|G+k|R) per atom type
        self.alm_b = {}
        f = 4 * math.pi / math.sqrt(uc.omega)
        for lab, at in uc.atom_types.items():
            jd = sbessel_deriv_at_R(at.lmax_apw, glen, at.rmt)  # [3, nG, l]
            z = np.array([1j ** l for l in range(at.lmax_apw + 1)])
            self.alm_b[lab] = jd * (f * z)[None, None, :]      # complex

    def generate(self, ia: int, asc, conjugate: bool = False) -> torch.Tensor:
        """A[igk, xi] for atom ia (xi over the AW basis of its type).

        conjugate=True returns conj(phase * zt) * Ylm — the 'row' variant
        (reference generate<true>), else phase * zt * conj(Ylm)."""
        uc = self.ctx.unit_cell
        lab, pos = uc.atoms[ia]
        at = uc.atom_types[lab]
        ng = self.gkvec.num_gvec
        phase = np.exp(2j * math.pi *
                       ((self.gkvec.miller + self.gkvec.k_frac) @ pos))
        alm_b = self.alm_b[lab]

        # inverse surface-derivative matrices per l
        Ainv = []
        for l in range(at.lmax_apw + 1):
            naw = at.aw_order(l)
            A = np.empty((naw, naw))
            for order in range(naw):
                idxrf = at.rf_index(l, order)
                for dm in range(naw):
                    A[dm, order] = asc.sd[dm, idxrf]
            Ainv.append(np.linalg.inv(A))

        # vectorized: zt[(l,nu)] = sum_dm alm_b[dm,:,l] Ainv_l[nu,dm]
        nxa = at.mt_aw_basis_size
        key = "_xi_maps"
        maps = getattr(at, key, None)
        if maps is None:
            lm_of = np.array([b[2] for b in at.indexb[:nxa]])
            l_of = np.array([b[0] for b in at.indexb[:nxa]])
            nu_of = np.array([b[3] for b in at.indexb[:nxa]])
            maps = (lm_of, l_of, nu_of)
            setattr(at, key, maps)
        lm_of, l_of, nu_of = maps
        max_nu = max(at.aw_order(l) for l in range(at.lmax_apw + 1))
        zt_ln = np.zeros((ng, at.lmax_apw + 1, max_nu), dtype=np.complex128)
        for l in range(at.lmax_apw + 1):
            naw = at.aw_order(l)
            # [ng, dm] @ [dm, nu] -> [ng, nu]
            zt_ln[:, l, :naw] = np.ascontiguousarray(
                alm_b[:naw, :, l].T) @ Ainv[l].T
        zt = zt_ln[:, l_of, nu_of]                    # [ng, nxa]
        yl = self.gkvec_ylm[:, lm_of]
        if conjugate:
            alm = np.conj(phase[:, None] * zt) * yl
        else:
            alm = phase[:, None] * zt * np.conj(yl)
        return torch.from_numpy(alm)
