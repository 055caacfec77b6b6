"""Gaunt coefficients of real spherical harmonics.

Reference behavior: src/core/sht/gaunt.hpp:46 (Gaunt_coefficients,
SHT::gaunt_rrr): G_{l1m1,l2m2,l3m3} = ∫ R_{l1m1}(Ω) R_{l2m2}(Ω) R_{l3m3}(Ω) dΩ.

Computed here by exact quadrature: Gauss-Legendre in cosθ × uniform φ —
exact for band-limited integrands (degree ≤ l1+l2+l3), no Wigner-3j
bookkeeping. Cached per (l1max, l2max, l3max).
"""

from __future__ import annotations

from functools import lru_cache

import numpy as np

from . import ylm as ylm_mod


@lru_cache(maxsize=16)
def gaunt_rrr(l1max: int, l2max: int, l3max: int) -> np.ndarray:
    """Dense real-Gaunt table [lmmax1, lmmax2, lmmax3]."""
    ltot = l1max + l2max + l3max
    nth = ltot // 2 + 2
    nph = ltot + 2
    x, wx = np.polynomial.legendre.leggauss(nth)
    theta = np.arccos(x)
    phi = np.arange(nph) * 2 * np.pi / nph
    wphi = 2 * np.pi / nph
    tt, pp = np.meshgrid(theta, phi, indexing="ij")
    ww = (wx[:, None] * wphi * np.ones(nph)[None, :]).reshape(-1)
    tt = tt.reshape(-1)
    pp = pp.reshape(-1)
    r1 = ylm_mod.rlm(l1max, tt, pp)          # [npts, lmmax1]
    r2 = ylm_mod.rlm(l2max, tt, pp) if l2max != l1max else r1
    r3 = ylm_mod.rlm(l3max, tt, pp) if l3max not in (l1max, l2max) else (
        r1 if l3max == l1max else r2)
    # G = Σ_p w_p R1[p,i] R2[p,j] R3[p,k]
    return np.einsum("p,pi,pj,pk->ijk", ww, r1, r2, r3, optimize=True)
