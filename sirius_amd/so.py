"""Spin-orbit coupling support for pseudopotentials with j-resolved
beta projectors.

Reference behavior: Eq.18-19 of PRB 71, 115106 as implemented in
- f-coefficients: Atom_type::generate_f_coefficients (atom_type.cpp:1074-1125)
  with ClebschGordan / calculate_U_sigma_m (sht.cpp:113-202);
- D matrix rotation: D_operator::initialize SO branch
  (non_local_operator.cpp:125-200), in the up/down basis with blocks
  s_idx = {{0,3},{2,1}};
- Q matrix rotation: Q_operator::initialize (non_local_operator.cpp:290-340).

The R↔Y transformation matrix is computed numerically from this package's
own harmonics (exact quadrature), guaranteeing internal consistency.
"""

from __future__ import annotations

import math
from functools import lru_cache

import numpy as np

from .core import ylm as ylm_mod


USE_ANALYTIC_TABLE = False


def _analytic_table(l: int) -> np.ndarray:
    """SIRIUS's analytic ylm_dot_rlm-based table (sht.hpp:355-389)."""
    isq = 1.0 / math.sqrt(2.0)

    def ylm_dot_rlm(m1, m2):
        if not (m1 == m2 or m1 == -m2):
            return 0.0
        if m1 == 0:
            return 1.0
        if m1 < 0:
            return -1j * isq if m2 < 0 else (-1.0) ** m2 * isq
        return (-1.0) ** m1 * 1j * isq if m2 < 0 else isq

    M = np.zeros((2 * l + 1, 2 * l + 1), dtype=np.complex128)
    for m in range(-l, l + 1):
        for mp in range(-l, l + 1):
            M[m + l, mp + l] = np.conj(ylm_dot_rlm(mp, m))
    return M


@lru_cache(maxsize=8)
def rlm_dot_ylm_table(l: int) -> np.ndarray:
    if USE_ANALYTIC_TABLE:
        return _analytic_table(l)
    return _numeric_table(l)


def _numeric_table(l: int) -> np.ndarray:
    """M[m, mp] = ⟨R_{l m}|Y_{l mp}⟩ = ∫ R_lm Y*_lmp dΩ, shape [2l+1, 2l+1]."""
    nth = 2 * l + 4
    nph = 4 * l + 4
    x, wx = np.polynomial.legendre.leggauss(nth)
    theta = np.arccos(x)
    phi = np.arange(nph) * 2 * math.pi / nph
    tt, pp = np.meshgrid(theta, phi, indexing="ij")
    w = np.broadcast_to(wx[:, None] * (2 * math.pi / nph), tt.shape).reshape(-1)
    tt = tt.reshape(-1)
    pp = pp.reshape(-1)
    R = ylm_mod.rlm(l, tt, pp)[:, l * l:(l + 1) ** 2]
    Y = ylm_mod.ylm(l, tt, pp)[:, l * l:(l + 1) ** 2]
    return np.einsum("p,pm,pn->mn", w, R, Y.conj())


def clebsch_gordan(l: int, j: float, mj: float, spin: int) -> float:
    """CG coefficient (sht.cpp:113-155)."""
    denom = math.sqrt(1.0 / (2 * l + 1))
    if abs(j - l - 0.5) < 1e-8:
        m = int(mj - 0.5)
        return denom * (math.sqrt(l + m + 1.0) if spin == 0 else math.sqrt(l - m))
    if abs(j - l + 0.5) < 1e-8:
        m = int(mj + 0.5)
        if m < 1 - l:
            return 0.0
        return denom * (math.sqrt(l - m + 1) if spin == 0 else -math.sqrt(l + m))
    raise ValueError("invalid j for l")


def u_sigma_m(l: int, j: float, mj2: int, mp: int, sigma: int) -> complex:
    """U^σ_{l j mj, m'} (sht.cpp:165-202); mj2 = 2·mj integer."""
    M = rlm_dot_ylm_table(l)

    def rdy(m, mp_):
        return M[m + l, mp_ + l]

    if abs(j - l - 0.5) < 1e-8:
        m1 = (mj2 - 1) >> 1
        if sigma == 0:
            return 0.0 if m1 < -l else rdy(m1, mp)
        return 0.0 if m1 + 1 > l else rdy(m1 + 1, mp)
    m1 = (mj2 + 1) >> 1
    if sigma == 0:
        return rdy(m1 - 1, mp)
    return rdy(m1, mp)


def f_coefficients(at) -> np.ndarray:
    """f^{σσ'}_{ξ1 ξ2} table [nbf, nbf, 2, 2] (atom_type.cpp:1074-1125)."""
    idx = at.beta_lm_index()   # (irf, l, m)
    jb = [at.beta[irf].j for irf, _, _ in idx]
    nbf = len(idx)
    f = np.zeros((nbf, nbf, 2, 2), dtype=np.complex128)
    for xi2, (irf2, l2, m2) in enumerate(idx):
        j2 = jb[xi2]
        for xi1, (irf1, l1, m1) in enumerate(idx):
            j1 = jb[xi1]
            if l1 != l2 or j1 is None or j2 is None or abs(j1 - j2) > 1e-8:
                continue
            jj1 = int(2 * j1 + 1e-8)
            for s1 in range(2):
                for s2 in range(2):
                    c = 0.0 + 0.0j
                    for mj2 in range(-jj1, jj1 + 1, 2):
                        c += (u_sigma_m(l1, j1, mj2, m1, s1)
                              * clebsch_gordan(l1, j1, mj2 / 2.0, s1)
                              * np.conj(u_sigma_m(l2, j2, mj2, m2, s2))
                              * clebsch_gordan(l2, j2, mj2 / 2.0, s2))
                    f[xi1, xi2, s1, s2] = c
    return f


# Pauli matrices in the (V, Bz, Bx, By) component order used by d_mtrx
PAULI = np.array([
    [[1, 0], [0, 1]],
    [[1, 0], [0, -1]],
    [[0, 1], [1, 0]],
    [[0, -1j], [1j, 0]],
], dtype=np.complex128)

# spin-block index: s_idx[sigma][sigma'] (non_local_operator.cpp:116)
S_IDX = [[0, 3], [2, 1]]


def so_d_blocks(at, d_alpha: list[np.ndarray], fcoef: np.ndarray) -> list[np.ndarray]:
    """Rotate component D matrices (V,Bz,Bx,By integrals) + D_ion into the
    up/down spin-block basis (D_operator::initialize SO branch).

    d_alpha: 4 matrices [nbf, nbf] (zeros when no augmentation).
    Returns [D00, D11, D01(up-dn), D10(dn-up)] complex [nbf, nbf].
    """
    idx = at.beta_lm_index()
    nbf = len(idx)
    jb = [at.beta[irf].j for irf, _, _ in idx]
    same = np.zeros((nbf, nbf), dtype=bool)
    for a, (irfa, la, _) in enumerate(idx):
        for b, (irfb, lb, _) in enumerate(idx):
            ja = jb[a] if jb[a] is not None else la + 0.5
            jb_ = jb[b] if jb[b] is not None else lb + 0.5
            same[a, b] = (la == lb) and abs(ja - jb_) < 1e-8

    d = np.stack(d_alpha)                       # [4, nbf, nbf]
    out = [np.zeros((nbf, nbf), dtype=np.complex128) for _ in range(4)]
    # result(xi1,xi2,σσ') = Σ_{x1p,x2p,α,σ1σ2} d[α,x1p,x2p] P[α,σ1,σ2]
    #                       f[xi1,x1p,σ,σ1] f[x2p,xi2,σ2,σ']
    # The x1p (x2p) sums are restricted to the SAME radial function as
    # xi1 (xi2) — compare_index_beta_functions demands equal idxrf, not
    # just equal (l, j) (non_local_operator.cpp:139-144, atom_type.hpp:1188)
    # — so cross-radial f elements (which the f table does contain for
    # equal (l,j)) must NOT enter the rotation.
    irf = np.array([irf_ for irf_, _, _ in idx])
    m_irf = (irf[:, None] == irf[None, :]).astype(np.float64)
    fm = fcoef * m_irf[:, :, None, None]
    for s in range(2):
        for sp in range(2):
            acc = np.zeros((nbf, nbf), dtype=np.complex128)
            for a in range(4):
                for s1 in range(2):
                    for s2 in range(2):
                        p = PAULI[a, s1, s2]
                        if p == 0:
                            continue
                        acc += p * (fm[:, :, s, s1] @ d[a] @ fm[:, :, s2, sp])
            out[S_IDX[s][sp]] += acc

    # ionic term: dion over same-am channels with single f factor
    for xi2, (irf2, l2, m2) in enumerate(idx):
        for xi1, (irf1, l1, m1) in enumerate(idx):
            if not same[xi1, xi2]:
                continue
            dion = at.d_ion[irf1, irf2]
            out[0][xi1, xi2] += dion * fcoef[xi1, xi2, 0, 0]
            out[1][xi1, xi2] += dion * fcoef[xi1, xi2, 1, 1]
            out[2][xi1, xi2] += dion * fcoef[xi1, xi2, 0, 1]
            out[3][xi1, xi2] += dion * fcoef[xi1, xi2, 1, 0]
    return out


def so_q_blocks(at, q_mtrx: np.ndarray, fcoef: np.ndarray) -> list[np.ndarray]:
    """Rotate the augmentation overlap charges into spin blocks (Eq.18;
    Q_operator::initialize, non_local_operator.cpp:307-340).

    Returns [Q00, Q11, Q01, Q10]. The primed sums are restricted to the
    same radial function (compare_index_beta_functions,
    non_local_operator.cpp:314-325), like the D rotation."""
    idx = at.beta_lm_index()
    nbf = q_mtrx.shape[0]
    irf = np.array([irf_ for irf_, _, _ in idx])
    m_irf = (irf[:, None] == irf[None, :]).astype(np.float64)
    fm = fcoef * m_irf[:, :, None, None]
    out = [np.zeros((nbf, nbf), dtype=np.complex128) for _ in range(4)]
    for si in range(2):
        for sj in range(2):
            acc = (fm[:, :, sj, 0] @ q_mtrx @ fm[:, :, 0, si]
                   + fm[:, :, sj, 1] @ q_mtrx @ fm[:, :, 1, si])
            # ind = si when diagonal else sj+2 == s_idx[si][sj]
            # (non_local_operator.cpp:330-338; verified against a
            # brute-force replica of the reference loops)
            out[S_IDX[si][sj]] = acc
    return out
