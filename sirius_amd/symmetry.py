"""Crystal symmetry: space-group detection, IBZ reduction, symmetrization.

Reference behavior: src/symmetry/ (Crystal_symmetry via spglib,
crystal_symmetry.cpp:209; get_irreducible_reciprocal_mesh.hpp:22;
symmetrize_pw_function.hpp; symmetrize_density_matrix.hpp). spglib is not
in this stack, so detection is implemented directly:

 - lattice point group: integer matrices W with W M W^T = M,
   M_ij = a_i·a_j (metric), entries enumerated over [-2,2];
 - space group: candidate fractional translations from atom coincidence,
   checked against the full atom permutation;
 - conventions: atoms x → x·W + t (rows, fractional), reciprocal
   m → m·W^T (so ρ̂(m·W^T) = e^{2πi m·t} ρ̂(m)).

Rotation matrices of real spherical harmonics (for the USPP density
matrix and Hubbard occupation symmetrization) are obtained numerically by
least squares on a direction set — no Wigner bookkeeping to get wrong.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np

from .core import ylm as ylm_mod


@dataclass
class SymOp:
    W: np.ndarray          # int [3,3], acts on fractional atom rows: x' = x W + t
    t: np.ndarray          # fractional translation
    perm: np.ndarray       # atom permutation: op maps atom a -> perm[a]
    S: np.ndarray          # cartesian rotation


class CrystalSymmetry:
    def __init__(self, cell, tol: float = 1e-6):
        self.cell = cell
        self.tol = tol
        self.ops = find_space_group(cell, tol)

    @property
    def rotations(self):
        return np.array([op.W for op in self.ops])

    @property
    def num_ops(self):
        return len(self.ops)


def _lattice_point_group(A: np.ndarray, tol: float) -> list[np.ndarray]:
    """Integer W (|det|=1) with W M W^T = M, M = A A^T."""
    M = A @ A.T
    rng = range(-2, 3)
    cands = [[], [], []]
    vecs = np.array([(i, j, k) for i in rng for j in rng for k in rng])
    vMv = np.einsum("ni,ij,nj->n", vecs, M, vecs)
    for i in range(3):
        sel = np.abs(vMv - M[i, i]) < tol * max(1.0, abs(M[i, i]))
        cands[i] = vecs[sel]
    out = []
    for w0 in cands[0]:
        m01 = w0 @ M
        for w1 in cands[1]:
            if abs(m01 @ w1 - M[0, 1]) > tol * max(1.0, abs(M[0, 0])):
                continue
            m02 = w0 @ M
            m12 = w1 @ M
            for w2 in cands[2]:
                if abs(m02 @ w2 - M[0, 2]) > tol * max(1.0, abs(M[0, 0])):
                    continue
                if abs(m12 @ w2 - M[1, 2]) > tol * max(1.0, abs(M[0, 0])):
                    continue
                W = np.array([w0, w1, w2])
                if abs(abs(round(np.linalg.det(W))) - 1) < 1e-9:
                    out.append(W)
    return out


def find_space_group(cell, tol: float = 1e-6) -> list[SymOp]:
    A = cell.lattice
    pos = cell.atom_positions_frac() % 1.0
    labels = [lab for lab, _ in cell.atoms]
    # collinear magnetic structure: distinguish sublattices by the sign
    # of the initial moment so the detected group is the magnetic
    # subgroup (reference uses spin rotations of the full group; the
    # colored subgroup is physically equivalent for symmetrization/IBZ)
    vf = getattr(cell, "vector_fields", None)
    if vf is not None and np.abs(vf).max() > 1e-8:
        def _color(m):
            if m > 1e-8:
                return "+"
            if m < -1e-8:
                return "-"
            return "0"
        labels = [f"{lab}{_color(vf[i][2])}" for i, (lab, _)
                  in enumerate(cell.atoms)]
    na = len(labels)
    Ws = _lattice_point_group(A, tol)

    # least-frequent type anchors candidate translations
    from collections import Counter

    cnt = Counter(labels)
    lab0 = min(cnt, key=lambda k: cnt[k])
    anchors = [i for i, l in enumerate(labels) if l == lab0]
    a0 = anchors[0]

    ops = []
    for W in Ws:
        x0 = pos[a0] @ W
        for b in anchors:
            t = (pos[b] - x0) % 1.0
            # check permutation
            newpos = (pos @ W + t) % 1.0
            perm = -np.ones(na, dtype=np.int64)
            ok = True
            for i in range(na):
                d = np.abs(newpos[i] - pos)
                d = np.minimum(d, 1.0 - d)
                dd = (d**2).sum(axis=1)
                j = int(np.argmin(dd))
                if dd[j] > tol or labels[j] != labels[i] or perm.tolist().count(j) > 0:
                    ok = False
                    break
                perm[i] = j
            if ok:
                S = np.linalg.inv(A) @ W.T @ A  # careful: see derivation below
                # derivation: a_i S^T = W_ij a_j => A S^T = W A => S = (A^{-1} W A)^T
                S = (np.linalg.inv(A) @ W @ A).T
                ops.append(SymOp(W=W, t=t, perm=perm, S=S))
                break  # one translation per W (primitive translations give dup perms)
    return ops


def ibz_mesh(cell, ngridk, shiftk, ops: list[SymOp]):
    """Irreducible k mesh. Reciprocal row vectors transform as m → m·W^T;
    + time reversal k → −k. Returns (kpoints [nk,3], weights)."""
    n = np.asarray(ngridk, dtype=np.int64)
    s = np.asarray(shiftk, dtype=np.float64)
    pts = []
    for i0 in range(n[0]):
        for i1 in range(n[1]):
            for i2 in range(n[2]):
                k = (np.array([i0, i1, i2]) + s / 2.0) / n
                pts.append(k - np.round(k))
    pts = np.array(pts)

    def key(k):
        ik = np.round((k % 1.0) * n * 2).astype(int)
        return tuple(ik % (2 * n))

    index = {key(k): i for i, k in enumerate(pts)}
    assigned = -np.ones(len(pts), dtype=np.int64)
    irr, weights = [], []
    for i, k in enumerate(pts):
        if assigned[i] >= 0:
            continue
        orbit = set()
        for op in ops:
            for sgn in (1.0, -1.0):
                kk = sgn * (k @ op.W.T)
                j = index.get(key(kk))
                if j is not None:
                    orbit.add(j)
        orbit.add(i)
        for j in orbit:
            assigned[j] = len(irr)
        irr.append(k)
        weights.append(len(orbit) / len(pts))
    return np.array(irr), np.array(weights)


class RhoSymmetrizer:
    """Precomputed gather indices + phases for PW symmetrization:
    ρ_sym(m0) = (1/N) Σ_op e^{2πi m_op·t} ρ̂(m_op), m_op = m0·W^{-T}
    (reference: symmetrize_pw_function.hpp via Gvec_shells). One batched
    gather per call — GPU-friendly, no per-op host work."""

    def __init__(self, gvec, ops: list[SymOp], device):
        import torch

        m = gvec.miller
        n1, n2, n3 = gvec.dims
        size = n1 * n2 * n3
        inv = np.full(size, -1, dtype=np.int64)
        lin0 = (np.mod(m[:, 0], n1) * n2 + np.mod(m[:, 1], n2)) * n3 + np.mod(m[:, 2], n3)
        inv[lin0] = np.arange(len(m))
        idx_all, ph_all = [], []
        for op in ops:
            Winvt = np.round(np.linalg.inv(op.W)).astype(np.int64).T
            m_op = m @ Winvt
            lin = (np.mod(m_op[:, 0], n1) * n2 + np.mod(m_op[:, 1], n2)) * n3 \
                + np.mod(m_op[:, 2], n3)
            idx = inv[lin]
            assert (idx >= 0).all(), "rotated G left the sphere"
            idx_all.append(idx)
            ph_all.append(np.exp(2j * math.pi * (m_op @ op.t)))
        self.idx = torch.from_numpy(np.stack(idx_all)).to(device)      # [nops, nG]
        self.ph = torch.from_numpy(np.stack(ph_all)).to(device)        # [nops, nG]
        self.nops = len(ops)

    def __call__(self, rho_g):
        return (self.ph * rho_g[self.idx]).sum(0) / self.nops


def symmetrize_rho_g(rho_g, gvec, ops: list[SymOp], cache={}):
    key = (id(gvec), id(ops), str(rho_g.device))
    if key not in cache:
        cache[key] = RhoSymmetrizer(gvec, ops, rho_g.device)
    return cache[key](rho_g)


def rlm_rotation_matrices(lmax: int, S: np.ndarray) -> list[np.ndarray]:
    """Per-l rotation matrices D^l with R_lm(S r̂) = Σ_m' D^l_{m m'} R_lm'(r̂),
    computed by least squares over a direction set (reference:
    src/core/sht rotation matrices)."""
    rng = np.random.default_rng(7)
    npts = max(64, 4 * ylm_mod.lmmax(lmax))
    u = rng.normal(size=(npts, 3))
    u /= np.linalg.norm(u, axis=1, keepdims=True)
    us = u @ S.T          # S r̂ (column action on each row vector)
    _, th1, ph1 = ylm_mod.spherical_coords(u)
    _, th2, ph2 = ylm_mod.spherical_coords(us)
    R1 = ylm_mod.rlm(lmax, th1, ph1)      # R(r̂)
    R2 = ylm_mod.rlm(lmax, th2, ph2)      # R(S r̂)
    out = []
    for l in range(lmax + 1):
        sl = slice(l * l, (l + 1) * (l + 1))
        # R2[:, sl] = R1[:, sl] @ D^T  => solve lsq
        D_T, *_ = np.linalg.lstsq(R1[:, sl], R2[:, sl], rcond=None)
        out.append(D_T.T)
    return out


def symmetrize_density_matrix(dm: dict, ctx, ops: list[SymOp]):
    """Average dm over the space group (reference:
    symmetrize_density_matrix.hpp). dm: {label: [na, nbf, nbf, nspin]}.

    Under op (atom a → perm[a], rotation S): the β-projector coefficients
    of atom perm[a] in the rotated frame relate by D^l; the symmetrized
    dm is (1/N) Σ_op T(op)† dm[perm[a]] T(op) blockwise in l.
    """
    import torch

    uc = ctx.unit_cell
    out = {lab: torch.zeros_like(d) for lab, d in dm.items()}
    nops = len(ops)
    key = ("dm_sym", id(ops))
    cache = getattr(ctx, "_dm_sym_cache", None)
    if cache is None or cache[0] != key:
        # precompute per (op, label): beta rotation matrix T and atom index map
        pre = {}
        for lab, d in dm.items():
            at = uc.atom_types[lab]
            if at.num_beta == 0:
                continue
            lmax = max(b.l for b in at.beta)
            idxb = at.beta_lm_index()
            nbf = len(idxb)
            ia_list = list(uc.atoms_of_type(lab))
            ia_pos = {ia: i for i, ia in enumerate(ia_list)}
            entries = []
            for op in ops:
                Dl = rlm_rotation_matrices(lmax, op.S)
                T = np.zeros((nbf, nbf))
                i = 0
                while i < nbf:
                    irf, l, m = idxb[i]
                    T[i:i + 2 * l + 1, i:i + 2 * l + 1] = Dl[l]
                    i += 2 * l + 1
                Tt = torch.from_numpy(T).to(d.device).to(d.dtype)
                src = torch.tensor([ia_pos[int(op.perm[ia])] for ia in ia_list],
                                   device=d.device)
                entries.append((Tt, src))
            pre[lab] = entries
        ctx._dm_sym_cache = (key, pre)
        cache = ctx._dm_sym_cache
    pre = cache[1]
    for lab, d in dm.items():
        if lab not in pre:
            continue
        for Tt, src in pre[lab]:
            dp = d[src]                                   # [na, nbf, nbf, nspin]
            out[lab] += torch.einsum("pq,aqrs,rt->apts", Tt.conj().T, dp, Tt)
    for lab in out:
        out[lab] /= nops
    return out
