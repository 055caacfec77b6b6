"""K-points and k-point sets.

Reference behavior: src/k_point/k_point.hpp (K_point: per-k G+k Gvec, own
wavefunction FFT, eigenvalues/occupancies, spinor wavefunctions, beta
projectors), k_point_set.hpp:29 (K_point_set: IBZ mesh, chunk split over
comm_k, find_band_occupancies k_point_set.cpp:286, sync_band :18-44).

MI355X design: k-points are chunk-split over the world process group
(one process per GPU, RCCL). Wavefunctions are [nspin][nbands, nGk]
complex128 torch tensors resident on the device.
"""

from __future__ import annotations

import numpy as np
import torch

from .core.gvec import Gvec
from .core.fft import SphericalFFT
from . import smearing as sm
from .parallel import get_comm


def kmesh_full(ngridk, shiftk) -> tuple[np.ndarray, np.ndarray]:
    """Full Monkhorst-Pack mesh: k = (i + shift/2)/n, reduced to (-1/2, 1/2]."""
    n = np.asarray(ngridk, dtype=np.int64)
    s = np.asarray(shiftk, dtype=np.float64)
    pts = []
    for i0 in range(n[0]):
        for i1 in range(n[1]):
            for i2 in range(n[2]):
                k = (np.array([i0, i1, i2]) + s / 2.0) / n
                k = k - np.round(k)  # into (-1/2, 1/2]
                pts.append(k)
    pts = np.array(pts)
    w = np.full(len(pts), 1.0 / len(pts))
    return pts, w


def kmesh_ibz(cell, ngridk, shiftk, sym_ops) -> tuple[np.ndarray, np.ndarray]:
    """Irreducible mesh under the given reciprocal rotations (+ time reversal).

    sym_ops: [nsym, 3, 3] integer rotation matrices in lattice coordinates
    (reference: get_irreducible_reciprocal_mesh via spglib,
    src/symmetry/get_irreducible_reciprocal_mesh.hpp:22).
    """
    pts, w = kmesh_full(ngridk, shiftk)
    n = np.asarray(ngridk)

    def key(k):
        ik = np.round((k + 0.5) * n * 2).astype(int)  # robust rational key
        return tuple(ik % (2 * n))

    index = {key(k): i for i, k in enumerate(pts)}
    mapped = -np.ones(len(pts), dtype=np.int64)
    irr = []
    weights = []
    for i, k in enumerate(pts):
        if mapped[i] >= 0:
            continue
        orbit = set()
        for R in sym_ops:
            for sgn in (1.0, -1.0):  # time reversal
                kk = sgn * (R.T @ k)
                kk = kk - np.round(kk)
                j = index.get(key(kk))
                if j is not None and mapped[j] < 0:
                    orbit.add(j)
        for j in orbit:
            mapped[j] = len(irr)
        irr.append(k)
        weights.append(len(orbit) / len(pts))
    return np.array(irr), np.array(weights)


class KPoint:
    def __init__(self, ctx, k_frac: np.ndarray, weight: float):
        self.ctx = ctx
        self.k_frac = np.asarray(k_frac, dtype=np.float64)
        self.weight = float(weight)
        uc = ctx.unit_cell
        self.gkvec = Gvec(uc.recip, ctx.gk_cutoff, k_frac=self.k_frac,
                          dims=ctx.coarse_dims, device=ctx.device)
        self.fft = SphericalFFT(self.gkvec)
        self.num_gkvec = self.gkvec.num_gvec
        nb = ctx.num_bands
        nss = ctx.num_spin_steps
        wf_len = self.num_gkvec * ctx.num_spinors   # spinor components stacked
        self.psi = torch.zeros(nss, nb, wf_len, dtype=ctx.dtype,
                               device=ctx.device)
        self.eigvals = np.zeros((nss, nb))
        self.occ = np.zeros((nss, nb))
        self.beta = None  # BetaProjectors, built lazily by Hamiltonian


class KPointSet:
    def __init__(self, ctx, vk: np.ndarray | None = None,
                 weights: np.ndarray | None = None):
        self.ctx = ctx
        self.comm = get_comm()
        p = ctx.cfg.parameters
        if vk is None:
            if p.use_symmetry and p.use_ibz and getattr(ctx, "symmetry", None):
                from .symmetry import ibz_mesh

                vk, weights = ibz_mesh(ctx.unit_cell, p.ngridk, p.shiftk,
                                       ctx.symmetry.ops)
            else:
                vk, weights = kmesh_full(p.ngridk, p.shiftk)
        self.vk = np.atleast_2d(vk)
        self.weights = np.asarray(weights if weights is not None
                                  else np.full(len(self.vk), 1.0 / len(self.vk)))
        self.num_kpoints = len(self.vk)
        # chunk split over k-GROUPS (reference splindex_chunk,
        # k_point_set.hpp:44); with band parallelism every rank of a
        # band group holds the same k-points (ctx.kcolor/num_kgroups)
        ngrp = getattr(ctx, "num_kgroups", self.comm.size if self.comm.active else 1)
        color = getattr(ctx, "kcolor", self.comm.rank if self.comm.active else 0)
        counts = [self.num_kpoints // ngrp +
                  (1 if r < self.num_kpoints % ngrp else 0)
                  for r in range(ngrp)]
        offs = np.cumsum([0] + counts)
        self.local_range = (int(offs[color]), int(offs[color + 1]))
        self.kpoints = [KPoint(ctx, self.vk[ik], self.weights[ik])
                        for ik in range(*self.local_range)]
        self.energy_fermi = 0.0
        self.band_gap = 0.0

    def __iter__(self):
        return iter(self.kpoints)

    def sync_band(self):
        """Allgather eigenvalues/occupancies (k_point_set.cpp:18-44)."""
        if not self.comm.active:
            self._all_eig = np.array([kp.eigvals for kp in self.kpoints])
            self._all_occ = np.array([kp.occ for kp in self.kpoints])
            self._all_w = self.weights.copy()
            return
        # gather (k_index, eig, occ); with band groups the same k is
        # held by several ranks — dedup by global k index
        local = [(self.local_range[0] + i, kp.eigvals, kp.occ)
                 for i, kp in enumerate(self.kpoints)]
        gathered = self.comm.allgather_object(local)
        by_k = {}
        for part in gathered:
            for ik, e, o in part:
                by_k[ik] = (e, o)
        eig = [by_k[ik][0] for ik in range(self.num_kpoints)]
        occ = [by_k[ik][1] for ik in range(self.num_kpoints)]
        self._all_eig = np.array(eig)
        self._all_occ = np.array(occ)
        self._all_w = self.weights.copy()

    def find_band_occupancies(self):
        """Fermi bisection + smearing (k_point_set.cpp:286)."""
        ctx = self.ctx
        p = ctx.cfg.parameters
        self.sync_band()
        ne = getattr(ctx, 'num_valence_electrons', None) or ctx.unit_cell.num_electrons
        eig = self._all_eig  # [nk, num_spin_steps, nb]
        nk, nspin, nb = eig.shape
        eigf = eig.reshape(nk * nspin, nb)
        wf = np.repeat(self._all_w, nspin)
        mu = sm.find_fermi(eigf, wf, ne, p.smearing, p.smearing_width,
                           ctx.max_occupancy)
        self.energy_fermi = mu
        occ_all = sm.occupancy(p.smearing, mu - eig, p.smearing_width) * ctx.max_occupancy
        # scatter back to local k-points
        for i, kp in enumerate(self.kpoints):
            kp.occ = occ_all[self.local_range[0] + i]
        self._all_occ = occ_all
        # band gap via per-band (min, max) ranges over the k set, sorted,
        # at integer filling (K_point_set::find_band_occupancies,
        # k_point_set.cpp:416-449)
        self.band_gap = 0.0
        nve = int(ne + 1e-12)
        if ctx.num_spins == 2 or (abs(nve - ne) < 1e-12 and nve % 2 == 0):
            emin = self._all_eig.min(axis=0).reshape(-1)   # per (spin, band)
            emax = self._all_eig.max(axis=0).reshape(-1)
            order = np.lexsort((emax, emin))
            emin, emax = emin[order], emax[order]
            ist = nve if ctx.num_spins == 2 else nve // 2
            if ist < len(emin) and emin[ist] > emax[ist - 1]:
                self.band_gap = float(emin[ist] - emax[ist - 1])

    def valence_eval_sum(self) -> float:
        """Σ_k w_k Σ_n f_nk ε_nk (over ALL k; every rank returns the total)."""
        s = (self._all_w[:, None, None] * self._all_occ * self._all_eig).sum()
        return float(s)

    def entropy_sum(self) -> float:
        p = self.ctx.cfg.parameters
        ent = sm.entropy(p.smearing, self.energy_fermi - self._all_eig, p.smearing_width)
        return float((self._all_w[:, None, None] * ent).sum() * self.ctx.max_occupancy)
