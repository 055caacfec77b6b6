"""G-vector sets and FFT-grid mappings.

Reference behavior: src/core/fft/gvec.hpp:124 (Gvec — distributed sphere of
G (or G+k) vectors inside a cutoff), fft3d_grid.hpp (grid dims).

MI355X-first design: on a 288 GB single GPU there is no need for the
reference's z-column (stick) slab distribution inside one node for the
fine grid — each rank holds its k-point-local spheres whole, and the
multi-GPU axis is k-point/band data parallelism over RCCL. G-vector
arrays live as torch tensors on the compute device; construction is
vectorized numpy.
"""

from __future__ import annotations

import math

import numpy as np
import torch


def next_fft_size(n: int) -> int:
    """Smallest {2,3,5}-smooth size >= n.

    Matches the reference's fft::Grid::find_grid_size (fft3d_grid.hpp:34-48)
    so that real-space integration grids — and therefore grid-quadrature
    XC energies — agree with the reference to the last digit. (rocFFT also
    handles 7-smooth sizes, but grid parity wins.)
    """
    while True:
        m = n
        for p in (2, 3, 4, 5):
            while m % p == 0:
                m //= p
        if m == 1:
            return n
        n += 1


def fft_grid_dims(lattice: np.ndarray, cutoff: float) -> tuple[int, int, int]:
    """FFT grid that circumscribes the |G| <= cutoff sphere.

    Mirrors fft::get_min_grid (fft3d_grid.hpp:160) + r3::find_translations
    (r3.hpp:588-603): N_i = int(2·cutoff·|a_i|/2π) + 1 + 2, rounded up to
    a {2,3,5}-smooth size.
    """
    dims = []
    for i in range(3):
        n = int(2 * cutoff * np.linalg.norm(lattice[i]) / (2 * math.pi)) + 3
        dims.append(next_fft_size(n))
    return tuple(dims)


def gvec_sphere(recip: np.ndarray, cutoff: float, k_frac=None,
                dims: tuple[int, int, int] | None = None) -> np.ndarray:
    """Miller indices m with |(m + k)·B| <= cutoff, sorted by |G+k|² (ties by
    (m1,m2,m3) lexicographic for a deterministic order).

    recip: rows b1,b2,b3. Returns int64 [nG,3].
    """
    # bound each miller index: |m_i| <= cutoff * |a_i| / 2pi ; recover a_i from B
    lattice = 2 * math.pi * np.linalg.inv(recip).T
    kf = np.zeros(3) if k_frac is None else np.asarray(k_frac, dtype=np.float64)
    mmax = [int(cutoff * np.linalg.norm(lattice[i]) / (2 * math.pi)) + 1 for i in range(3)]
    r = [np.arange(-m, m + 1) for m in mmax]
    grid = np.stack(np.meshgrid(*r, indexing="ij"), axis=-1).reshape(-1, 3)
    gk = (grid + kf) @ recip
    g2 = np.einsum("ij,ij->i", gk, gk)
    sel = g2 <= cutoff * cutoff + 1e-12
    m = grid[sel]
    g2 = g2[sel]
    order = np.lexsort((m[:, 2], m[:, 1], m[:, 0], np.round(g2, 10)))
    return m[order].astype(np.int64)


class Gvec:
    """A set of G (or G+k) vectors inside a cutoff sphere, with FFT mapping."""

    def __init__(self, recip: np.ndarray, cutoff: float, k_frac=None,
                 dims: tuple[int, int, int] | None = None,
                 device: str | torch.device = "cpu"):
        self.recip = recip
        self.cutoff = cutoff
        self.k_frac = np.zeros(3) if k_frac is None else np.asarray(k_frac, np.float64)
        self.miller = gvec_sphere(recip, cutoff, k_frac)
        self.num_gvec = len(self.miller)
        gk = (self.miller + self.k_frac) @ recip
        self.gkvec_cart = gk                             # numpy [nG,3]
        self.gk_len = np.linalg.norm(gk, axis=1)
        self.g_cart = self.miller @ recip                # G without k
        self.dims = dims
        self.device = torch.device(device)
        # torch mirrors
        self.gkvec_t = torch.from_numpy(gk.copy()).to(self.device)
        self.gk2_t = (self.gkvec_t ** 2).sum(-1)
        self._fft_index = None
        if dims is not None:
            self.set_fft_dims(dims)
        # shells of equal |G| (for form factors / symmetrization)
        lens = np.round(self.gk_len, 9)
        uniq, inv = np.unique(lens, return_inverse=True)
        self.shell_len = uniq
        self.shell_of_g = inv

    def set_fft_dims(self, dims):
        self.dims = tuple(dims)
        n1, n2, n3 = self.dims
        m = self.miller
        i1 = np.mod(m[:, 0], n1)
        i2 = np.mod(m[:, 1], n2)
        i3 = np.mod(m[:, 2], n3)
        idx = (i1 * n2 + i2) * n3 + i3
        assert len(np.unique(idx)) == len(idx), "FFT grid too small for sphere"
        self._fft_index = torch.from_numpy(idx).to(self.device)

    @property
    def fft_index(self) -> torch.Tensor:
        assert self._fft_index is not None
        return self._fft_index

    def index_of_zero(self) -> int:
        """Index of G=0 in this set (k must be 0), or -1."""
        z = np.where((self.miller == 0).all(axis=1))[0]
        return int(z[0]) if len(z) else -1

    def gvec_map_to(self, other: "Gvec") -> torch.Tensor:
        """For each G in self, its index in `other` (must contain all of self).

        Used for the coarse→fine transfer of ρ(G) and fine→coarse map of
        V_eff (reference: gvec_base_mapping, local_operator.cpp:91-118).
        """
        key = {tuple(mm): i for i, mm in enumerate(other.miller)}
        idx = np.array([key[tuple(mm)] for mm in self.miller], dtype=np.int64)
        return torch.from_numpy(idx).to(self.device)

    def to(self, device) -> "Gvec":
        self.device = torch.device(device)
        self.gkvec_t = self.gkvec_t.to(device)
        self.gk2_t = self.gk2_t.to(device)
        if self._fft_index is not None:
            self._fft_index = self._fft_index.to(device)
        return self
