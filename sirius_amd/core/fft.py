"""G-sphere <-> real-space grid transforms.

Reference behavior: src/core/fft (SpFFT-backed sphere-limited transforms,
fft.hpp:17) and Smooth_periodic_function::fft_transform
(src/function3d/smooth_periodic_function.hpp:262).

MI355X-native path: batched dense 3D FFTs through torch.fft (rocFFT) with
sphere pack/unpack as hand-written HIP kernels (sirius_amd/ops/src/
scf_ops.hip) — including the fused local-operator pipeline
pack → IFFT → ×V_eff(r) → FFT → unpack+kinetic, which is the Davidson
hot loop (reference: Local_operator::apply_h, local_operator.cpp:275,
GPU twins mul_by_veff_* / add_to_hphi_pw in local_operator.cu).

On a GPU the HIP kernels are mandatory (loud failure if the extension is
missing); the CPU path uses the equivalent torch indexing ops.

Conventions: f(r) = Σ_G c_G e^{iG·r};  c_G = (1/N) Σ_r f(r) e^{-iG·r}.
"""

from __future__ import annotations

import torch

from .gvec import Gvec


def _ext_for(t: torch.Tensor):
    """Native extension when running on GPU (required there), else None.
    The HIP kernels are fp64 (double2); complex64 tensors (fp32
    wave-function mode) take the torch path (cgemm/cFFT, still fully on
    device)."""
    if t.is_cuda and t.dtype in (torch.complex128, torch.float64):
        from .. import ops

        return ops.get_ext(required=True)
    return None


class SphericalFFT:
    """Batched transform between sphere coefficients [.., nG] and the real
    grid [.., n1, n2, n3]."""

    def __init__(self, gvec: Gvec):
        assert gvec.dims is not None
        self.gvec = gvec
        self.dims = gvec.dims
        self.size = gvec.dims[0] * gvec.dims[1] * gvec.dims[2]
        self._pack_buf = None  # persistent zeroed staging buffer

    # -- pack/unpack -------------------------------------------------------

    def _pack(self, coeffs: torch.Tensor) -> torch.Tensor:
        """[.., nG] -> dense flat [.., size] (zero outside sphere).

        The staging buffer is PERSISTENT: pack writes the same fixed sphere
        locations every call and the FFT reads it out-of-place, so entries
        outside the sphere stay zero for the lifetime of the buffer — no
        per-call zero-fill (was 4.7% of GPU time as FillFunctor kernels,
        profiles/r01..._v3)."""
        batch = coeffs.shape[:-1]
        nb = int(coeffs.numel() // coeffs.shape[-1]) if coeffs.shape[-1] else 0
        buf = self._pack_buf
        if (buf is None or buf.shape[0] < nb or buf.device != coeffs.device
                or buf.dtype != coeffs.dtype):
            if buf is not None and buf.is_cuda:
                # threaded k-loop: the retiring buffer may belong to a
                # different stream's allocator pool — keep it alive until
                # this stream's enqueued reads complete
                buf.record_stream(torch.cuda.current_stream())
            buf = torch.zeros(max(nb, 1), self.size, dtype=coeffs.dtype,
                              device=coeffs.device)
            self._pack_buf = buf
        grid = buf[:nb].reshape(*batch, self.size)
        ext = _ext_for(coeffs)
        if ext is not None:
            ext.pack_sphere(coeffs.contiguous(), self.gvec.fft_index, grid)
        else:
            grid[..., self.gvec.fft_index] = coeffs
        return grid

    def _unpack(self, grid_flat: torch.Tensor) -> torch.Tensor:
        ext = _ext_for(grid_flat)
        if ext is not None:
            out = torch.empty(*grid_flat.shape[:-1], self.gvec.num_gvec,
                              dtype=grid_flat.dtype, device=grid_flat.device)
            ext.unpack_sphere(grid_flat.contiguous(), self.gvec.fft_index, out)
            return out
        return grid_flat[..., self.gvec.fft_index]

    # -- public ------------------------------------------------------------

    def to_real(self, coeffs: torch.Tensor) -> torch.Tensor:
        """[.., nG] complex -> [.., n1, n2, n3] complex."""
        grid = self._pack(coeffs).reshape(*coeffs.shape[:-1], *self.dims)
        return torch.fft.ifftn(grid, dim=(-3, -2, -1), norm="forward")

    def to_pw(self, fr: torch.Tensor) -> torch.Tensor:
        """[.., n1, n2, n3] complex -> [.., nG] complex.

        fftn runs UNNORMALIZED; the 1/N lands in the sphere-sized unpack
        kernel instead of a full-grid scale pass."""
        ext = _ext_for(fr)
        if ext is not None:
            g = torch.fft.fftn(fr, dim=(-3, -2, -1), norm="backward")
            flat = g.reshape(*fr.shape[:-3], self.size)
            out = torch.empty(*flat.shape[:-1], self.gvec.num_gvec,
                              dtype=flat.dtype, device=flat.device)
            ext.unpack_sphere(flat.contiguous(), self.gvec.fft_index, out,
                              1.0 / self.size)
            return out
        g = torch.fft.fftn(fr, dim=(-3, -2, -1), norm="forward")
        return self._unpack(g.reshape(*fr.shape[:-3], self.size))

    # keep the dense FFT working set bounded (large Γ-only cells: a
    # 1000-band block on a 144³ grid would be ~50 GB of grids at once)
    MAX_GRID_BYTES = 4 << 30

    def _band_chunk(self, nb: int, itemsize: int = 16) -> int:
        per_band = self.size * itemsize
        return max(1, min(nb, int(self.MAX_GRID_BYTES // max(per_band, 1))))

    def apply_veff_kinetic(self, psi: torch.Tensor, veff_r: torch.Tensor,
                           gk2: torch.Tensor) -> torch.Tensor:
        """Fused local-operator apply: FFT⁻¹ψ → ×V(r) → FFT → +½|G+k|²ψ.

        psi [nb, nG], veff_r [n1,n2,n3] real, gk2 [nG] real.
        Returns hpsi_local [nb, nG].  Band-chunked to bound grid memory.
        """
        nb = psi.shape[0]
        chunk = self._band_chunk(nb)
        if chunk < nb:
            out = torch.empty_like(psi)
            for i in range(0, nb, chunk):
                out[i:i + chunk] = self.apply_veff_kinetic(
                    psi[i:i + chunk].contiguous(), veff_r, gk2)
            return out
        ext = _ext_for(psi)
        grid = self._pack(psi).reshape(nb, *self.dims)
        psi_r = torch.fft.ifftn(grid, dim=(-3, -2, -1), norm="forward")
        if ext is not None:
            ext.mul_veff(psi_r, veff_r.contiguous())
            g = torch.fft.fftn(psi_r, dim=(-3, -2, -1), norm="backward")
            g = g.reshape(nb, self.size)
            out = torch.empty_like(psi)
            ext.unpack_add_kinetic(g.contiguous(), self.gvec.fft_index,
                                   gk2.contiguous(), psi.contiguous(), out,
                                   1.0 / self.size)
            return out
        vpsi = psi_r * veff_r
        g = torch.fft.fftn(vpsi, dim=(-3, -2, -1), norm="forward")
        g = g.reshape(nb, self.size)
        return g[..., self.gvec.fft_index] + gk2 * psi

    def density_accumulate(self, psi: torch.Tensor, weights: torch.Tensor,
                           rho_r: torch.Tensor):
        """rho_r += Σ_b w_b |FFT⁻¹ψ_b(r)|² (in place; rho_r real [n1,n2,n3])."""
        nb = psi.shape[0]
        chunk = self._band_chunk(nb)
        if chunk < nb:
            for i in range(0, nb, chunk):
                self.density_accumulate(psi[i:i + chunk].contiguous(),
                                        weights[i:i + chunk], rho_r)
            return
        psi_r = self.to_real(psi)
        ext = _ext_for(psi)
        if ext is not None:
            ext.density_acc(psi_r.contiguous(), weights.contiguous(),
                            rho_r.reshape(-1))
        else:
            rho_r += torch.einsum("b,bxyz->xyz", weights,
                                  psi_r.real**2 + psi_r.imag**2)
