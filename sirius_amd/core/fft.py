"""G-sphere <-> real-space grid transforms.

Reference behavior: src/core/fft (SpFFT-backed sphere-limited transforms,
fft.hpp:17) and Smooth_periodic_function::fft_transform
(src/function3d/smooth_periodic_function.hpp:262).

MI355X-native path: batched dense 3D FFTs through torch.fft (rocFFT) with
sphere pack/unpack as indexed scatter/gather. The pack/unpack (and the
fusions around them: kinetic-energy add, V_eff multiply) are the custom
HIP kernel targets (sirius_amd/ops); this module provides the reference
torch implementation used on CPU and as the numerics baseline.

Conventions: f(r) = Σ_G c_G e^{iG·r};  c_G = (1/N) Σ_r f(r) e^{-iG·r}.
"""

from __future__ import annotations

import torch

from .gvec import Gvec


class SphericalFFT:
    """Batched transform between sphere coefficients [.., nG] and the real
    grid [.., n1, n2, n3]."""

    def __init__(self, gvec: Gvec):
        assert gvec.dims is not None
        self.gvec = gvec
        self.dims = gvec.dims
        self.size = gvec.dims[0] * gvec.dims[1] * gvec.dims[2]

    def to_real(self, coeffs: torch.Tensor) -> torch.Tensor:
        """[.., nG] complex -> [.., n1, n2, n3] complex."""
        batch = coeffs.shape[:-1]
        grid = torch.zeros(*batch, self.size, dtype=coeffs.dtype, device=coeffs.device)
        grid[..., self.gvec.fft_index] = coeffs
        grid = grid.reshape(*batch, *self.dims)
        return torch.fft.ifftn(grid, dim=(-3, -2, -1), norm="forward")

    def to_pw(self, fr: torch.Tensor) -> torch.Tensor:
        """[.., n1, n2, n3] complex -> [.., nG] complex."""
        g = torch.fft.fftn(fr, dim=(-3, -2, -1), norm="forward")
        return g.reshape(*fr.shape[:-3], self.size)[..., self.gvec.fft_index]
