"""Checkpoint / resume.

Reference behavior: src/context/simulation_context.cpp:1153-1191
(create_storage_file) + Periodic_function::hdf5_write — HDF5 file
`sirius.h5` with groups /parameters (num_spins, num_mag_dims, num_bands,
num_gvec, gvec Miller indices), /effective_potential, /density,
/magnetization/{j}.

This stack has no HDF5 library (no h5py, no libhdf5), so the same TREE is
serialized to an .npz archive with '/'-joined keys mirroring the HDF5
layout; load() validates the stored G-vector list and re-maps
coefficients so ordering differences are tolerated (reference:
density.hpp:615-633). An HDF5 writer can replace the backend without
changing the tree.
"""

from __future__ import annotations

import numpy as np
import torch

STORAGE_FILE = "sirius.npz"


def save_state(path: str, dft) -> None:
    """Write density + potential PW coefficients with the reference tree."""
    ctx = dft.ctx
    data = {
        "parameters/num_spins": np.array(ctx.num_spins),
        "parameters/num_mag_dims": np.array(ctx.num_mag_dims),
        "parameters/num_bands": np.array(ctx.num_bands),
        "parameters/num_gvec": np.array(ctx.gvec_fine.num_gvec),
        "parameters/gvec": ctx.gvec_fine.miller,
        "density": dft.density.rho_g.cpu().numpy(),
        "effective_potential": dft.potential.veff_g.cpu().numpy(),
    }
    if ctx.num_mag_dims:
        data["magnetization/0"] = dft.density.mag_g.cpu().numpy()
        if dft.potential.bz_g is not None:
            data["effective_magnetic_field/0"] = dft.potential.bz_g.cpu().numpy()
    if dft.density.density_matrix is not None:
        for lab, t in dft.density.density_matrix.items():
            data[f"density_matrix/{lab}"] = t.cpu().numpy()
    np.savez_compressed(path, **data)


def load_state(path: str, dft) -> None:
    """Load and re-map coefficients onto the current G-vector order."""
    ctx = dft.ctx
    z = np.load(path)
    stored = z["parameters/gvec"]
    cur = ctx.gvec_fine.miller
    if stored.shape == cur.shape and (stored == cur).all():
        remap = None
    else:
        key = {tuple(m): i for i, m in enumerate(stored)}
        remap = np.array([key[tuple(m)] for m in cur], dtype=np.int64)

    def to_dev(arr):
        a = arr if remap is None else arr[remap]
        return torch.from_numpy(a).to(ctx.device)

    dft.density.rho_g = to_dev(z["density"])
    dft.density.rho_r = ctx.fft_fine.to_real(dft.density.rho_g).real
    if "magnetization/0" in z and ctx.num_mag_dims:
        dft.density.mag_g = to_dev(z["magnetization/0"])
        dft.density.mag_r = ctx.fft_fine.to_real(dft.density.mag_g).real
    for lab in list(dft.density.density_matrix or {}):
        k = f"density_matrix/{lab}"
        if k in z:
            dft.density.density_matrix[lab] = torch.from_numpy(z[k]).to(ctx.device)
    # potential is regenerated from the density
    dft.potential.generate(dft.density)
    dft.potential.generate_paw(dft.density)
