"""Checkpoint / resume.

Reference behavior: src/context/simulation_context.cpp:1153-1191
(create_storage_file) + Periodic_function::hdf5_write — HDF5 file
`sirius.h5` with groups /parameters (num_spins, num_mag_dims, num_bands,
num_gvec, gvec Miller indices), /effective_potential, /density,
/magnetization/{j}.

Two backends write the same tree:
- `save_state_h5`/`load_state_h5`: a real `sirius.h5` in the HDF5
  classic binary format via sirius_amd.utils.hdf5 (this stack has no
  libhdf5/h5py, so the format is emitted directly — files are readable
  by h5py elsewhere).  This is the reference-parity checkpoint.
- `save_state`/`load_state`: the same tree in an .npz archive with
  '/'-joined keys (compact fallback).
load() validates the stored G-vector list and re-maps coefficients so
ordering differences are tolerated (reference: density.hpp:615-633).
"""

from __future__ import annotations

import numpy as np
import torch

STORAGE_FILE = "sirius.h5"


def save_state_h5(path: str, dft) -> None:
    """Write the reference `sirius.h5` tree (create_storage_file +
    Periodic_function::hdf5_write layout) as a real HDF5 file."""
    from .utils.hdf5 import H5Writer

    ctx = dft.ctx
    w = H5Writer()
    w.create_group("parameters")
    w.write("parameters", "num_spins", ctx.num_spins)
    w.write("parameters", "num_mag_dims", ctx.num_mag_dims)
    w.write("parameters", "num_bands", ctx.num_bands)
    w.write("parameters", "num_gvec", ctx.gvec_fine.num_gvec)
    w.write("parameters", "gvec",
            np.ascontiguousarray(ctx.gvec_fine.miller.T))   # [3, num_gvec]
    w.create_group("effective_potential")
    w.write("effective_potential", "f_pw", dft.potential.veff_g.cpu().numpy())
    w.create_group("density")
    w.write("density", "f_pw", dft.density.rho_g.cpu().numpy())
    w.create_group("magnetization")
    w.create_group("effective_magnetic_field")
    if ctx.num_mag_dims:
        w.create_group("magnetization/0")
        w.write("magnetization/0", "f_pw", dft.density.mag_g.cpu().numpy())
        if dft.potential.bz_g is not None:
            w.create_group("effective_magnetic_field/0")
            w.write("effective_magnetic_field/0", "f_pw",
                    dft.potential.bz_g.cpu().numpy())
    w.create_group("unit_cell/atoms")
    for ia, (lab, _) in enumerate(ctx.unit_cell.atoms):
        w.create_group(f"unit_cell/atoms/{ia}")
        w.write(f"unit_cell/atoms/{ia}", "mt_basis_size", 0)
    w.save(path)


def load_state_h5(path: str, dft) -> None:
    from .utils.hdf5 import read

    ctx = dft.ctx
    d = read(path)
    stored = np.ascontiguousarray(d["parameters"]["gvec"]).T  # [num_gvec, 3]
    cur = ctx.gvec_fine.miller
    if stored.shape == cur.shape and (stored == cur).all():
        remap = None
    else:
        key = {tuple(m): i for i, m in enumerate(stored)}
        remap = np.array([key[tuple(m)] for m in cur], dtype=np.int64)

    def to_dev(arr):
        a = arr.view(np.complex128)
        a = a if remap is None else a[remap]
        return torch.from_numpy(a.copy()).to(ctx.device)

    dft.density.rho_g = to_dev(d["density"]["f_pw"])
    dft.density.rho_r = ctx.fft_fine.to_real(dft.density.rho_g).real
    if ctx.num_mag_dims and "0" in d.get("magnetization", {}):
        dft.density.mag_g = to_dev(d["magnetization"]["0"]["f_pw"])
        dft.density.mag_r = ctx.fft_fine.to_real(dft.density.mag_g).real
    dft.potential.generate(dft.density)
    dft.potential.generate_paw(dft.density)


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
