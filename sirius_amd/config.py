"""Input configuration.

Reads the same ``sirius.json`` decks as the reference (the de-facto input
format; reference schema: src/context/input_schema.json). Defaults below
mirror the reference schema defaults for every option we consume; unknown
keys are kept (and queryable) so existing decks load unmodified.
"""

from __future__ import annotations

import copy
import json
from typing import Any

# Defaults mirroring /root/reference/src/context/input_schema.json.
_DEFAULTS: dict[str, dict[str, Any]] = {
    "control": {
        "processing_unit": "auto",          # auto|cpu|gpu
        "verbosity": 1,
        "std_evp_solver_name": "auto",
        "gen_evp_solver_name": "auto",
        "fft_mode": "serial",
        "reduce_gvec": True,
        "print_forces": False,
        "print_stress": False,
        "beta_chunk_size": 256,
        "gvec_chunk_size": 500000,
        "cyclic_block_size": -1,
        "mpi_grid_dims": [1, 1],
        "rmt_max": 2.2,
        "verification": 0,
        # MI355X extension (no reference counterpart): worker threads for
        # the per-k-point Davidson loop, one HIP stream per worker.
        # MEASURED on sto-uspp (64 k, 1 GPU): 1 thr 0.50 s/iter, 2 thr
        # 0.62, 4 thr 1.06 — the loop is GIL/launch-bound, so threads only
        # add contention.  0 = auto = serial; the knob stays for
        # experiments (results are bit-identical, verified on GPU).
        "num_kpoint_threads": 0,
    },
    "parameters": {
        "electronic_structure_method": "pseudopotential",
        "xc_functionals": [],
        "core_relativity": "dirac",
        "valence_relativity": "zora",
        "num_fv_states": -1,
        "num_bands": -1,
        "smearing": "gaussian",
        "smearing_width": 0.01,             # Ha
        "use_symmetry": True,
        "use_ibz": True,
        "num_mag_dims": 0,
        "precision_wf": "fp64",
        "pw_cutoff": 20.0,                  # a.u.^-1, fine grid |G| cutoff
        "gk_cutoff": 6.0,                   # a.u.^-1, |G+k| cutoff
        "aw_cutoff": 0.0,
        "lmax_apw": 8,
        "lmax_rho": 8,
        "lmax_pot": 8,
        "num_dft_iter": 100,
        "energy_tol": 1e-6,
        "density_tol": 1e-6,
        "ngridk": [1, 1, 1],
        "shiftk": [0, 0, 0],
        "vk": [],
        "gamma_point": False,
        "nn_radius": -1,
        "molecule": False,
        "auto_rmt": 1,
        "hubbard_correction": False,
        "so_correction": False,
    },
    "iterative_solver": {
        "type": "davidson",                 # davidson|exact
        "num_steps": 20,
        "subspace_size": 2,
        "locking": True,
        "early_restart": 0.5,
        "energy_tolerance": 1e-2,
        "residual_tolerance": 1e-6,
        "empty_states_tolerance": 0.0,
        "converge_by_energy": 1,
        "min_num_res": 0,
        "init_subspace": "lcao",            # lcao|random
        "extra_ortho": False,
        "min_tolerance": 1e-13,
        "tolerance_ratio": 0.0,
        # MI355X extension: cap the Davidson expansion block (0 = all
        # unconverged residuals).  MEASURED on si512 (1228 bands, 1 GPU):
        # cap 256/384 runs ~3x SLOWER per SCF iteration than uncapped —
        # the pending bands keep every Davidson call at its step limit.
        # Kept as an experiment knob; default stays off.
        "max_block_size": 0,
        "tolerance_scale": [0.1, 0.5],
        "relative_tolerance": 0,
        "init_eval_old": True,
        "num_singular": -1,
        "min_occupancy": 1e-14,
    },
    "mixer": {
        "type": "anderson",                 # linear|anderson|anderson_stable|broyden2
        "beta": 0.7,
        "beta0": 0.15,
        "max_history": 8,
        "beta_scaling_factor": 1.0,
        "use_hartree": False,
        "rms_min": 1e-16,
    },
    "settings": {
        "nprii_vloc": 200,
        "nprii_beta": 20,
        "nprii_aug": 20,
        "nprii_rho_core": 20,
        "itsol_tol_min": 1e-13,
        "itsol_tol_ratio": 0.0,
        "itsol_tol_scale": [0.1, 0.5],
        "min_occupancy": 1e-14,
        "mixer_rms_min": 1e-16,
        "auto_enu_tol": 0,
        "radial_grid": "exponential, 1.0",
        "fft_grid_size": [0, 0, 0],
        "pseudo_grid_cutoff": 10.0,
        "fp32_to_fp64_rms": 0.0,
    },
    "unit_cell": {
        "lattice_vectors": [[1, 0, 0], [0, 1, 0], [0, 0, 1]],
        "lattice_vectors_scale": 1.0,
        "atom_types": [],
        "atom_files": {},
        "atoms": {},
        "atom_coordinate_units": "lattice",
    },
    "hubbard": {},
    "nlcg": {},
    "vcsqnm": {},
}


class Section:
    """Attribute/dict access wrapper over one config section."""

    def __init__(self, data: dict):
        self._data = data

    def __getattr__(self, key):
        try:
            return self._data[key]
        except KeyError as e:
            raise AttributeError(key) from e

    def __getitem__(self, key):
        return self._data[key]

    def __contains__(self, key):
        return key in self._data

    def get(self, key, default=None):
        return self._data.get(key, default)

    def set(self, key, value):
        self._data[key] = value

    def to_dict(self):
        return copy.deepcopy(self._data)


def _deep_merge(base: dict, override: dict) -> dict:
    out = copy.deepcopy(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


class Config:
    """Full input configuration with schema defaults applied."""

    def __init__(self, data: dict | None = None):
        data = data or {}
        merged = _deep_merge(_DEFAULTS, data)
        self._data = merged
        for name in merged:
            if isinstance(merged[name], dict):
                setattr(self, name, Section(merged[name]))

    @classmethod
    def from_json(cls, path: str) -> "Config":
        with open(path) as f:
            return cls(json.load(f))

    @classmethod
    def from_string(cls, s: str) -> "Config":
        return cls(json.loads(s))

    def to_dict(self) -> dict:
        return copy.deepcopy(self._data)

    def override(self, dotted_key: str, value):
        """CLI-style override: section.key=value."""
        sec, key = dotted_key.split(".", 1)
        self._data[sec][key] = value
