"""Python implementation layer behind the C API (libsirius_amd).

Reference behavior: src/api/sirius_api.cpp — the handler model (opaque
context / k-set / ground-state handlers) and the function semantics.
The C shim (sirius_amd/api/sirius_amd_api.cpp) embeds CPython and calls
the functions in this module with plain Python scalars/lists; handlers
are the Python objects themselves (the shim owns references).

Supports both setup styles the reference offers:
- deck style: import_parameters(json) with unit_cell/atom_files
- programmatic style (what QE uses): set_lattice_vectors + add_atom_type
  + add_atom_type_radial_function(...) pushing pseudopotential data
  through the API (sirius_api.cpp:2100-2170).
"""

from __future__ import annotations

import json
import math

import numpy as np


class CtxHandle:
    def __init__(self):
        self.cfg_data = {}
        self.base_dir = "."
        self.lattice = None
        self.types = {}          # label -> AtomType (programmatic)
        self.atoms = []          # (label, pos, vector_field)
        self.xc = []
        self.ctx = None

    @property
    def initialized(self):
        return self.ctx is not None


class KsetHandle:
    def __init__(self, kset):
        self.kset = kset


class GsHandle:
    def __init__(self, dft):
        self.dft = dft
        self.result = None


def create_context():
    return CtxHandle()


def import_parameters(h: CtxHandle, s: str):
    d = json.loads(s) if s.strip() else {}
    _deep_update(h.cfg_data, d)


def _deep_update(dst, src):
    for k, v in src.items():
        if isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_update(dst[k], v)
        else:
            dst[k] = v


def set_parameter(h: CtxHandle, section: str, key: str, value):
    h.cfg_data.setdefault(section, {})[key] = value


def add_xc_functional(h: CtxHandle, name: str):
    h.xc.append(name)


def set_lattice_vectors(h: CtxHandle, a1, a2, a3):
    h.lattice = np.array([a1, a2, a3], dtype=np.float64)


def add_atom_type(h: CtxHandle, label: str, fname: str = "", zn: int = 0,
                  symbol: str = "", mass: float = 0.0,
                  spin_orbit: bool = False):
    from .cell import AtomType

    if fname:
        h.cfg_data.setdefault("unit_cell", {}).setdefault(
            "atom_files", {})[label] = fname
        h.types[label] = None    # from file at initialize time
        return
    at = AtomType(label)
    at.symbol = symbol or label
    at.zn = int(zn)
    at.spin_orbit = bool(spin_orbit)
    h.types[label] = at


def set_atom_type_radial_grid(h: CtxHandle, label: str, points):
    h.types[label].r = np.asarray(points, dtype=np.float64)


def add_atom_type_radial_function(h: CtxHandle, atom_type: str, label: str,
                                  rf, n: int = -1, l: int = -1,
                                  idxrf1: int = -1, idxrf2: int = -1,
                                  occ: float = 0.0):
    """Push one radial function (sirius_api.cpp:2100-2170 label set)."""
    from .cell import BetaProjector, AtomicWf, QRadialFunction

    at = h.types[atom_type]
    f = np.zeros_like(at.r)
    rf = np.asarray(rf, dtype=np.float64)
    f[:len(rf)] = rf[:len(at.r)]
    if label == "beta":
        at.beta.append(BetaProjector(l=l, j=None, f_r=f))
        at.is_norm_conserving = len(at.q_radial) == 0
    elif label == "ps_atomic_wf":
        at.atomic_wfs.append(AtomicWf(n=n, l=l, occ=occ, f_r=f))
    elif label == "ps_rho_core":
        at.rho_core_r = f
        at.core_correction = True
    elif label == "ps_rho_total":
        at.rho_total_4pir2 = f
    elif label == "vloc":
        at.vloc_r = f
    elif label == "q_aug":
        at.q_radial.append(QRadialFunction(i=idxrf1 - 1, j=idxrf2 - 1,
                                           l=l, f_r=f))
        at.is_ultrasoft = True
        at.is_norm_conserving = False
    elif label == "ae_paw_wf":
        at.paw_ae_wfs.append(f)
        at.is_paw = True
    elif label == "ps_paw_wf":
        at.paw_ps_wfs.append(f)
    elif label == "ae_paw_core":
        at.paw_ae_core = f
    else:
        raise ValueError(f"unknown radial function label: {label}")


def set_atom_type_dion(h: CtxHandle, label: str, dion_flat, nbf: int):
    at = h.types[label]
    at.d_ion = np.asarray(dion_flat, dtype=np.float64).reshape(nbf, nbf)


def add_atom(h: CtxHandle, label: str, position, vector_field=None):
    h.atoms.append((label, np.asarray(position, dtype=np.float64),
                    np.asarray(vector_field or [0, 0, 0], dtype=np.float64)))


def set_atom_position(h: CtxHandle, ia: int, position):
    if h.ctx is not None:
        lab, _ = h.ctx.unit_cell.atoms[ia]
        h.ctx.unit_cell.atoms[ia] = (lab, np.asarray(position, np.float64))
    else:
        lab, _, vf = h.atoms[ia]
        h.atoms[ia] = (lab, np.asarray(position, np.float64), vf)


def initialize_context(h: CtxHandle):
    from .config import Config
    from .context import SimulationContext
    from .cell import UnitCell

    cfg_data = dict(h.cfg_data)
    if h.xc:
        cfg_data.setdefault("parameters", {}).setdefault(
            "xc_functionals", list(h.xc))
    if h.lattice is not None:
        cfg_data.setdefault("unit_cell", {})["lattice_vectors"] = \
            h.lattice.tolist()
        cfg_data["unit_cell"]["lattice_vectors_scale"] = 1.0
    if h.atoms:
        am = {}
        for lab, pos, vf in h.atoms:
            am.setdefault(lab, []).append(list(pos) + list(vf))
        cfg_data.setdefault("unit_cell", {})["atoms"] = am
        cfg_data["unit_cell"]["atom_types"] = list(
            dict.fromkeys(lab for lab, _, _ in h.atoms))
    cfg = Config(cfg_data)
    if any(at is not None for at in h.types.values()):
        lat = h.lattice
        pos = [(lab, p) for lab, p, _ in h.atoms]
        uc = UnitCell(lat, {k: v for k, v in h.types.items()}, pos)
        uc.vector_fields = np.array([vf for _, _, vf in h.atoms])
        h.ctx = SimulationContext(cfg, unit_cell=uc, base_dir=h.base_dir)
    else:
        h.ctx = SimulationContext(cfg, base_dir=h.base_dir)
    return h


def context_initialized(h: CtxHandle) -> bool:
    return h.initialized


def create_kset_from_grid(h: CtxHandle, k_grid, k_shift, use_symmetry: bool):
    from .kpoint import KPointSet

    return KsetHandle(KPointSet(h.ctx))


def create_kset(h: CtxHandle, kpoints, weights, init: bool):
    from .kpoint import KPointSet

    vk = np.asarray(kpoints, dtype=np.float64).reshape(-1, 3)
    w = np.asarray(weights, dtype=np.float64)
    return KsetHandle(KPointSet(h.ctx, vk=vk, weights=w))


def create_ground_state(ks: KsetHandle):
    from .dft import DFTGroundState

    return GsHandle(DFTGroundState(ks.kset))


def find_ground_state(gs: GsHandle, density_tol=None, energy_tol=None,
                      itsol_tol=None, initial_guess=True, max_niter=None,
                      save_state=False):
    if initial_guess:
        gs.dft.initial_state()
    res = gs.dft.find(density_tol=density_tol, energy_tol=energy_tol,
                      itsol_tol=itsol_tol, num_dft_iter=max_niter)
    gs.result = res
    if save_state:
        from .checkpoint import save_state_h5

        save_state_h5("sirius.h5", gs.dft)
    rho_min = float(gs.dft.density.rho_r.min())
    return bool(res["converged"]), int(res["num_scf_iterations"]), rho_min


def get_energy(gs: GsHandle, label: str) -> float:
    dft = gs.dft
    en = gs.result["energy"] if gs.result else dft.total_energy_components()
    kset = dft.kset
    m = {
        "total": lambda: en["total"],
        "evalsum": lambda: en.get("valence_eval_sum",
                                  kset.valence_eval_sum()),
        "exc": lambda: en["exc"],
        "vxc": lambda: en.get("vxc", 0.0),
        "bxc": lambda: en.get("bxc", 0.0),
        "veff": lambda: en.get("veff", 0.0),
        "vha": lambda: en["vha"],
        "ewald": lambda: en.get("ewald", 0.0),
        "demet": lambda: kset.entropy_sum(),
        "fermi": lambda: kset.energy_fermi,
        "band-gap": lambda: kset.band_gap,
        "descf": lambda: en.get("scf_correction", 0.0),
        "paw": lambda: en.get("paw", 0.0),
        "hubbard": lambda: en.get("hubbard", 0.0),
    }
    if label not in m:
        raise ValueError(f"wrong energy label: {label}")
    return float(m[label]())


def get_forces(gs: GsHandle, label: str):
    f = gs.dft.forces()
    if label not in f:
        raise ValueError(f"wrong force label: {label}")
    return np.asarray(f[label], dtype=np.float64).reshape(-1).tolist()


def get_stress_tensor(gs: GsHandle, label: str):
    st = gs.dft.stress()
    if label not in st:
        raise ValueError(f"wrong stress label: {label}")
    return np.asarray(st[label], dtype=np.float64).reshape(-1).tolist()


def get_num_kpoints(ks: KsetHandle) -> int:
    return int(ks.kset.num_kpoints)


def get_band_energies(ks: KsetHandle, ik: int, ispn: int):
    kset = ks.kset
    kset.sync_band()
    return kset._all_eig[ik, ispn, :].tolist()


def get_band_occupancies(ks: KsetHandle, ik: int, ispn: int):
    kset = ks.kset
    kset.sync_band()
    return kset._all_occ[ik, ispn, :].tolist()


def get_kpoint_properties(ks: KsetHandle, ik: int):
    kset = ks.kset
    return float(kset.weights[ik]), kset.vk[ik].tolist()


def save_state(gs: GsHandle, fname: str):
    from .checkpoint import save_state_h5, save_state as save_npz

    if fname.endswith(".npz"):
        save_npz(fname, gs.dft)
    else:
        save_state_h5(fname, gs.dft)


def load_state(gs: GsHandle, fname: str):
    from .checkpoint import load_state_h5, load_state as load_npz

    if fname.endswith(".npz"):
        load_npz(fname, gs.dft)
    else:
        load_state_h5(fname, gs.dft)
