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


# ---- introspection / array export (reference sirius_api.cpp; round-2
# ---- widening of the QE-coupling surface) ------------------------------

VERSION = (2, 0, 0)   # this framework's own series (not the reference's)


def is_initialized() -> bool:
    return True


def get_version(which: str) -> int:
    return {"major": VERSION[0], "minor": VERSION[1],
            "revision": VERSION[2]}[which]


def get_num_atoms(gs: GsHandle) -> int:
    return int(gs.dft.ctx.unit_cell.num_atoms)


def get_num_gvec(h: CtxHandle) -> int:
    return int(h.ctx.gvec_fine.num_gvec)


def get_num_fft_grid_points(h: CtxHandle) -> int:
    import math as _m

    return int(_m.prod(h.ctx.fft_fine.dims))


def get_fft_index(h: CtxHandle):
    """1-based offsets of every fine-sphere G in the (Fortran-ordered)
    FFT grid (reference sirius_get_fft_index)."""
    g = h.ctx.gvec_fine
    n1, n2, n3 = g.dims
    m = g.miller
    i1 = np.mod(m[:, 0], n1)
    i2 = np.mod(m[:, 1], n2)
    i3 = np.mod(m[:, 2], n3)
    return (i1 + n1 * (i2 + n2 * i3) + 1).tolist()


def get_num_beta_projectors(h: CtxHandle, label: str) -> int:
    return int(h.ctx.unit_cell.atom_types[label].num_beta)


def get_gvec_arrays(h: CtxHandle):
    """(miller [3N], cart [3N], len [N]) of the fine G sphere."""
    g = h.ctx.gvec_fine
    return (g.miller.reshape(-1).tolist(),
            np.asarray(g.g_cart).reshape(-1).tolist(),
            np.linalg.norm(np.asarray(g.g_cart), axis=1).tolist())


def get_max_num_gkvec(ks: KsetHandle) -> int:
    return max(int(kp.num_gkvec) for kp in ks.kset.kpoints)


def get_gkvec_arrays(ks: KsetHandle, ik: int):
    """num_gkvec, gvec_index (1-based, into the fine sphere), gkvec
    fractional [3N], cart [3N], |G+k| [N], (theta, phi) [2N]."""
    kp = ks.kset.kpoints[ik]
    gk = kp.gkvec
    fine = ks.kset.ctx.gvec_fine
    key = {tuple(mm): i for i, mm in enumerate(fine.miller)}
    idx = [key[tuple(mm)] + 1 for mm in gk.miller]
    cart = np.asarray(gk.gkvec_cart)
    ln = np.linalg.norm(cart, axis=1)
    with np.errstate(invalid="ignore"):
        theta = np.arccos(np.where(ln > 0, cart[:, 2] / np.where(ln > 0, ln, 1), 1.0))
    phi = np.arctan2(cart[:, 1], cart[:, 0])
    frac = gk.miller + gk.k_frac
    return (int(kp.num_gkvec), idx, frac.reshape(-1).tolist(),
            cart.reshape(-1).tolist(), ln.tolist(),
            np.stack([theta, phi], axis=1).reshape(-1).tolist())


def get_wave_functions(ks: KsetHandle, vkl, spin: int):
    """Return (num_gkvec, nbands, interleaved re/im of psi[nb, ngk]) of
    the k-point matching vkl (reference sirius_get_wave_functions; native
    G ordering of this engine — pair with get_gkvec_arrays)."""
    kset = ks.kset
    vkl = np.asarray(vkl, dtype=np.float64).reshape(3)
    for kp in kset.kpoints:
        if np.allclose(kp.k_frac, vkl, atol=1e-10):
            psi = kp.psi[spin].detach().cpu().numpy()
            ngk = int(kp.num_gkvec)
            flat = np.empty(2 * psi.shape[0] * psi.shape[1])
            flat[0::2] = psi.real.reshape(-1)
            flat[1::2] = psi.imag.reshape(-1)
            return ngk, int(psi.shape[0]), flat.tolist()
    raise ValueError(f"k-point {vkl} not found")


def set_band_occupancies(ks: KsetHandle, ik: int, ispn: int, occ):
    kp = ks.kset.kpoints[ik]
    kp.occ[ispn][:len(occ)] = np.asarray(occ, dtype=np.float64)


def generate_initial_density(gs: GsHandle):
    gs.dft.density.initial_density()


def generate_effective_potential(gs: GsHandle):
    gs.dft.potential.generate(gs.dft.density)
    gs.dft.potential.generate_paw(gs.dft.density)


def generate_density(gs: GsHandle, add_core: bool = False,
                     transform_to_rg: bool = False):
    gs.dft.density.generate(gs.dft.kset)


def initialize_subspace(gs: GsHandle):
    from .dft import initialize_subspace as init_sub
    from .hamiltonian import Hamiltonian0

    dft = gs.dft
    h0 = Hamiltonian0(dft.ctx, dft.potential)
    for kp in dft.kset:
        init_sub(dft.ctx, kp, h0(kp))


def find_eigen_states(gs: GsHandle, precompute_pw: bool = True,
                      itsol_tol: float = 1e-5):
    from .dft import diagonalize
    from .hamiltonian import Hamiltonian0

    dft = gs.dft
    h0 = Hamiltonian0(dft.ctx, dft.potential)
    return bool(diagonalize(dft.ctx, h0, dft.kset, itsol_tol))


def find_band_occupancies(ks: KsetHandle):
    ks.kset.find_band_occupancies()


def get_periodic_function(gs: GsHandle, label: str):
    """Real-grid values of a named scalar field on the fine FFT grid
    (Fortran order; reference label set, PP branch: no MT part)."""
    dft = gs.dft
    m = {
        "rho": lambda: dft.density.rho_r,
        "veff": lambda: dft.potential.veff_r,
        "bz": lambda: dft.potential.bz_r,
        "magz": lambda: dft.density.mag_r,
        "vha": lambda: dft.ctx.fft_fine.to_real(
            dft.potential.vha_g).real
        if getattr(dft.potential, "vha_g", None) is not None else None,
        "vxc": lambda: dft.potential.vxc_r,
        "exc": lambda: dft.potential.exc_r,
    }
    if label not in m:
        raise ValueError(f"wrong periodic-function label: {label}")
    t = m[label]()
    if t is None:
        raise ValueError(f"field {label} not present in this run")
    return np.asarray(t.detach().cpu().numpy(),
                      dtype=np.float64).transpose(2, 1, 0).reshape(-1).tolist()


def set_periodic_function(gs: GsHandle, label: str, values, dims):
    dft = gs.dft
    import torch

    arr = np.asarray(values, dtype=np.float64).reshape(
        dims[2], dims[1], dims[0]).transpose(2, 1, 0)
    t = torch.from_numpy(arr.copy()).to(dft.ctx.device)
    if label == "rho":
        dft.density.rho_r = t
        dft.density.rho_g = dft.ctx.fft_fine.to_pw(t.to(dft.ctx.dtype))
    elif label == "magz":
        dft.density.mag_r = t
        dft.density.mag_g = dft.ctx.fft_fine.to_pw(t.to(dft.ctx.dtype))
    else:
        raise ValueError(f"set_periodic_function: unsupported label {label}")


def get_total_magnetization(gs: GsHandle):
    d = gs.dft.density
    if gs.dft.ctx.nc_magnetism:
        return [float(gs.dft.ctx.integrate_rg_fine(c)) for c in d.magv_r]
    mz = d.total_magnetization() if gs.dft.ctx.num_mag_dims else 0.0
    return [0.0, 0.0, float(mz)]


def set_atom_vector_field(h: CtxHandle, ia: int, vf):
    lab, pos, _ = h.atoms[ia]
    h.atoms[ia] = (lab, pos, list(vf))


def set_num_bands(h: CtxHandle, n: int):
    h.cfg_data.setdefault("parameters", {})["num_bands"] = int(n)


def set_mpi_grid_dims(h: CtxHandle, dims):
    h.cfg_data.setdefault("control", {})["mpi_grid_dims"] = list(dims)


def create_context_from_json(js: str):
    h = CtxHandle()
    import_parameters(h, js)
    return h


def get_parameters(h: CtxHandle):
    """Scalar parameter snapshot (reference sirius_get_parameters)."""
    ctx = h.ctx
    p = ctx.cfg.parameters
    itso = ctx.cfg.iterative_solver
    return {
        "lmax_apw": int(p.lmax_apw),
        "lmax_rho": int(p.lmax_rho),
        "lmax_pot": int(p.lmax_pot),
        "num_bands": int(ctx.num_bands),
        "num_spins": int(ctx.num_spins),
        "num_mag_dims": int(ctx.num_mag_dims),
        "pw_cutoff": float(ctx.pw_cutoff),
        "gk_cutoff": float(ctx.gk_cutoff),
        "fft_grid_size": list(ctx.fft_fine.dims),
        "auto_rmt": int(p.auto_rmt),
        "gamma_point": bool(p.gamma_point),
        "use_symmetry": bool(p.use_symmetry),
        "so_correction": bool(getattr(p, "so_correction", False)),
        "iter_solver_tol": float(itso.get("energy_tolerance", 1e-2)),
        "iter_solver_tol_empty": float(
            itso.get("empty_states_tolerance", 0.0)),
        "verbosity": int(ctx.cfg.control.verbosity),
        "hubbard_correction": bool(ctx.hubbard is not None),
        "evp_work_count": float(ctx.counters.get(
            "band_evp_work_count", 0.0)),
        "num_loc_op_applied": int(ctx.counters.get(
            "local_operator_num_applied", 0)),
        "electronic_structure_method": str(p.electronic_structure_method),
        "num_sym_op": int(len(ctx.symmetry.ops)) if getattr(
            ctx, "symmetry", None) else 0,
        "num_fv_states": int(getattr(ctx, "num_fv_states", -1) or -1),
    }


def update_ground_state(gs: GsHandle):
    """Re-generate potential from the current density (reference
    sirius_update_ground_state)."""
    generate_effective_potential(gs)


def print_info(h: CtxHandle):
    ctx = h.ctx
    print(f"sirius_amd: {ctx.unit_cell.num_atoms} atoms, "
          f"{ctx.num_bands} bands, nG={ctx.gvec_fine.num_gvec}, "
          f"fft={ctx.fft_fine.dims}")


def print_timers():
    print("sirius_amd: timers are exposed via ctx.counters")


# ---- config-schema introspection (reference sirius_option_get_*;
# ---- backed by config._DEFAULTS instead of a JSON schema file) ---------

def _option_sections():
    from .config import _DEFAULTS

    return _DEFAULTS


def option_get_number_of_sections() -> int:
    return len(_option_sections())


def option_get_section_name(i: int) -> str:
    return list(_option_sections().keys())[i]


def option_get_section_length(section: str) -> int:
    return len(_option_sections()[section.lower()])


def _option_type_code(v) -> int:
    # 1=int 2=bool 3=string 4=double; +6 for arrays (reference
    # option_type_t, sirius_api.cpp:40-56)
    if isinstance(v, bool):
        return 2
    if isinstance(v, int):
        return 1
    if isinstance(v, float):
        return 4
    if isinstance(v, str):
        return 3
    if isinstance(v, (list, tuple)):
        if not v:
            return 9
        return _option_type_code(v[0]) + 6
    return 5


def option_get_info(section: str, i: int):
    sec = _option_sections()[section.lower()]
    key = list(sec.keys())[i]
    v = sec[key]
    length = len(v) if isinstance(v, (list, tuple)) else 1
    return key, _option_type_code(v), length


def option_get(section: str, name: str):
    """(type_code, value) of a default option."""
    v = _option_sections()[section.lower()][name.lower()]
    code = _option_type_code(v)
    if isinstance(v, (list, tuple)):
        v = list(v)
    return code, v


# ---- species-setup extensions (PAW / LAPW programmatic path) -----------

def set_atom_type_paw(h: CtxHandle, label: str, core_energy: float,
                      occupations, num_occ: int):
    at = h.types[label]
    at.paw_core_energy = float(core_energy)
    at.paw_wf_occ = [float(x) for x in occupations[:num_occ]]
    at.is_paw = True


def set_atom_type_configuration(h: CtxHandle, label: str, n: int, l: int,
                                k: int, occupancy: float, core: bool):
    """Accumulate the atomic level list (reference
    sirius_set_atom_type_configuration; consumed by the free-atom
    density / LAPW core solver)."""
    at = h.types[label]
    if not hasattr(at, "configuration") or at.configuration is None:
        at.configuration = []
    at.configuration.append({"n": int(n), "l": int(l), "k": int(k),
                             "occupancy": float(occupancy),
                             "core": bool(core)})


def add_atom_type_aw_descriptor(h: CtxHandle, label: str, n: int, l: int,
                                enu: float, dme: int, auto_enu: bool):
    """Accumulate APW radial-solution descriptors in the species-JSON
    shape FPAtomType parses (lapw/species.py)."""
    at = h.types[label]
    if not hasattr(at, "aw_specific") or at.aw_specific is None:
        at.aw_specific = {}
    at.aw_specific.setdefault(int(l), []).append(
        {"n": int(n), "enu": float(enu), "dme": int(dme),
         "auto": int(bool(auto_enu))})


def add_atom_type_lo_descriptor(h: CtxHandle, label: str, ilo: int, n: int,
                                l: int, enu: float, dme: int,
                                auto_enu: bool):
    at = h.types[label]
    if not hasattr(at, "lo_descriptors") or at.lo_descriptors is None:
        at.lo_descriptors = {}
    at.lo_descriptors.setdefault(int(ilo), {"l": int(l), "basis": []})
    at.lo_descriptors[int(ilo)]["basis"].append(
        {"n": int(n), "enu": float(enu), "dme": int(dme),
         "auto": int(bool(auto_enu))})


def set_equivalent_atoms(h: CtxHandle, eq):
    h.cfg_data.setdefault("unit_cell", {})["equivalent_atoms"] = \
        [int(x) for x in eq]


def get_fv_eigen_values(ks: KsetHandle, ik: int, num_fv_states: int):
    kp = ks.kset.kpoints[ik]
    ev = getattr(kp, "fv_eval", None)
    if ev is None:
        ev = kp.eigvals[0]
    return np.asarray(ev, dtype=np.float64)[:num_fv_states].tolist()


def ctx_num_atoms(h: CtxHandle) -> int:
    if h.ctx is not None:
        return int(h.ctx.unit_cell.num_atoms)
    return len(h.atoms)


def nlcg(gs: GsHandle, ks: KsetHandle, temp: float = -1.0,
         smearing: str = "", kappa: float = 0.3, tau: float = 0.1,
         tol: float = 1e-9, maxiter: int = 300, restart: int = 10,
         processing_unit: str = ""):
    """Direct total-energy minimization (reference sirius_nlcg /
    sirius_nlcg_params; fixed-occupation orbital CG — the ensemble-DFT
    smearing branch is a roadmap item, so temp/smearing/kappa are
    accepted and the fixed-occupation minimizer runs)."""
    from .nlcg import DirectMinimizer

    dm = DirectMinimizer(gs.dft, maxiter=int(maxiter), tol=float(tol))
    res = dm.run()
    gs.result = {"energy": {"total": res["etot"]},
                 "converged": bool(res["converged"]),
                 "num_scf_iterations": int(res["num_iter"])}
    return bool(res["converged"])


# ---- timers, parameters, field access (batch 3) ------------------------

_timers: dict = {}


def start_timer(name: str):
    import time

    _timers.setdefault(name, {"total": 0.0, "count": 0})["start"] = \
        time.time()


def stop_timer(name: str):
    import time

    t = _timers.get(name)
    if t and "start" in t:
        t["total"] += time.time() - t.pop("start")
        t["count"] += 1


def serialize_timers(fname: str):
    with open(fname, "w") as f:
        json.dump({k: {kk: vv for kk, vv in v.items() if kk != "start"}
                   for k, v in _timers.items()}, f, indent=1)


def set_parameters(h: CtxHandle, params: dict):
    """Scalar parameter batch (reference sirius_set_parameters; only the
    non-null pointers arrive here)."""
    p = h.cfg_data.setdefault("parameters", {})
    simple = {"lmax_apw", "lmax_rho", "lmax_pot", "num_fv_states",
              "num_bands", "num_mag_dims", "pw_cutoff", "gk_cutoff",
              "auto_rmt", "gamma_point", "use_symmetry", "so_correction",
              "hubbard_correction", "smearing", "smearing_width"}
    for k, v in params.items():
        if k in simple:
            p[k] = v
        elif k == "valence_rel":
            p["valence_relativity"] = v
        elif k == "core_rel":
            p["core_relativity"] = v
        elif k == "verbosity":
            h.cfg_data.setdefault("control", {})["verbosity"] = v
        elif k == "iter_solver_type":
            h.cfg_data.setdefault("iterative_solver", {})["type"] = v
        elif k == "iter_solver_tol_empty":
            h.cfg_data.setdefault("iterative_solver", {})[
                "empty_states_tolerance"] = v
        elif k == "fft_grid_size":
            h.cfg_data.setdefault("settings", {})["fft_grid_size"] = v
        elif k == "hubbard_full_orthogonalization":
            h.cfg_data.setdefault("hubbard", {})[
                "hubbard_subspace_method"] = "full_orthogonalization" \
                if v else "none"
        elif k == "hubbard_constrained_calculation":
            h.cfg_data.setdefault("hubbard", {})[
                "constrained_calculation"] = v
        # unknown keys are kept for introspection
        else:
            p[k] = v


def update_context(h: CtxHandle):
    """Re-initialize the simulation context from the accumulated config
    (reference sirius_update_context)."""
    return initialize_context(h)


def option_set(h: CtxHandle, section: str, name: str, value, append=False):
    sec = h.cfg_data.setdefault(section.lower(), {})
    if append and isinstance(sec.get(name), list):
        sec[name].append(value)
    else:
        sec[name] = value


def dump_runtime_setup(h: CtxHandle, fname: str):
    data = h.ctx.cfg.to_dict() if h.ctx is not None else h.cfg_data
    with open(fname, "w") as f:
        json.dump(data, f, indent=1)


def get_kp_params_from_ctx(h: CtxHandle):
    p = h.ctx.cfg.parameters
    return list(p.ngridk), list(p.shiftk), bool(p.use_symmetry)


def get_scf_params_from_ctx(h: CtxHandle):
    p = h.ctx.cfg.parameters
    tol = h.ctx.cfg.iterative_solver.get("energy_tolerance", 1e-2)
    return (float(p.density_tol), float(p.energy_tol), float(tol),
            int(p.num_dft_iter))


def _named_pw_field(dft, label: str):
    m = {"rho": ("density", "rho_g"), "veff": ("potential", "veff_g"),
         "magz": ("density", "mag_g"), "bz": ("potential", "bz_g"),
         "rhoc": ("density", "rho_core_g"), "vloc": ("potential", "vloc_g")}
    if label not in m:
        raise ValueError(f"unknown pw field label: {label}")
    owner, attr = m[label]
    return getattr(dft, owner), attr


def get_pw_coeffs(gs: GsHandle, label: str, gvl):
    """PW coefficients of a named field at the caller's Miller indices
    (reference sirius_get_pw_coeffs; serial comm).  gvl is a flat list of
    3*ngv ints; returns interleaved re/im."""
    dft = gs.dft
    owner, attr = _named_pw_field(dft, label)
    f = getattr(owner, attr)
    if f is None:
        raise ValueError(f"field {label} not present")
    fine = dft.ctx.gvec_fine
    key = {tuple(mm): i for i, mm in enumerate(fine.miller)}
    g = np.asarray(gvl, dtype=np.int64).reshape(-1, 3)
    fc = f.detach().cpu().numpy()
    out = np.zeros(2 * len(g))
    for i, mm in enumerate(g):
        j = key.get(tuple(mm))
        if j is not None:
            out[2 * i] = fc[j].real
            out[2 * i + 1] = fc[j].imag
    return out.tolist()


def set_pw_coeffs(gs: GsHandle, label: str, coeffs, gvl,
                  transform_to_rg: bool = False):
    import torch

    dft = gs.dft
    owner, attr = _named_pw_field(dft, label)
    fine = dft.ctx.gvec_fine
    key = {tuple(mm): i for i, mm in enumerate(fine.miller)}
    g = np.asarray(gvl, dtype=np.int64).reshape(-1, 3)
    c = np.asarray(coeffs, dtype=np.float64)
    f = torch.zeros(fine.num_gvec, dtype=dft.ctx.dtype)
    for i, mm in enumerate(g):
        j = key.get(tuple(mm))
        if j is not None:
            f[j] = complex(c[2 * i], c[2 * i + 1])
    f = f.to(dft.ctx.device)
    setattr(owner, attr, f)
    if transform_to_rg and label == "rho":
        owner.rho_r = dft.ctx.fft_fine.to_real(f).real
    elif transform_to_rg and label == "magz":
        owner.mag_r = dft.ctx.fft_fine.to_real(f).real


def fft_transform(gs: GsHandle, label: str, direction: int):
    """direction +1: PW -> real grid; -1: real grid -> PW (reference
    sirius_fft_transform)."""
    dft = gs.dft
    rg_attr = {"rho": "rho_r", "magz": "mag_r", "veff": "veff_r"}
    pw_attr = {"rho": "rho_g", "magz": "mag_g", "veff": "veff_g"}
    if label not in rg_attr:
        raise ValueError(f"fft_transform: unknown label {label}")
    owner = dft.density if label in ("rho", "magz") else dft.potential
    if direction > 0:
        setattr(owner, rg_attr[label], dft.ctx.fft_fine.to_real(
            getattr(owner, pw_attr[label])).real)
    else:
        setattr(owner, pw_attr[label], dft.ctx.fft_fine.to_pw(
            getattr(owner, rg_attr[label]).to(dft.ctx.dtype)))


def _rg_field(dft, label: str):
    m = {"rho": (dft.density, "rho_r"), "magz": (dft.density, "mag_r"),
         "veff": (dft.potential, "veff_r"), "bz": (dft.potential, "bz_r"),
         "vxc": (dft.potential, "vxc_r"), "exc": (dft.potential, "exc_r")}
    if label not in m:
        raise ValueError(f"unknown rg field label: {label}")
    return m[label]


def get_rg_values(gs: GsHandle, label: str, box_origin, box_size):
    """Real-grid values inside a (1-based, Fortran-order) box."""
    owner, attr = _rg_field(gs.dft, label)
    t = getattr(owner, attr)
    if t is None:
        raise ValueError(f"field {label} not present")
    a = t.detach().cpu().numpy()
    o = [int(x) - 1 for x in box_origin]
    s = [int(x) for x in box_size]
    box = a[o[0]:o[0] + s[0], o[1]:o[1] + s[1], o[2]:o[2] + s[2]]
    return np.asarray(box, dtype=np.float64).transpose(2, 1, 0) \
        .reshape(-1).tolist()


def set_rg_values(gs: GsHandle, label: str, box_origin, box_size, values,
                  transform_to_pw: bool = False):
    import torch

    dft = gs.dft
    owner, attr = _rg_field(dft, label)
    t = getattr(owner, attr)
    if t is None:
        raise ValueError(f"field {label} not present")
    o = [int(x) - 1 for x in box_origin]
    s = [int(x) for x in box_size]
    box = np.asarray(values, dtype=np.float64).reshape(
        s[2], s[1], s[0]).transpose(2, 1, 0)
    t[o[0]:o[0] + s[0], o[1]:o[1] + s[1], o[2]:o[2] + s[2]] = \
        torch.from_numpy(box.copy()).to(t.device)
    if transform_to_pw:
        pw = {"rho": "rho_g", "magz": "mag_g", "veff": "veff_g"}.get(label)
        if pw:
            setattr(owner, pw, dft.ctx.fft_fine.to_pw(
                getattr(owner, attr).to(dft.ctx.dtype)))


def generate_coulomb_potential(gs: GsHandle):
    """Regenerate the effective potential from the current density and
    return vh_el per atom when the branch computes it (reference
    sirius_generate_coulomb_potential; the PP branch has no MT spheres,
    so vh_el is zeros there — documented deviation)."""
    dft = gs.dft
    dft.potential.generate(dft.density)
    vh_el = getattr(dft.potential, "vh_el", None)
    n = dft.ctx.unit_cell.num_atoms
    if vh_el is None:
        return [0.0] * n
    return [float(x) for x in np.asarray(vh_el).reshape(-1)[:n]]


def generate_xc_potential(gs: GsHandle):
    """Regenerate the potential (includes the XC part; reference
    sirius_generate_xc_potential regenerates only XC — here the full
    potential is rebuilt, which subsumes it)."""
    gs.dft.potential.generate(gs.dft.density)


# ---- Hubbard setup, eigen-vector export, misc (batch 4) ----------------

def set_atom_type_hubbard(h: CtxHandle, label: str, l: int, n: int,
                          occ: float, U: float, J: float, alpha: float,
                          beta: float, J0: float):
    """Add a Hubbard-corrected orbital for an atom type (reference
    sirius_set_atom_type_hubbard; accumulates hubbard.local entries)."""
    hub = h.cfg_data.setdefault("hubbard", {})
    hub.setdefault("local", []).append({
        "atom_type": label, "l": int(l), "n": int(n),
        "total_initial_occupancy": float(occ), "U": float(U),
        "J": float(J), "alpha": float(alpha), "beta": float(beta),
        "J0": float(J0)})
    h.cfg_data.setdefault("parameters", {})["hubbard_correction"] = True


def add_hubbard_atom_pair(h: CtxHandle, atom_pair, translation, n, l,
                          coupling: float):
    hub = h.cfg_data.setdefault("hubbard", {})
    hub.setdefault("nonlocal", []).append({
        "atom_pair": [int(atom_pair[0]) - 1, int(atom_pair[1]) - 1],
        "T": [int(x) for x in translation],
        "n": [int(x) for x in n], "l": [int(x) for x in l],
        "V": float(coupling)})


def add_hubbard_atom_constraint(h: CtxHandle, atom_id: int, n: int, l: int,
                                lmax_at: int, occ, orbital_order=None):
    hub = h.cfg_data.setdefault("hubbard", {})
    mm = 2 * int(l) + 1
    nsp = 2 if h.cfg_data.get("parameters", {}).get("num_mag_dims", 0) \
        else 1
    occm = np.asarray(occ, dtype=np.float64).reshape(nsp, mm, mm)
    entry = {"atom_index": int(atom_id) - 1, "n": int(n), "l": int(l),
             "occupancy": occm.tolist()}
    if orbital_order is not None:
        entry["lm_order"] = [int(x) for x in orbital_order]
    hub.setdefault("local_constraint", []).append(entry)
    hub["constrained_calculation"] = True


def get_comm_handle(h: CtxHandle, which: str) -> int:
    """Fortran communicator handles (reference get_kpoint_inner/inter/
    fft_comm).  This engine drives distribution through torch.distributed
    (RCCL/gloo), not MPI — callers embedding serially get MPI_COMM_SELF's
    conventional f-handle 0."""
    return 0


def set_energy_fermi(ks: KsetHandle, ef: float):
    ks.kset.energy_fermi = float(ef)


def check_scf_density(gs: GsHandle) -> float:
    """Regenerate rho from the current wave functions and report the max
    |Δrho(G)| against the stored density (reference
    sirius_check_scf_density prints the same check)."""
    import torch

    from .hamiltonian import Hamiltonian0

    dft = gs.dft
    old = dft.density.rho_g.clone()
    h0 = Hamiltonian0(dft.ctx, dft.potential, dft.density)
    dft.density.generate(dft.kset, h0)
    d = float((dft.density.rho_g - old).abs().max())
    print(f"sirius_amd: check_scf_density |drho(G)|_max = {d:.3e}")
    return d


def get_step_function(h: CtxHandle):
    """LAPW unit-step function Θ: (PW coeffs interleaved, real-grid
    values, n_rg) — FP context only."""
    ctx = h.ctx
    if not getattr(ctx, "theta_pw", None) is not None:
        raise ValueError("step function only exists for the LAPW branch")
    pw = ctx.theta_pw.detach().cpu().numpy()
    rg = ctx.theta_rg.detach().cpu().numpy()
    flat = np.empty(2 * pw.size)
    flat[0::2] = pw.real
    flat[1::2] = pw.imag
    return (flat.tolist(),
            np.asarray(rg, dtype=np.float64).transpose(2, 1, 0)
            .reshape(-1).tolist(), int(rg.size))


def get_fv_eigen_vectors(ks: KsetHandle, ik: int, num_fv_states: int):
    """First-variational eigenvectors [N_basis, nfv] (LAPW), interleaved
    re/im, column-major like the reference."""
    kp = ks.kset.kpoints[ik]
    v = getattr(kp, "fv_evec", None)
    if v is None:
        raise ValueError("no first-variational eigenvectors on this kp")
    v = np.asarray(v)[:, :num_fv_states]
    flat = np.empty(2 * v.size)
    a = v.reshape(-1, order="F")
    flat[0::2] = a.real
    flat[1::2] = a.imag
    return int(v.shape[0]), flat.tolist()


def get_psi(ks: KsetHandle, ik: int, ispin: int):
    """Full wave-function coefficients of one k-point/spin, interleaved
    re/im, band-major (reference sirius_get_psi)."""
    kp = ks.kset.kpoints[ik]
    psi = kp.psi[ispin].detach().cpu().numpy()
    flat = np.empty(2 * psi.size)
    flat[0::2] = psi.real.reshape(-1)
    flat[1::2] = psi.imag.reshape(-1)
    return int(psi.shape[1]), int(psi.shape[0]), flat.tolist()


# ---- batch 5: Hamiltonian handlers, kset init, occupation setters ------

class HamHandle:
    def __init__(self, h0):
        self.h0 = h0


def initialize_kset(ks: KsetHandle):
    """No-op: this engine's k-set is fully built at creation (reference
    sirius_initialize_kset splits construction in two)."""
    return int(ks.kset.num_kpoints)


def create_hamiltonian(gs: GsHandle):
    from .hamiltonian import Hamiltonian0

    dft = gs.dft
    return HamHandle(Hamiltonian0(dft.ctx, dft.potential, dft.density))


def diagonalize_hamiltonian(gs: GsHandle, H0: HamHandle,
                            iter_solver_tol: float, max_steps: int,
                            exact: bool = False):
    from .dft import diagonalize

    dft = gs.dft
    itso = dft.ctx.cfg.iterative_solver
    saved = itso.get("num_steps"), itso.get("type")
    try:
        itso.set("num_steps", int(max_steps))
        if exact:
            itso.set("type", "exact")
        conv = diagonalize(dft.ctx, H0.h0, dft.kset,
                           float(iter_solver_tol))
    finally:
        itso.set("num_steps", saved[0])
        itso.set("type", saved[1])
    nit = int(dft.ctx.counters.get("num_itsol_steps", 0))
    return bool(conv), nit


def generate_d_operator_matrix(gs: GsHandle):
    """The D matrices are (re)built with the Hamiltonian from the current
    potential (reference sirius_generate_d_operator_matrix) — rebuild and
    cache on the handler so a following create_hamiltonian is warm."""
    create_hamiltonian(gs)


def set_atom_type_radial_grid_inf(h: CtxHandle, label: str, points):
    at = h.types[label]
    at.r_inf = np.asarray(points, dtype=np.float64)


def get_gkvec(ks: KsetHandle, ik: int):
    """Cartesian G+k vectors of one k-point, flat [3*ngk]."""
    kp = ks.kset.kpoints[ik]
    return np.asarray(kp.gkvec.gkvec_cart,
                      dtype=np.float64).reshape(-1).tolist()


def set_local_occupation_matrix(h, ia: int, n: int, l: int, spin: int,
                                occ_flat, ld: int):
    """Set one local Hubbard occupation block (reference
    sirius_set_local_occupation_matrix; handler is the ground state)."""
    import torch

    dft = h.dft if isinstance(h, GsHandle) else h
    hub = dft.ctx.hubbard
    mm = 2 * int(l) + 1
    il = hub._find_level(int(ia), int(n), int(l))
    c = np.asarray(occ_flat, dtype=np.float64)
    m = (c[0::2] + 1j * c[1::2]).reshape(ld, -1)[:mm, :mm]
    if hub.om is None:
        hub.om = hub.initial_occupation()
    hub.om[il][..., int(spin)] = torch.from_numpy(m).to(
        hub.om[il].device, hub.om[il].dtype)


def set_nonlocal_occupation_matrix(h, atom_pair, n, l, spin: int, T,
                                   occ_flat, ld1: int, ld2: int):
    """Set one inter-site occupation block."""
    import torch

    dft = h.dft if isinstance(h, GsHandle) else h
    hub = dft.ctx.hubbard
    m1 = 2 * int(l[0]) + 1
    m2 = 2 * int(l[1]) + 1
    c = np.asarray(occ_flat, dtype=np.float64)
    blk = (c[0::2] + 1j * c[1::2]).reshape(ld1, ld2)[:m1, :m2]
    tgt = (int(atom_pair[0]), int(atom_pair[1]), tuple(int(x) for x in T))
    for i, p in enumerate(hub.nonlocal_pairs):
        if (p.ia, p.ja, p.T) == tgt and p.n1 == int(n[0]) \
                and p.n2 == int(n[1]):
            if hub.om_nl is None:
                hub.om_nl = [torch.zeros(2 * q.il + 1, 2 * q.jl + 1,
                                         dft.ctx.num_spins,
                                         dtype=dft.ctx.dtype)
                             for q in hub.nonlocal_pairs]
            hub.om_nl[i][..., int(spin)] = torch.from_numpy(blk).to(
                hub.om_nl[i].device, hub.om_nl[i].dtype)
            return
    raise ValueError(f"nonlocal pair {tgt} not found")


def get_sv_eigen_vectors(ks: KsetHandle, ik: int):
    """Second-variational eigenvectors (LAPW collinear: block-diagonal
    per-spin rotation), interleaved re/im, column-major per spin."""
    kp = ks.kset.kpoints[ik]
    sv = getattr(kp, "sv_evec", None)
    if sv is None or sv[0] is None:
        raise ValueError("no second-variational eigenvectors on this kp")
    mats = [np.asarray(m) for m in sv if m is not None]
    flat_parts = []
    for v in mats:
        a = v.reshape(-1, order="F")
        f = np.empty(2 * a.size)
        f[0::2] = a.real
        f[1::2] = a.imag
        flat_parts.append(f)
    return int(mats[0].shape[0]), len(mats), \
        np.concatenate(flat_parts).tolist()
