"""Unit cell, atom types and pseudopotential ingestion.

Reference behavior: src/unit_cell/unit_cell.hpp:33 (Unit_cell),
src/unit_cell/atom_type.hpp:86 (Atom_type; UPF-as-JSON parsing
atom_type.cpp:498 `read_pseudo_uspp`, :644 `read_pseudo_paw`, UPF v2 XML
:771). Units: Hartree atomic units. UPF-JSON files are already in Ha;
UPF v2 XML carries Ry for PP_LOCAL / PP_DIJ (converted here ×0.5, as the
reference does at atom_type.cpp:789/:864).
"""

from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass, field

import numpy as np


@dataclass
class BetaProjector:
    l: int                      # angular momentum
    j: float | None             # total angular momentum (SO pseudos), else None
    f_r: np.ndarray             # r*beta(r) on the radial grid (UPF convention)


@dataclass
class AtomicWf:
    n: int
    l: int
    occ: float
    f_r: np.ndarray             # r*chi(r) (UPF convention)


@dataclass
class QRadialFunction:
    i: int
    j: int
    l: int
    f_r: np.ndarray


class AtomType:
    """One atomic species with its pseudopotential data."""

    def __init__(self, label: str):
        self.label = label
        self.symbol = label
        self.zn = 0                      # valence charge Z_p
        self.r = np.zeros(0)             # radial grid
        self.vloc_r = np.zeros(0)        # local potential V(r) [Ha]
        self.rho_core_r = np.zeros(0)    # core charge density rho_c(r) (plain)
        self.rho_total_4pir2 = np.zeros(0)  # 4*pi*r^2*rho(r) (UPF PP_RHOATOM)
        self.beta: list[BetaProjector] = []
        self.atomic_wfs: list[AtomicWf] = []
        self.d_ion = np.zeros((0, 0))    # ionic D matrix [Ha]
        self.q_radial: list[QRadialFunction] = []
        self.is_ultrasoft = False
        self.is_paw = False
        self.is_norm_conserving = True
        self.spin_orbit = False
        self.core_correction = False
        # PAW extras
        self.paw_core_energy = 0.0
        self.paw_ae_wfs: list[np.ndarray] = []
        self.paw_ps_wfs: list[np.ndarray] = []
        self.paw_ae_core = np.zeros(0)
        self.paw_wf_occ: list[float] = []
        self.paw_cutoff_index = 0

    # -- parsing ----------------------------------------------------------

    @classmethod
    def from_file(cls, label: str, path: str) -> "AtomType":
        if path.endswith(".json"):
            with open(path) as f:
                return cls.from_upf_json(label, json.load(f))
        # UPF v2 XML
        return cls.from_upf_xml(label, path)

    @classmethod
    def from_upf_json(cls, label: str, d: dict) -> "AtomType":
        at = cls(label)
        pp = d["pseudo_potential"]
        h = pp["header"]
        at.symbol = h.get("element", label).strip()
        at.zn = int(h["z_valence"] + 1e-10)
        at.r = np.asarray(pp["radial_grid"], dtype=np.float64)
        at.vloc_r = np.asarray(pp["local_potential"], dtype=np.float64)
        at.rho_core_r = np.asarray(
            pp.get("core_charge_density", np.zeros_like(at.r)), dtype=np.float64)
        at.rho_total_4pir2 = np.asarray(pp["total_charge_density"], dtype=np.float64)
        at.spin_orbit = bool(h.get("spin_orbit", False))
        at.core_correction = bool(h.get("core_correction", False))
        at.is_ultrasoft = bool(h.get("is_ultrasoft", False)) or h.get("pseudo_type") in ("US", "USPP")
        ptype = h.get("pseudo_type", "NC")
        at.is_paw = ptype in ("PAW",)
        at.is_norm_conserving = not (at.is_ultrasoft or at.is_paw)

        nbf = int(h.get("number_of_proj", 0))
        for i in range(nbf):
            b = pp["beta_projectors"][i]
            f = np.zeros_like(at.r)
            fr = np.asarray(b["radial_function"], dtype=np.float64)
            f[: len(fr)] = fr[: len(at.r)]
            l = int(b["angular_momentum"])
            jtot = b.get("total_angular_momentum") if at.spin_orbit else None
            at.beta.append(BetaProjector(l=l, j=jtot, f_r=f))

        if nbf:
            v = np.asarray(pp["D_ion"], dtype=np.float64)
            at.d_ion = v.reshape(nbf, nbf).T.copy()  # stored column-major (atom_type.cpp:575-580)

        for a in pp.get("augmentation", []):
            at.q_radial.append(QRadialFunction(
                i=int(a["i"]), j=int(a["j"]), l=int(a["angular_momentum"]),
                f_r=np.asarray(a["radial_function"], dtype=np.float64)))

        for w in pp.get("atomic_wave_functions", []):
            f = np.asarray(w["radial_function"], dtype=np.float64)
            lab = w.get("label", "")
            n = int(lab[0]) if lab and lab[0].isdigit() else -1
            at.atomic_wfs.append(AtomicWf(
                n=n, l=int(w["angular_momentum"]),
                occ=float(w.get("occupation", 0.0)), f_r=f))

        if at.is_paw:
            at.paw_core_energy = float(h.get("paw_core_energy", 0.0))
            at.paw_cutoff_index = int(h.get("cutoff_radius_index", len(at.r)))
            paw = pp.get("paw_data", {})
            at.paw_ae_core = np.asarray(
                paw.get("ae_core_charge_density", np.zeros_like(at.r)),
                dtype=np.float64)
            at.paw_wf_occ = [float(x) for x in paw.get("occupations", [])]
            ncut = at.paw_cutoff_index
            for w in paw.get("ae_wfc", []):
                f = np.zeros_like(at.r)
                v = np.asarray(w["radial_function"], dtype=np.float64)[:ncut]
                f[:len(v)] = v
                at.paw_ae_wfs.append(f)
            for w in paw.get("ps_wfc", []):
                f = np.zeros_like(at.r)
                v = np.asarray(w["radial_function"], dtype=np.float64)[:ncut]
                f[:len(v)] = v
                at.paw_ps_wfs.append(f)

        return at

    @classmethod
    def from_upf_xml(cls, label: str, path: str) -> "AtomType":
        """UPF v2 XML (reference: atom_type.cpp:771 via pugixml; here ElementTree)."""
        import xml.etree.ElementTree as ET

        at = cls(label)
        root = ET.parse(path).getroot()
        if root.tag != "UPF":
            upf = root.find("UPF")
            root = upf if upf is not None else root
        h = root.find("PP_HEADER").attrib
        at.symbol = h["element"].strip()
        at.zn = int(float(h["z_valence"]) + 1e-10)
        at.spin_orbit = h.get("has_so", "F").upper().startswith("T")
        at.core_correction = h.get("core_correction", "F").upper().startswith("T")
        at.is_ultrasoft = h.get("is_ultrasoft", "F").upper().startswith("T")
        at.is_paw = h.get("is_paw", "F").upper().startswith("T")
        at.is_norm_conserving = not (at.is_ultrasoft or at.is_paw)

        def vec(node, scale=1.0):
            return scale * np.fromstring(node.text.replace("\n", " "), sep=" ")

        mesh = root.find("PP_MESH")
        at.r = vec(mesh.find("PP_R"))
        at.vloc_r = vec(root.find("PP_LOCAL"), 0.5)   # Ry -> Ha
        nlcc = root.find("PP_NLCC")
        at.rho_core_r = vec(nlcc) if nlcc is not None else np.zeros_like(at.r)
        at.rho_total_4pir2 = vec(root.find("PP_RHOATOM"))

        nl = root.find("PP_NONLOCAL")
        nbf = int(h.get("number_of_proj", 0))
        for i in range(nbf):
            b = nl.find(f"PP_BETA.{i + 1}")
            f = np.zeros_like(at.r)
            fr = vec(b)
            # truncate at the projector's own cutoff (atom_type.cpp:45-56:
            # attribute, else the last |v| > 1e-80)
            nr = int(float(b.attrib.get("cutoff_radius_index", 0)))
            if nr == 0:
                nz = np.nonzero(np.abs(fr) > 1e-80)[0]
                nr = int(nz[-1]) + 1 if len(nz) else len(fr)
            n = min(nr, len(at.r), len(fr))
            f[:n] = fr[:n]
            at.beta.append(BetaProjector(l=int(b.attrib["angular_momentum"]), j=None, f_r=f))
        if nbf:
            at.d_ion = vec(nl.find("PP_DIJ"), 0.5).reshape(nbf, nbf)  # Ry -> Ha
        aug = nl.find("PP_AUGMENTATION")
        if aug is not None:
            for ch in aug:
                if not ch.tag.startswith("PP_QIJL"):
                    continue
                # tag PP_QIJL.i.j.l (1-based i,j)
                parts = ch.tag.split(".")
                i, j, l = int(parts[1]) - 1, int(parts[2]) - 1, int(parts[3])
                f = np.zeros_like(at.r)
                v = vec(ch)
                f[:min(len(v), len(f))] = v[:len(f)]
                at.q_radial.append(QRadialFunction(i=i, j=j, l=l, f_r=f))

        if at.is_paw:
            aug_node = nl.find("PP_AUGMENTATION")
            at.paw_cutoff_index = int(float(aug_node.attrib.get(
                "cutoff_r_index", len(at.r)))) if aug_node is not None else len(at.r)
            paw = root.find("PP_PAW")
            if paw is not None:
                at.paw_core_energy = 0.5 * float(paw.attrib.get("core_energy", 0.0))
                nlcc_ae = paw.find("PP_AE_NLCC")
                if nlcc_ae is not None:
                    at.paw_ae_core = vec(nlcc_ae)
                occ = paw.find("PP_OCCUPATIONS")
                if occ is not None:
                    at.paw_wf_occ = list(vec(occ))
            full = root.find("PP_FULL_WFC")
            ncut = at.paw_cutoff_index
            if full is not None:
                for i in range(nbf):
                    for tag, dest in (("PP_AEWFC", at.paw_ae_wfs),
                                      ("PP_PSWFC", at.paw_ps_wfs)):
                        node = full.find(f"{tag}.{i + 1}")
                        f = np.zeros_like(at.r)
                        if node is not None:
                            v = vec(node)[:ncut]
                            f[:len(v)] = v
                        dest.append(f)
            if at.paw_ae_core.size == 0:
                at.paw_ae_core = np.zeros_like(at.r)

        pswfc = root.find("PP_PSWFC")
        if pswfc is not None:
            for ch in pswfc:
                if not ch.tag.startswith("PP_CHI"):
                    continue
                lab = ch.attrib.get("label", "")
                n = int(lab[0]) if lab and lab[0].isdigit() else -1
                at.atomic_wfs.append(AtomicWf(
                    n=n, l=int(ch.attrib["l"]),
                    occ=float(ch.attrib.get("occupation", 0.0)), f_r=vec(ch)))
        return at

    # -- derived ----------------------------------------------------------

    @property
    def num_beta(self) -> int:
        return len(self.beta)

    @property
    def augment(self) -> bool:
        """True when the type carries augmentation charges (USPP/PAW)."""
        return len(self.q_radial) > 0

    @property
    def num_beta_lm(self) -> int:
        """Total number of beta projectors including m-degeneracy."""
        return sum(2 * b.l + 1 for b in self.beta)

    def beta_lm_index(self):
        """List of (idxrf, l, m) over the full lm-resolved beta set,
        ordered radial-function-major, m from -l..l (reference ordering:
        atom_type indexb)."""
        out = []
        for i, b in enumerate(self.beta):
            for m in range(-b.l, b.l + 1):
                out.append((i, b.l, m))
        return out


class UnitCell:
    """Lattice + atoms (reference: src/unit_cell/unit_cell.hpp:33)."""

    def __init__(self, lattice: np.ndarray, atom_types: dict[str, AtomType],
                 positions: list[tuple[str, np.ndarray]]):
        """lattice: rows are lattice vectors a1,a2,a3 (Bohr).
        positions: list of (type_label, fractional coordinate)."""
        self.lattice = np.asarray(lattice, dtype=np.float64)
        self.recip = 2 * math.pi * np.linalg.inv(self.lattice).T  # rows b1,b2,b3
        self.omega = abs(np.linalg.det(self.lattice))
        self.atom_types = atom_types
        self.type_labels = list(atom_types.keys())
        self.atoms = [(lab, np.asarray(pos, dtype=np.float64)) for lab, pos in positions]
        self.vector_fields = np.zeros((len(self.atoms), 3))  # initial moments

    @classmethod
    def from_config(cls, cfg, base_dir: str = ".") -> "UnitCell":
        uc = cfg.unit_cell
        scale = uc.get("lattice_vectors_scale", 1.0)
        lattice = np.asarray(uc.lattice_vectors, dtype=np.float64) * scale
        types = {}
        for lab in uc.atom_types:
            fname = uc.atom_files.get(lab, f"{lab}.json")
            path = fname if os.path.isabs(fname) else os.path.join(base_dir, fname)
            types[lab] = AtomType.from_file(lab, path)
        units = uc.get("atom_coordinate_units", "lattice")
        inv_lat = np.linalg.inv(lattice)
        from .constants import bohr_to_ang

        positions = []
        vfields = []
        # global atom order follows the atom_types list (reference behavior:
        # atoms are added per type in unit_cell initialization)
        type_order = list(uc.atom_types) + [t for t in uc.atoms if t not in uc.atom_types]
        for lab in type_order:
            plist = uc.atoms.get(lab, [])
            for p in plist:
                p = np.asarray(p, dtype=np.float64)
                pos = p[:3]
                if units in ("au", "a.u."):
                    pos = pos @ inv_lat          # cartesian bohr -> fractional
                elif units in ("A", "angstrom"):
                    pos = (pos / bohr_to_ang) @ inv_lat
                positions.append((lab, pos))
                vf = np.zeros(3)
                if len(p) >= 6:
                    vf = p[3:6]
                elif len(p) == 4:
                    vf[2] = p[3]
                vfields.append(vf)
        cell = cls(lattice, types, positions)
        cell.vector_fields = np.array(vfields)  # initial moments (atom entries 4-6)
        return cell

    # -- queries -----------------------------------------------------------

    @property
    def num_atoms(self) -> int:
        return len(self.atoms)

    def atom_positions_frac(self) -> np.ndarray:
        return np.array([p for _, p in self.atoms])

    def atom_positions_cart(self) -> np.ndarray:
        return self.atom_positions_frac() @ self.lattice

    def atoms_of_type(self, label: str) -> np.ndarray:
        """Indices of atoms of a given type."""
        return np.array([i for i, (lab, _) in enumerate(self.atoms) if lab == label],
                        dtype=np.int64)

    @property
    def num_electrons(self) -> float:
        return float(sum(self.atom_types[lab].zn for lab, _ in self.atoms))

    def nearest_neighbours(self, r_cut: float):
        """(i, j, distance) pairs within r_cut, for the Ewald real-space
        sum (reference: unit_cell nearest-neighbour list used in energy.cpp:52-60)."""
        return [(ia, ja, d) for ia, ja, d, _, _ in
                self.nearest_neighbours_full(r_cut)]

    def nearest_neighbours_full(self, r_cut: float):
        """(ia, ja, distance, T_int [3], rc [3]) within r_cut; rc is the
        Cartesian connecting vector pos(ja)+T·A−pos(ia) and T the integer
        lattice translation (reference: Unit_cell::find_nearest_neighbours,
        unit_cell.cpp:389-437 — nnd.translation and nnd.rc). Needed by the
        Ewald force/stress real-space sums."""
        latt = self.lattice
        pos = self.atom_positions_cart()
        inv_len = np.linalg.norm(np.linalg.inv(latt), axis=0)  # 1/interplanar dist
        nmax = np.ceil(r_cut * inv_len).astype(int) + 1
        tint = np.array([(i, j, k)
                         for i in range(-nmax[0], nmax[0] + 1)
                         for j in range(-nmax[1], nmax[1] + 1)
                         for k in range(-nmax[2], nmax[2] + 1)], dtype=np.int64)
        svec = tint.astype(np.float64) @ latt
        pairs = []
        for ia in range(self.num_atoms):
            for ja in range(self.num_atoms):
                rc = pos[ja] + svec - pos[ia]
                d = np.linalg.norm(rc, axis=1)
                sel = np.nonzero((d > 1e-8) & (d < r_cut))[0]
                for s in sel:
                    pairs.append((ia, ja, d[s], tint[s], rc[s]))
        return pairs
