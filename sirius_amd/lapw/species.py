"""Full-potential LAPW species files (He.json-style).

Reference behavior: src/unit_cell/atom_type.cpp:699-741 (read_input, FP
branch), :375-436 (read_input_core), :438-463 (read_input_aw), :465-496
(read_input_lo), and atom_type.hpp:305-328 (init_aw_descriptors: the
default AW basis is replicated for every l = 0..lmax_apw with n = l+1;
l-specific sets override when l < lmax_apw).
"""

from __future__ import annotations

import json
import math
from dataclasses import dataclass, field

import numpy as np

from .atomic_conf import atomic_configuration

_L_BY_CHAR = {"s": 0, "p": 1, "d": 2, "f": 3}


@dataclass
class RSD:
    """Radial solution descriptor (one radial function of the basis)."""
    n: int = -1
    l: int = -1
    enu: float = 0.15
    dme: int = 0
    auto: int = 0


@dataclass
class LocalOrbital:
    l: int
    rsd_set: list = field(default_factory=list)


class FPAtomType:
    """One LAPW species parsed from a JSON file."""

    def __init__(self, label: str, data: dict, lmax_apw: int = 8,
                 radial_grid: str = "exponential, 1.0"):
        self.label = label
        self.name = data.get("name", label)
        self.symbol = data["symbol"]
        self.mass = float(data.get("mass", 0.0))
        self.zn = int(data["number"])
        self.rmin = float(data["rmin"])
        self.rmt = float(data["rmt"])
        self.nmtp = int(data["nrmt"])
        self.lmax_apw = int(data.get("lmax_apw", lmax_apw))

        # MT radial grid (reference Radial_grid_exp: x_i = r0 (R/r0)^{t^p})
        kind, p = [s.strip() for s in radial_grid.split(",")]
        p = float(p)
        self._grid_p = p
        t = (np.arange(self.nmtp) / (self.nmtp - 1)) ** p
        if kind != "exponential":
            raise ValueError(f"unsupported radial grid: {kind}")
        self.r = self.rmin * (self.rmt / self.rmin) ** t
        self.r[0] = self.rmin
        self.r[-1] = self.rmt

        # atomic levels: mark those named in the "core" string as core
        core_str = data.get("core", "").strip()
        core_shells = set()
        i = 0
        while i < len(core_str):
            n = int(core_str[i])
            l = _L_BY_CHAR[core_str[i + 1]]
            core_shells.add((n, l))
            i += 2
        self.atomic_levels = []   # (n, l, k, occ, is_core)
        for (n, l, k, occ) in atomic_configuration(self.zn):
            self.atomic_levels.append((n, l, k, occ, (n, l) in core_shells))
        self.num_core_electrons = sum(occ for (_, _, _, occ, c)
                                      in self.atomic_levels if c)
        self.num_valence_electrons = self.zn - self.num_core_electrons

        # augmented-wave descriptors
        val = data["valence"]
        default = [RSD(enu=float(b["enu"]), dme=int(b["dme"]),
                       auto=int(b["auto"])) for b in val[0]["basis"]]
        self.aw_descriptors: list[list[RSD]] = []
        for l in range(self.lmax_apw + 1):
            self.aw_descriptors.append(
                [RSD(n=l + 1, l=l, enu=b.enu, dme=b.dme, auto=b.auto)
                 for b in default])
        for spec in val[1:]:
            l = int(spec["l"])
            n = int(spec["n"])
            if l < self.lmax_apw:
                self.aw_descriptors[l] = [
                    RSD(n=n, l=l, enu=float(b["enu"]), dme=int(b["dme"]),
                        auto=int(b["auto"])) for b in spec["basis"]]

        # local orbitals
        self.lo_descriptors: list[LocalOrbital] = []
        for lo in data.get("lo", []):
            l = int(lo["l"])
            rsds = [RSD(n=int(b["n"]), l=l, enu=float(b["enu"]),
                        dme=int(b["dme"]), auto=int(b["auto"]))
                    for b in lo["basis"]]
            self.lo_descriptors.append(LocalOrbital(l=l, rsd_set=rsds))

        # free atom density on its own grid
        self.free_atom_r = np.asarray(data["free_atom"]["radial_grid"],
                                      dtype=np.float64)
        self.free_atom_rho = np.asarray(data["free_atom"]["density"],
                                        dtype=np.float64)

        self._build_indices()

    # ------------------------------------------------------------- indices
    def _build_indices(self):
        """indexr (radial functions) and indexb (basis = radial x lm).

        APW radial functions are ordered l-major then order; local
        orbitals are appended in descriptor order with orders continuing
        after the AW orders of the same l (reference
        radial_functions_index).  Basis functions are idxrf-major,
        m-minor (reference basis_functions_index).
        """
        self.indexr = []          # (l, order, idxlo or -1)
        order_count = {}
        for l in range(self.lmax_apw + 1):
            for _ in range(len(self.aw_descriptors[l])):
                o = order_count.get(l, 0)
                self.indexr.append((l, o, -1))
                order_count[l] = o + 1
        self.num_aw_rf = len(self.indexr)
        for ilo, lo in enumerate(self.lo_descriptors):
            o = order_count.get(lo.l, 0)
            self.indexr.append((lo.l, o, ilo))
            order_count[lo.l] = o + 1
        self.max_order_by_l = dict(order_count)

        # reverse map (l, order) -> idxrf
        self._rf_by_lo = {}
        for idx, (l, o, _) in enumerate(self.indexr):
            self._rf_by_lo[(l, o)] = idx

        # basis index: APW part first (all aw radial functions), then lo
        self.indexb = []          # (l, m, lm, order, idxrf)
        for idxrf in range(self.num_aw_rf):
            l, o, _ = self.indexr[idxrf]
            for m in range(-l, l + 1):
                lm = l * l + l + m
                self.indexb.append((l, m, lm, o, idxrf))
        self.mt_aw_basis_size = len(self.indexb)
        for idxrf in range(self.num_aw_rf, len(self.indexr)):
            l, o, _ = self.indexr[idxrf]
            for m in range(-l, l + 1):
                lm = l * l + l + m
                self.indexb.append((l, m, lm, o, idxrf))
        self.mt_basis_size = len(self.indexb)
        self.mt_lo_basis_size = self.mt_basis_size - self.mt_aw_basis_size

    def set_rmt(self, rmt: float):
        """Rebuild the MT radial grid with a new sphere radius (used by
        auto_rmt; reference unit_cell.cpp:850-855 keeps nmtp and rmin)."""
        self.rmt = float(rmt)
        t = (np.arange(self.nmtp) / (self.nmtp - 1)) ** self._grid_p
        self.r = self.rmin * (self.rmt / self.rmin) ** t
        self.r[0] = self.rmin
        self.r[-1] = self.rmt

    def rf_index(self, l: int, order: int) -> int:
        return self._rf_by_lo[(l, order)]

    def aw_order(self, l: int) -> int:
        return len(self.aw_descriptors[l])

    @property
    def num_rf(self) -> int:
        return len(self.indexr)

    @property
    def num_lo(self) -> int:
        return len(self.lo_descriptors)

    def free_atom_density(self, x):
        """Interpolated free-atom density (clamped to the grid tail)."""
        return np.interp(x, self.free_atom_r, self.free_atom_rho)

    @classmethod
    def from_file(cls, label: str, path: str, **kw) -> "FPAtomType":
        with open(path) as f:
            return cls(label, json.load(f), **kw)
