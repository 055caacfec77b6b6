"""Synthetic benchmark systems: analytic norm-conserving pseudopotentials and
silicon-like supercells.

There is no network access for real pseudopotential files in the benchmark
environment, so the flagship bench (BASELINE.json configs: "Si N-atom
supercell, NC PP-PW") runs on an analytically generated Si-like NC
pseudopotential: Coulomb-tail local potential −(Z/r)erf(r/r_loc),
Gaussian beta projectors for l=0,1, Gaussian atomic density/orbitals.
Same shapes, sizes, cutoffs and SCF work as a real Si run.
"""

from __future__ import annotations

import math

import numpy as np
from scipy.special import erf

from ..cell import AtomType, BetaProjector, AtomicWf, UnitCell
from ..config import Config


def silicon_like_atom_type(label: str = "Si", zn: int = 4) -> AtomType:
    at = AtomType(label)
    at.symbol = label
    at.zn = zn
    r = np.geomspace(1e-7, 12.0, 1400)
    at.r = r
    rloc = 1.1
    at.vloc_r = -(zn / r) * erf(r / rloc)

    # beta projectors (file convention r*beta)
    for l, sig in enumerate([0.9, 1.0]):
        # file convention stores r*β(r); with β ∝ r^l e^{-r²/2σ²} that is r^{l+1} e^{-r²/2σ²}
        b = r ** (l + 1) * np.exp(-(r**2) / (2 * sig**2))
        b /= math.sqrt(np.trapezoid(b * b, r))  # ∫ (rβ)² dr = 1
        at.beta.append(BetaProjector(l=l, j=None, f_r=b))
    at.d_ion = np.diag([1.2, 0.8])

    # atomic density: Gaussian with zn electrons; 4πr²ρ
    alpha = 0.35
    rho = zn * (alpha / math.pi) ** 1.5 * np.exp(-alpha * r**2)
    at.rho_total_4pir2 = 4 * math.pi * r**2 * rho

    # atomic wavefunctions (LCAO init): s and p
    for l, (sig, occ) in enumerate([(1.6, 2.0), (1.8, 2.0)]):
        chi = r ** (l + 1) * np.exp(-(r**2) / (2 * sig**2))
        chi /= math.sqrt(np.trapezoid(chi * chi, r))
        at.atomic_wfs.append(AtomicWf(n=3, l=l, occ=occ, f_r=chi))

    at.is_norm_conserving = True
    return at


# diamond conventional cell: 8 atoms, fractional coordinates
_DIAMOND8 = np.array([
    [0.00, 0.00, 0.00], [0.50, 0.50, 0.00], [0.50, 0.00, 0.50], [0.00, 0.50, 0.50],
    [0.25, 0.25, 0.25], [0.75, 0.75, 0.25], [0.75, 0.25, 0.75], [0.25, 0.75, 0.75],
])
A0_SI = 10.2631  # bohr


def make_synthetic_cell(natoms: int) -> UnitCell:
    at = silicon_like_atom_type()
    if natoms == 2:
        # primitive fcc cell, 2 atoms
        lat = 0.5 * A0_SI * np.array([[0, 1, 1], [1, 0, 1], [1, 1, 0]], dtype=float)
        pos = [("Si", np.array([0.0, 0.0, 0.0])), ("Si", np.array([0.25, 0.25, 0.25]))]
        return UnitCell(lat, {"Si": at}, pos)
    n = round((natoms / 8) ** (1 / 3))
    if 8 * n**3 != natoms:
        raise ValueError("natoms must be 2 or 8*n^3")
    lat = np.eye(3) * A0_SI * n
    pos = []
    for i in range(n):
        for j in range(n):
            for k in range(n):
                for p in _DIAMOND8:
                    pos.append(("Si", (p + [i, j, k]) / n))
    return UnitCell(lat, {"Si": at}, pos)


def make_synthetic_config(natoms: int = 8, gk_cutoff: float = 5.0,
                          pw_cutoff: float = 14.0, ngridk=(1, 1, 1),
                          num_bands: int = -1, smearing_width: float = 0.01) -> tuple:
    cfg = Config({
        "parameters": {
            "xc_functionals": ["XC_LDA_X", "XC_LDA_C_PZ"],
            "gk_cutoff": gk_cutoff,
            "pw_cutoff": pw_cutoff,
            "ngridk": list(ngridk),
            "smearing_width": smearing_width,
            "num_bands": num_bands,
            "use_symmetry": False,
        },
        "mixer": {"type": "anderson", "beta": 0.7},
    })
    return cfg, None


def make_context(natoms: int = 8, device: str | None = None,
                 lattice_scale: float = 1.0, **kwargs):
    from ..context import SimulationContext

    cfg, _ = make_synthetic_config(natoms=natoms, **kwargs)
    uc = make_synthetic_cell(natoms)
    if lattice_scale != 1.0:
        from ..cell import UnitCell

        uc = UnitCell(uc.lattice * float(lattice_scale), uc.atom_types,
                      [(lab, pos) for lab, pos in uc.atoms])
    return SimulationContext(cfg, unit_cell=uc, device=device)
