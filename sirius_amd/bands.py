"""Non-self-consistent band structure along a k-path, and an
equation-of-state scan.

Reference behavior: apps/bands/bands.cpp (fix the converged density and
potential, diagonalize H(k) on a user-supplied k-list, emit the band
energies), and the mini-app "eos" task (total energy vs volume).
"""

from __future__ import annotations

import numpy as np


def band_structure(cfg, kpoints, base_dir: str = ".", device=None,
                   num_scf_iter=None) -> dict:
    """SCF on the deck's mesh, then one diagonalization per path point.

    kpoints: [nk, 3] fractional coordinates. Returns band energies
    [nk, num_spin_steps, num_bands] (Ha) and the Fermi energy.
    """
    from .context import SimulationContext
    from .kpoint import KPointSet, KPoint
    from .dft import DFTGroundState, initialize_subspace, diagonalize
    from .hamiltonian import Hamiltonian0

    ctx = SimulationContext(cfg, base_dir=base_dir, device=device)
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset).initial_state()
    res = dft.find(num_dft_iter=num_scf_iter)

    h0 = Hamiltonian0(ctx, dft.potential, dft.density)
    kpoints = np.atleast_2d(np.asarray(kpoints, dtype=np.float64))
    bands = []
    # tighter tolerance for the fixed-potential eigensolve
    tol = ctx.cfg.iterative_solver.energy_tolerance * 1e-2
    for k in kpoints:
        kp = KPoint(ctx, k, 1.0 / len(kpoints))
        initialize_subspace(ctx, kp, h0(kp))
        class _OneK:
            kpoints = [kp]

            def __iter__(self):
                return iter([kp])
        diagonalize(ctx, h0, _OneK(), tol)
        bands.append(kp.eigvals.copy())
    return {
        "kpoints": kpoints.tolist(),
        "bands": np.array(bands).tolist(),       # [nk, nss, nb]
        "efermi": res["efermi"],
        "etot": res["energy"]["total"],
    }


def eos_scan(cfg, scales, base_dir: str = ".", device=None,
             num_scf_iter=None) -> dict:
    """Total energy vs isotropic lattice scaling (mini-app "eos" task)."""
    from .cell import UnitCell
    from .context import SimulationContext
    from .kpoint import KPointSet
    from .dft import DFTGroundState

    c0 = UnitCell.from_config(cfg, base_dir)
    out = []
    for s in scales:
        cell = UnitCell(c0.lattice * float(s), c0.atom_types,
                        [(lab, pos.copy()) for lab, pos in c0.atoms])
        cell.vector_fields = c0.vector_fields.copy()
        ctx = SimulationContext(cfg, unit_cell=cell, base_dir=base_dir,
                                device=device)
        kset = KPointSet(ctx)
        dft = DFTGroundState(kset).initial_state()
        res = dft.find(num_dft_iter=num_scf_iter)
        out.append({"scale": float(s), "volume": ctx.unit_cell.omega,
                    "etot": res["energy"]["total"],
                    "converged": res["converged"]})
    return {"eos": out}
