"""Atomic ground-state configurations (n, l, k, occupancy).

Standard NIST ground-state electron configurations generated from the
Madelung (aufbau) rule plus the well-known exceptions, split into
relativistic (n, l, k) sub-shells the way the reference stores them
(reference: src/unit_cell/atomic_data.hpp `atomic_conf`): for every
(n, l>0) shell there are two j sub-levels, k = l (j = l-1/2, capacity
2l) and k = l+1 (j = l+1/2, capacity 2l+2).

Only *closed* shells matter for LAPW core states (cores are closed), and
closed shells split exactly: occ(k=l) = 2l, occ(k=l+1) = 2l+2.  Open
(valence) shells are split here by filling k=l first — a convention that
only affects the standalone free-atom solver, never the LAPW SCF (open
shells are valence there).
"""

from __future__ import annotations

SYMBOLS = [
    "H", "He", "Li", "Be", "B", "C", "N", "O", "F", "Ne", "Na", "Mg", "Al",
    "Si", "P", "S", "Cl", "Ar", "K", "Ca", "Sc", "Ti", "V", "Cr", "Mn", "Fe",
    "Co", "Ni", "Cu", "Zn", "Ga", "Ge", "As", "Se", "Br", "Kr", "Rb", "Sr",
    "Y", "Zr", "Nb", "Mo", "Tc", "Ru", "Rh", "Pd", "Ag", "Cd", "In", "Sn",
    "Sb", "Te", "I", "Xe", "Cs", "Ba", "La", "Ce", "Pr", "Nd", "Pm", "Sm",
    "Eu", "Gd", "Tb", "Dy", "Ho", "Er", "Tm", "Yb", "Lu", "Hf", "Ta", "W",
    "Re", "Os", "Ir", "Pt", "Au", "Hg", "Tl", "Pb", "Bi", "Po", "At", "Rn",
    "Fr", "Ra", "Ac", "Th", "Pa", "U", "Np", "Pu", "Am", "Cm", "Bk", "Cf",
    "Es", "Fm", "Md", "No", "Lr", "Rf",
]

# Madelung order: (n, l) by increasing n+l, then n
_MADELUNG = sorted(
    [(n, l) for n in range(1, 9) for l in range(0, min(n, 4))],
    key=lambda nl: (nl[0] + nl[1], nl[0]))

# Well-known exceptions to the aufbau rule: z -> {(n, l): occupancy
# override}; shells not listed keep their aufbau filling minus the moved
# electrons.  Expressed as (from_shell, to_shell, n_electrons) moves.
_EXCEPTIONS = {
    24: [((4, 0), (3, 2), 1)],   # Cr  3d5 4s1
    29: [((4, 0), (3, 2), 1)],   # Cu  3d10 4s1
    41: [((5, 0), (4, 2), 1)],   # Nb  4d4 5s1
    42: [((5, 0), (4, 2), 1)],   # Mo  4d5 5s1
    44: [((5, 0), (4, 2), 1)],   # Ru  4d7 5s1
    45: [((5, 0), (4, 2), 1)],   # Rh  4d8 5s1
    46: [((5, 0), (4, 2), 2)],   # Pd  4d10
    47: [((5, 0), (4, 2), 1)],   # Ag  4d10 5s1
    57: [((4, 3), (5, 2), 1)],   # La  5d1 (4f0)
    58: [((4, 3), (5, 2), 1)],   # Ce  4f1 5d1
    64: [((4, 3), (5, 2), 1)],   # Gd  4f7 5d1
    78: [((6, 0), (5, 2), 1)],   # Pt  5d9 6s1
    79: [((6, 0), (5, 2), 1)],   # Au  5d10 6s1
    89: [((5, 3), (6, 2), 1)],   # Ac  6d1
    90: [((5, 3), (6, 2), 2)],   # Th  6d2
    91: [((5, 3), (6, 2), 1)],   # Pa  5f2 6d1
    92: [((5, 3), (6, 2), 1)],   # U   5f3 6d1
    96: [((5, 3), (6, 2), 1)],   # Cm  5f7 6d1
    103: [((5, 3), (7, 1), 1)],  # Lr  7p1 (simplified)
}


def shell_occupations(z: int) -> dict:
    """{(n, l): occ} for the neutral atom of charge z."""
    occ = {}
    left = z
    for (n, l) in _MADELUNG:
        if left <= 0:
            break
        cap = 2 * (2 * l + 1)
        take = min(cap, left)
        occ[(n, l)] = take
        left -= take
    for (src, dst, ne) in _EXCEPTIONS.get(z, []):
        occ[src] = occ.get(src, 0) - ne
        occ[dst] = occ.get(dst, 0) + ne
        if occ[src] <= 0:
            del occ[src]
    return occ


def atomic_configuration(z: int):
    """[(n, l, k, occupancy)] with relativistic sub-shell splitting.

    Closed shells: occ(k=l) = 2l, occ(k=l+1) = 2l+2 (exact).  Open
    shells: k=l filled first (convention, see module docstring).
    """
    out = []
    for (n, l), ne in sorted(shell_occupations(z).items()):
        if l == 0:
            out.append((n, 0, 1, float(ne)))
            continue
        cap_lo = 2 * l        # j = l - 1/2
        lo = min(cap_lo, ne)
        hi = ne - lo
        if lo > 0:
            out.append((n, l, l, float(lo)))
        if hi > 0:
            out.append((n, l, l + 1, float(hi)))
    return out


def zn_by_symbol(symbol: str) -> int:
    return SYMBOLS.index(symbol) + 1
