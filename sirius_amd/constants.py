"""Physical and numerical constants (Hartree atomic units throughout)."""

import math

pi = math.pi
twopi = 2 * math.pi
fourpi = 4 * math.pi

ha_to_ev = 27.211386245988
bohr_to_ang = 0.529177210903

# |G|-spline step for radial-integral interpolation tables
# (reference: Radial_integrals_base, src/radial/radial_integrals.hpp:27 —
# uses ppd points per a.u.^-1; we tabulate on a uniform q grid instead).
RI_QPOINTS_PER_AU = 20
