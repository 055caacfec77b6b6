"""FP-LAPW branch: full-potential linearized augmented plane waves.

MI355X-native implementation of the reference's second electronic-
structure method (reference: src/lapw/, src/unit_cell/atom_symmetry_class.*,
src/hamiltonian/hamiltonian_k.cpp set_fv_h_o/apply_fv_h_o,
src/potential/poisson.cpp + xc_mt.cpp, src/density/density.cpp
generate_valence_mt).  Dense FV generalized eigenproblems are solved
with LAPACK on host / rocSOLVER via torch on device — at LAPW matrix
sizes (10^3-10^4) the dense solve is the right tool on MI355X.
"""
