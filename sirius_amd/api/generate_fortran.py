#!/usr/bin/env python3
"""Generate the Fortran bindings module (sirius.f90) for libsirius_amd.

Mirrors the role of the reference's src/api/generate_api.py: a single
declarative signature table produces ISO-C-binding interfaces plus
friendly wrapper subroutines (handler types, logical/integer/real
conversion, optional error_code).  Run:

    python sirius_amd/api/generate_fortran.py > sirius_amd/api/sirius.f90
"""

from __future__ import annotations

# (name, [(argname, ftype, intent)], doc)
# ftype: handler | handler_out | bool | int | double | string |
#        int(n) | double(n) | int_out | double_out | bool_out
API = [
    ("sirius_initialize", [("call_mpi_init", "bool", "in")],
     "Initialize the library (embeds the Python engine)."),
    ("sirius_finalize", [("call_mpi_fin", "bool", "in"),
                         ("call_device_reset", "bool", "in"),
                         ("call_fftw_fin", "bool", "in")],
     "Shut down the library."),
    ("sirius_create_context", [("fcomm", "int_val", "in"),
                               ("handler", "handler_out", "out"),
                               ("fcomm_k", "int_out", "out"),
                               ("fcomm_band", "int_out", "out")],
     "Create an empty simulation context."),
    ("sirius_import_parameters", [("handler", "handler", "in"),
                                  ("json_str", "string", "in")],
     "Import a JSON parameter string."),
    ("sirius_add_xc_functional", [("handler", "handler", "in"),
                                  ("name", "string", "in")],
     "Add an exchange-correlation functional."),
    ("sirius_set_lattice_vectors", [("handler", "handler", "in"),
                                    ("a1", "double(3)", "in"),
                                    ("a2", "double(3)", "in"),
                                    ("a3", "double(3)", "in")],
     "Set the lattice vectors (bohr)."),
    ("sirius_add_atom_type", [("handler", "handler", "in"),
                              ("label", "string", "in"),
                              ("fname", "string", "in"),
                              ("zn", "int", "in"),
                              ("symbol", "string", "in"),
                              ("mass", "double", "in"),
                              ("spin_orbit", "bool", "in")],
     "Add an atom type (from file or programmatically)."),
    ("sirius_set_atom_type_radial_grid",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("num_radial_points", "int", "in"),
      ("radial_points", "double(*)", "in")],
     "Set the radial grid of an atom type."),
    ("sirius_add_atom_type_radial_function",
     [("handler", "handler", "in"), ("atom_type", "string", "in"),
      ("label", "string", "in"), ("rf", "double(*)", "in"),
      ("num_points", "int", "in"), ("n", "int", "in"), ("l", "int", "in"),
      ("idxrf1", "int", "in"), ("idxrf2", "int", "in"),
      ("occ", "double", "in")],
     "Push one pseudopotential radial function."),
    ("sirius_add_atom", [("handler", "handler", "in"),
                         ("label", "string", "in"),
                         ("position", "double(3)", "in"),
                         ("vector_field", "double(3)", "in")],
     "Add an atom."),
    ("sirius_set_atom_position", [("handler", "handler", "in"),
                                  ("ia", "int", "in"),
                                  ("position", "double(3)", "in")],
     "Move an atom (1-based index)."),
    ("sirius_initialize_context", [("handler", "handler", "in")],
     "Initialize the simulation context."),
    ("sirius_context_initialized", [("handler", "handler", "in"),
                                    ("status", "bool_out", "out")],
     "Query initialization status."),
    ("sirius_create_kset_from_grid",
     [("handler", "handler", "in"), ("k_grid", "int(3)", "in"),
      ("k_shift", "int(3)", "in"), ("use_symmetry", "bool", "in"),
      ("kset_handler", "handler_out", "out")],
     "Create a k-point set from a Monkhorst-Pack grid."),
    ("sirius_create_ground_state", [("ks_handler", "handler", "in"),
                                    ("gs_handler", "handler_out", "out")],
     "Create a ground-state instance."),
    ("sirius_find_ground_state",
     [("gs_handler", "handler", "in"), ("density_tol", "double", "in"),
      ("energy_tol", "double", "in"), ("iter_solver_tol", "double", "in"),
      ("initial_guess", "bool", "in"), ("max_niter", "int", "in"),
      ("save_state", "bool", "in"), ("converged", "bool_out", "out"),
      ("niter", "int_out", "out"), ("rho_min", "double_out", "out")],
     "Run the SCF loop."),
    ("sirius_get_energy", [("gs_handler", "handler", "in"),
                           ("label", "string", "in"),
                           ("energy", "double_out", "out")],
     "Get an energy component by label."),
    ("sirius_get_forces", [("gs_handler", "handler", "in"),
                           ("label", "string", "in"),
                           ("forces", "double_arr_out", "out")],
     "Get forces [3, num_atoms]."),
    ("sirius_get_stress_tensor", [("gs_handler", "handler", "in"),
                                  ("label", "string", "in"),
                                  ("stress_tensor", "double_arr_out", "out")],
     "Get a stress tensor component [3, 3]."),
    ("sirius_get_num_kpoints", [("ks_handler", "handler", "in"),
                                ("num_kpoints", "int_out", "out")],
     "Number of k-points in the set."),
    ("sirius_get_band_energies", [("ks_handler", "handler", "in"),
                                  ("ik", "int", "in"), ("ispn", "int", "in"),
                                  ("band_energies", "double_arr_out", "out")],
     "Band energies of one k-point."),
    ("sirius_get_band_occupancies",
     [("ks_handler", "handler", "in"), ("ik", "int", "in"),
      ("ispn", "int", "in"), ("band_occupancies", "double_arr_out", "out")],
     "Band occupancies of one k-point."),
    ("sirius_get_kpoint_properties", [("ks_handler", "handler", "in"),
                                      ("ik", "int", "in"),
                                      ("weight", "double_out", "out"),
                                      ("coordinates", "double_arr_out", "out")],
     "Weight and coordinates of one k-point."),
    ("sirius_save_state", [("gs_handler", "handler", "in"),
                           ("file_name", "string", "in")],
     "Save the density/potential state (sirius.h5)."),
    ("sirius_load_state", [("gs_handler", "handler", "in"),
                           ("file_name", "string", "in")],
     "Load a saved state."),
    ("sirius_free_object_handler", [("handler", "handler", "inout")],
     "Release a handler."),
    # -- round-2 widening (introspection, arrays, SCF seams).  The
    # -- string-out helpers (option_get_section_name/info) and the
    # -- no-error-code version getters are C-only.
    ("sirius_is_initialized", [("status", "bool_out", "out")],
     "Check if the library is initialized."),
    ("sirius_get_num_atoms", [("gs_handler", "handler", "in"),
                              ("num_atoms", "int_out", "out")],
     "Number of atoms in the unit cell."),
    ("sirius_get_num_gvec", [("handler", "handler", "in"),
                             ("num_gvec", "int_out", "out")],
     "Size of the fine G-vector sphere."),
    ("sirius_get_num_fft_grid_points",
     [("handler", "handler", "in"), ("num_fft_grid_points", "int_out", "out")],
     "Number of fine FFT grid points."),
    ("sirius_get_fft_index", [("handler", "handler", "in"),
                              ("fft_index", "int(*)_out", "out")],
     "1-based FFT-grid offsets of the fine G sphere."),
    ("sirius_get_num_beta_projectors",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("num_bp", "int_out", "out")],
     "Number of beta projectors of an atom type."),
    ("sirius_get_gvec_arrays",
     [("handler", "handler", "in"), ("gvec", "int(*)_out", "out"),
      ("gvec_cart", "double(*)_out", "out"),
      ("gvec_len", "double(*)_out", "out"),
      ("index_by_gvec", "int(*)_out", "out")],
     "Miller indices / cartesian coords / lengths of the G sphere."),
    ("sirius_get_max_num_gkvec", [("ks_handler", "handler", "in"),
                                  ("max_num_gkvec", "int_out", "out")],
     "Maximum number of G+k vectors over the k-set."),
    ("sirius_get_gkvec_arrays",
     [("ks_handler", "handler", "in"), ("ik", "int", "in"),
      ("num_gkvec", "int_out", "out"), ("gvec_index", "int(*)_out", "out"),
      ("gkvec", "double(*)_out", "out"), ("gkvec_cart", "double(*)_out", "out"),
      ("gkvec_len", "double(*)_out", "out"),
      ("gkvec_tp", "double(*)_out", "out")],
     "G+k vector arrays of one k-point (1-based ik)."),
    ("sirius_set_band_occupancies",
     [("ks_handler", "handler", "in"), ("ik", "int", "in"),
      ("ispn", "int", "in"), ("band_occupancies", "double(*)", "in"),
      ("num_bands", "int", "in")],
     "Set band occupancies of one k-point/spin."),
    ("sirius_generate_initial_density", [("gs_handler", "handler", "in")],
     "Generate the superposition-of-atoms initial density."),
    ("sirius_generate_effective_potential",
     [("gs_handler", "handler", "in")],
     "Generate the effective potential from the current density."),
    ("sirius_generate_density",
     [("gs_handler", "handler", "in"), ("add_core", "bool", "in"),
      ("transform_to_rg", "bool", "in"), ("paw_only", "bool", "in")],
     "Generate the charge density from the current wave functions."),
    ("sirius_initialize_subspace",
     [("gs_handler", "handler", "in"), ("ks_handler", "handler", "in")],
     "Initialize the Davidson trial subspace."),
    ("sirius_find_eigen_states",
     [("gs_handler", "handler", "in"), ("ks_handler", "handler", "in"),
      ("precompute_pw", "bool", "in"), ("precompute_rf", "bool", "in"),
      ("precompute_ri", "bool", "in"), ("iter_solver_tol", "double", "in")],
     "Diagonalize the Hamiltonian for all k-points."),
    ("sirius_find_band_occupancies", [("ks_handler", "handler", "in")],
     "Find the Fermi level and band occupancies."),
    ("sirius_get_periodic_function",
     [("gs_handler", "handler", "in"), ("label", "string", "in"),
      ("f_mt", "double(*)_out", "out"), ("lmmax", "int", "in"),
      ("nrmtmax", "int", "in"), ("num_atoms", "int", "in"),
      ("f_rg", "double(*)_out", "out"), ("size_x", "int", "in"),
      ("size_y", "int", "in"), ("size_z", "int", "in"),
      ("offset_z", "int", "in")],
     "Real-grid values of a named scalar field."),
    ("sirius_set_periodic_function",
     [("gs_handler", "handler", "in"), ("label", "string", "in"),
      ("f_mt", "double(*)", "in"), ("lmmax", "int", "in"),
      ("nrmtmax", "int", "in"), ("num_atoms", "int", "in"),
      ("f_rg", "double(*)", "in"), ("size_x", "int", "in"),
      ("size_y", "int", "in"), ("size_z", "int", "in"),
      ("offset_z", "int", "in")],
     "Set a named scalar field from real-grid values."),
    ("sirius_get_total_magnetization",
     [("gs_handler", "handler", "in"), ("mag", "double_arr_out", "out")],
     "Total magnetization vector (3 components)."),
    ("sirius_set_atom_vector_field",
     [("handler", "handler", "in"), ("ia", "int", "in"),
      ("vector_field", "double(3)", "in")],
     "Set the initial magnetization vector of one atom (1-based ia)."),
    ("sirius_set_num_bands", [("handler", "handler", "in"),
                              ("num_bands", "int", "in")],
     "Set the number of bands."),
    ("sirius_set_mpi_grid_dims",
     [("handler", "handler", "in"), ("ndims", "int", "in"),
      ("dims", "int(*)", "in")],
     "Set the MPI grid dimensions (k-groups x band ranks)."),
    ("sirius_create_context_from_json",
     [("fcomm", "int_val", "in"), ("handler", "handler_out", "out"),
      ("fname", "string", "in")],
     "Create a context directly from a JSON string/file."),
    ("sirius_update_ground_state", [("gs_handler", "handler", "in")],
     "Re-generate the potential from the current density."),
    ("sirius_print_info", [("handler", "handler", "in")],
     "Print a short context summary."),
    ("sirius_print_timers", [("flatten", "bool", "in")],
     "Print timer statistics."),
    ("sirius_option_get_number_of_sections", [("length", "int_out", "out")],
     "Number of config schema sections."),
    ("sirius_option_get_section_length",
     [("section", "string", "in"), ("length", "int_out", "out")],
     "Number of options in a schema section."),
    ("sirius_set_atom_type_dion",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("num_beta", "int", "in"), ("dion", "double(*)", "in")],
     "Set the ionic D matrix of an atom type."),
    ("sirius_set_atom_type_paw",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("core_energy", "double", "in"), ("occupations", "double(*)", "in"),
      ("num_occ", "int", "in")],
     "Set PAW core energy and wave occupations."),
    ("sirius_set_atom_type_configuration",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("n", "int", "in"), ("l", "int", "in"), ("k", "int", "in"),
      ("occupancy", "double", "in"), ("core", "bool", "in")],
     "Add one atomic level to the configuration."),
    ("sirius_add_atom_type_aw_descriptor",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("n", "int", "in"), ("l", "int", "in"), ("enu", "double", "in"),
      ("dme", "int", "in"), ("auto_enu", "bool", "in")],
     "Add an APW radial-solution descriptor (LAPW)."),
    ("sirius_add_atom_type_lo_descriptor",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("ilo", "int", "in"), ("n", "int", "in"), ("l", "int", "in"),
      ("enu", "double", "in"), ("dme", "int", "in"),
      ("auto_enu", "bool", "in")],
     "Add a local-orbital descriptor (LAPW)."),
    ("sirius_set_equivalent_atoms",
     [("handler", "handler", "in"), ("equivalent_atoms", "int(*)", "in")],
     "Set the equivalent-atom map (overrides autodetection)."),
    ("sirius_get_fv_eigen_values",
     [("ks_handler", "handler", "in"), ("ik", "int", "in"),
      ("fv_eval", "double(*)_out", "out"), ("num_fv_states", "int", "in")],
     "First-variational eigenvalues of one k-point."),
    ("sirius_nlcg", [("gs_handler", "handler", "in"),
                     ("ks_handler", "handler", "in")],
     "Direct total-energy minimization (orbital CG)."),
    ("sirius_nlcg_params",
     [("gs_handler", "handler", "in"), ("ks_handler", "handler", "in"),
      ("temp", "double", "in"), ("smearing", "string", "in"),
      ("kappa", "double", "in"), ("tau", "double", "in"),
      ("tol", "double", "in"), ("maxiter", "int", "in"),
      ("restart", "int", "in"), ("processing_unit", "string", "in"),
      ("converged", "bool_out", "out")],
     "Direct minimization with explicit parameters."),
    ("sirius_start_timer", [("name", "string", "in")], "Start a timer."),
    ("sirius_stop_timer", [("name", "string", "in")], "Stop a timer."),
    ("sirius_serialize_timers", [("fname", "string", "in")],
     "Write timer statistics to a JSON file."),
    ("sirius_update_context", [("handler", "handler", "in")],
     "Re-initialize the context from the accumulated config."),
    ("sirius_dump_runtime_setup",
     [("handler", "handler", "in"), ("filename", "string", "in")],
     "Dump the merged runtime configuration to a JSON file."),
    ("sirius_get_kp_params_from_ctx",
     [("handler", "handler", "in"), ("k_grid", "int(3)_out", "out"),
      ("k_shift", "int(3)_out", "out"), ("use_symmetry", "bool_out", "out")],
     "k-grid parameters stored in the context."),
    ("sirius_get_scf_params_from_ctx",
     [("handler", "handler", "in"), ("density_tol", "double_out", "out"),
      ("energy_tol", "double_out", "out"),
      ("iter_solver_tol", "double_out", "out"),
      ("max_niter", "int_out", "out")],
     "SCF convergence parameters stored in the context."),
    ("sirius_fft_transform",
     [("gs_handler", "handler", "in"), ("label", "string", "in"),
      ("direction", "int", "in")],
     "Transform a named field between PW and real grid."),
    ("sirius_generate_coulomb_potential",
     [("gs_handler", "handler", "in"), ("vh_el", "double(*)_out", "out")],
     "Generate the Coulomb potential from the current density."),
    ("sirius_generate_xc_potential", [("gs_handler", "handler", "in")],
     "Generate the XC potential from the current density."),
    ("sirius_get_rg_values",
     [("gs_handler", "handler", "in"), ("label", "string", "in"),
      ("grid_dims", "int(3)", "in"), ("local_box_origin", "int(3)", "in"),
      ("local_box_size", "int(3)", "in"), ("fcomm", "int", "in"),
      ("values", "double(*)_out", "out"),
      ("transform_to_rg", "bool", "in")],
     "Real-grid values of a named field inside a box."),
    ("sirius_set_rg_values",
     [("gs_handler", "handler", "in"), ("label", "string", "in"),
      ("grid_dims", "int(3)", "in"), ("local_box_origin", "int(3)", "in"),
      ("local_box_size", "int(3)", "in"), ("fcomm", "int", "in"),
      ("values", "double(*)", "in"), ("transform_to_pw", "bool", "in")],
     "Set real-grid values of a named field inside a box."),
    ("sirius_set_atom_type_hubbard",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("l", "int", "in"), ("n", "int", "in"), ("occ", "double", "in"),
      ("U", "double", "in"), ("J", "double", "in"),
      ("alpha", "double", "in"), ("beta", "double", "in"),
      ("J0", "double", "in")],
     "Add a Hubbard-corrected orbital to an atom type."),
    ("sirius_add_hubbard_atom_pair",
     [("handler", "handler", "in"), ("atom_pair", "int(*)", "in"),
      ("translation", "int(3)", "in"), ("n", "int(*)", "in"),
      ("l", "int(*)", "in"), ("coupling", "double", "in")],
     "Add an inter-site Hubbard V pair."),
    ("sirius_get_kpoint_inner_comm",
     [("handler", "handler", "in"), ("fcomm", "int_out", "out")],
     "Communicator handle inside a k-group (serial embedding: 0)."),
    ("sirius_get_kpoint_inter_comm",
     [("handler", "handler", "in"), ("fcomm", "int_out", "out")],
     "Communicator handle across k-groups (serial embedding: 0)."),
    ("sirius_get_fft_comm",
     [("handler", "handler", "in"), ("fcomm", "int_out", "out")],
     "FFT communicator handle (serial embedding: 0)."),
    ("sirius_set_energy_fermi",
     [("ks_handler", "handler", "in"), ("energy_fermi", "double", "in")],
     "Set the Fermi energy."),
    ("sirius_check_scf_density", [("gs_handler", "handler", "in")],
     "Regenerate rho from the wave functions and report max |drho(G)|."),
    ("sirius_get_step_function",
     [("handler", "handler", "in"), ("cfunig", "double(*)_out", "out"),
      ("cfunrg", "double(*)_out", "out"),
      ("num_rg_points", "int_out", "out")],
     "LAPW unit-step function (PW re/im interleaved + real grid)."),
    ("sirius_initialize_kset",
     [("ks_handler", "handler", "in"), ("count", "int_out", "out")],
     "Finalize k-set construction (no-op here; returns num_kpoints)."),
    ("sirius_create_hamiltonian",
     [("gs_handler", "handler", "in"), ("H0_handler", "handler_out", "out")],
     "Create a Hamiltonian handler from the current potential."),
    ("sirius_generate_d_operator_matrix",
     [("gs_handler", "handler", "in")],
     "Rebuild the D-operator matrices from the current potential."),
    ("sirius_set_atom_type_radial_grid_inf",
     [("handler", "handler", "in"), ("label", "string", "in"),
      ("num_radial_points", "int", "in"),
      ("radial_points", "double(*)", "in")],
     "Set the extended (free-atom) radial grid of an atom type."),
    ("sirius_get_gkvec",
     [("ks_handler", "handler", "in"), ("ik", "int", "in"),
      ("gvec", "double(*)_out", "out")],
     "Cartesian G+k vectors of one k-point (1-based ik)."),
]

FT = {
    "handler": ("type(sirius_context_handler), intent(in)", "type(c_ptr)"),
    "handler_out": ("type(sirius_context_handler), intent(out)", "type(c_ptr)"),
    "bool": ("logical, intent(in)", "logical(c_bool)"),
    "bool_out": ("logical, intent(out)", "logical(c_bool)"),
    "int": ("integer, intent(in)", "integer(c_int)"),
    "int_val": ("integer, value", "integer(c_int), value"),
    "int_out": ("integer, intent(out)", "integer(c_int)"),
    "double": ("real(8), intent(in)", "real(c_double)"),
    "double_out": ("real(8), intent(out)", "real(c_double)"),
    "string": ("character(*), intent(in)", "character(c_char)"),
    "int(3)": ("integer, intent(in)", "integer(c_int)"),
    "double(3)": ("real(8), intent(in)", "real(c_double)"),
    "double(*)": ("real(8), intent(in)", "real(c_double)"),
    "double_arr_out": ("real(8), intent(out)", "real(c_double)"),
}


def emit():
    out = []
    w = out.append
    w("!> @file sirius.f90")
    w("!! @brief Autogenerated Fortran bindings for libsirius_amd.")
    w("!! Generated by sirius_amd/api/generate_fortran.py — edit the")
    w("!! signature table there, not this file.")
    w("module sirius")
    w("use, intrinsic :: iso_c_binding")
    w("implicit none")
    w("")
    w("type sirius_context_handler")
    w("    type(c_ptr) :: handler_ptr_ = c_null_ptr")
    w("end type")
    w("")
    w("interface")
    for name, args, doc in API:
        cargs = []
        for an, ft, intent in args:
            if ft == "int_val":
                cargs.append(f"{an}")
            else:
                cargs.append(f"{an}")
        cargs.append("error_code")
        w(f"    subroutine {name}_aux({', '.join(cargs)}) &")
        w(f"        bind(C, name=\"{name}\")")
        w("        use, intrinsic :: iso_c_binding")
        for an, ft, intent in args:
            if ft in ("handler", "handler_out"):
                w(f"        type(c_ptr) :: {an}")
            elif ft == "int_val":
                w(f"        integer(c_int), value :: {an}")
            elif ft == "string":
                w(f"        character(c_char), dimension(*) :: {an}")
            elif ft.startswith("int"):
                dim = ", dimension(*)" if "(" in ft else ""
                w(f"        integer(c_int){dim} :: {an}")
            elif ft.startswith("bool"):
                w(f"        logical(c_bool) :: {an}")
            else:
                dim = ", dimension(*)" if ("(" in ft or "arr" in ft) else ""
                w(f"        real(c_double){dim} :: {an}")
        w("        integer(c_int) :: error_code")
        w(f"    end subroutine {name}_aux")
        w("")
    w("end interface")
    w("")
    w("contains")
    w("")
    w("function string_f2c(f_string) result(res)")
    w("    character(*), intent(in) :: f_string")
    w("    character(c_char) :: res(len_trim(f_string) + 1)")
    w("    integer :: i")
    w("    do i = 1, len_trim(f_string)")
    w("        res(i) = f_string(i:i)")
    w("    end do")
    w("    res(len_trim(f_string) + 1) = c_null_char")
    w("end function string_f2c")
    w("")
    for name, args, doc in API:
        fargs = [an for an, _, _ in args] + ["error_code"]
        w(f"!> @brief {doc}")
        w(f"subroutine {name}({', '.join(fargs)})")
        w("    implicit none")
        for an, ft, intent in args:
            if ft in ("handler", "handler_out"):
                io = "out" if ft.endswith("out") else "inout"
                w(f"    type(sirius_context_handler), intent({io}) :: {an}")
            elif ft == "int_val":
                w(f"    integer, value :: {an}")
            elif ft == "string":
                w(f"    character(*), intent(in), target :: {an}")
            elif ft.startswith("int"):
                dim = ", dimension(*)" if "(" in ft else ""
                io = "out" if ft.endswith("out") else "in"
                w(f"    integer, intent({io}){dim}, target :: {an}")
            elif ft.startswith("bool"):
                io = "out" if ft.endswith("out") else "in"
                w(f"    logical, intent({io}), target :: {an}")
            else:
                dim = ", dimension(*)" if ("(" in ft or "arr" in ft) else ""
                io = "out" if ("out" in ft) else "in"
                w(f"    real(8), intent({io}){dim}, target :: {an}")
        w("    integer, intent(out), optional, target :: error_code")
        # locals
        for an, ft, intent in args:
            if ft.startswith("bool"):
                w(f"    logical(c_bool), target :: {an}_c")
        w("    integer(c_int), target :: error_code_c")
        for an, ft, intent in args:
            if ft == "bool":
                w(f"    {an}_c = {an}")
        call_args = []
        for an, ft, intent in args:
            if ft in ("handler", "handler_out"):
                call_args.append(f"{an}%handler_ptr_")
            elif ft == "string":
                call_args.append(f"string_f2c({an})")
            elif ft.startswith("bool"):
                call_args.append(f"{an}_c")
            else:
                call_args.append(an)
        call_args.append("error_code_c")
        w(f"    call {name}_aux({', '.join(call_args)})")
        for an, ft, intent in args:
            if ft == "bool_out":
                w(f"    {an} = {an}_c")
        w("    if (present(error_code)) error_code = error_code_c")
        w(f"end subroutine {name}")
        w("")
    w("end module sirius")
    return "\n".join(out) + "\n"


if __name__ == "__main__":
    import sys

    sys.stdout.write(emit())
