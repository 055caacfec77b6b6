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
