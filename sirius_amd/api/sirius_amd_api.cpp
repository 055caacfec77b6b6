// C API for the sirius_amd engine (libsirius_amd.so).
//
// Reference behavior: src/api/sirius_api.cpp — the same `sirius_*`
// function names and ABI (opaque void* handlers, pointer arguments,
// trailing int* error_code).  This shim embeds CPython and forwards to
// sirius_amd.api_impl (the handler objects ARE Python objects; this
// layer owns references).  Built standalone with g++ against
// libpython3 — no torch headers needed at this level.
//
// Fortran bindings: sirius.f90 is generated from the same signature
// table by sirius_amd/api/generate_fortran.py (the reference generates
// its module with src/api/generate_api.py).

#include <Python.h>

#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

namespace {

bool owns_interpreter = false;
PyObject* impl_module = nullptr;

struct Gil {
    PyGILState_STATE st;
    Gil() { st = PyGILState_Ensure(); }
    ~Gil() { PyGILState_Release(st); }
};

void set_err(int* ec, int v) {
    if (ec) *ec = v;
}

// call impl.<name>(args); returns new reference or nullptr (error printed)
PyObject* call_impl(const char* name, PyObject* args) {
    PyObject* fn = PyObject_GetAttrString(impl_module, name);
    if (!fn) {
        PyErr_Print();
        Py_XDECREF(args);
        return nullptr;
    }
    PyObject* res = PyObject_CallObject(fn, args);
    Py_DECREF(fn);
    Py_XDECREF(args);
    if (!res) {
        PyErr_Print();
        return nullptr;
    }
    return res;
}

PyObject* list_from_doubles(const double* x, int n) {
    PyObject* l = PyList_New(n);
    for (int i = 0; i < n; i++) PyList_SET_ITEM(l, i, PyFloat_FromDouble(x[i]));
    return l;
}

PyObject* list_from_ints(const int* x, int n) {
    PyObject* l = PyList_New(n);
    for (int i = 0; i < n; i++) PyList_SET_ITEM(l, i, PyLong_FromLong(x[i]));
    return l;
}

void doubles_from_seq(PyObject* seq, double* out) {
    PyObject* fast = PySequence_Fast(seq, "expected sequence");
    Py_ssize_t n = PySequence_Fast_GET_SIZE(fast);
    for (Py_ssize_t i = 0; i < n; i++)
        out[i] = PyFloat_AsDouble(PySequence_Fast_GET_ITEM(fast, i));
    Py_DECREF(fast);
}

}  // namespace

extern "C" {

void sirius_initialize(bool const* call_mpi_init, int* error_code) {
    if (!Py_IsInitialized()) {
        Py_InitializeEx(0);
        owns_interpreter = true;
    }
    Gil g;
    if (!impl_module) {
        impl_module = PyImport_ImportModule("sirius_amd.api_impl");
        if (!impl_module) {
            PyErr_Print();
            set_err(error_code, 1);
            return;
        }
    }
    set_err(error_code, 0);
}

void sirius_finalize(bool const* call_mpi_fin, bool const* call_device_reset,
                     bool const* call_fftw_fin, int* error_code) {
    set_err(error_code, 0);
}

void sirius_create_context(int fcomm, void** handler, int* fcomm_k,
                           int* fcomm_band, int* error_code) {
    Gil g;
    PyObject* h = call_impl("create_context", PyTuple_New(0));
    if (!h) { set_err(error_code, 1); return; }
    *handler = h;
    set_err(error_code, 0);
}

void sirius_import_parameters(void* const* handler, char const* str,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("import_parameters",
                            Py_BuildValue("(Os)", (PyObject*)*handler, str));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_xc_functional(void* const* handler, char const* name,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("add_xc_functional",
                            Py_BuildValue("(Os)", (PyObject*)*handler, name));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_lattice_vectors(void* const* handler, double const* a1,
                                double const* a2, double const* a3,
                                int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue("(ONNN)", (PyObject*)*handler,
                                   list_from_doubles(a1, 3),
                                   list_from_doubles(a2, 3),
                                   list_from_doubles(a3, 3));
    PyObject* r = call_impl("set_lattice_vectors", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_atom_type(void* const* handler, char const* label,
                          char const* fname, int const* zn,
                          char const* symbol, double const* mass,
                          bool const* spin_orbit, int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(OssisdO)", (PyObject*)*handler, label, fname ? fname : "",
        zn ? *zn : 0, symbol ? symbol : "", mass ? *mass : 0.0,
        (spin_orbit && *spin_orbit) ? Py_True : Py_False);
    PyObject* r = call_impl("add_atom_type", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_atom_type_radial_grid(void* const* handler, char const* label,
                                      int const* num_radial_points,
                                      double const* radial_points,
                                      int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(OsN)", (PyObject*)*handler, label,
        list_from_doubles(radial_points, *num_radial_points));
    PyObject* r = call_impl("set_atom_type_radial_grid", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_atom_type_radial_function(
    void* const* handler, char const* atom_type, char const* label,
    double const* rf, int const* num_points, int const* n, int const* l,
    int const* idxrf1, int const* idxrf2, double const* occ,
    int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(OssNiiiid)", (PyObject*)*handler, atom_type, label,
        list_from_doubles(rf, *num_points), n ? *n : -1, l ? *l : -1,
        idxrf1 ? *idxrf1 : -1, idxrf2 ? *idxrf2 : -1, occ ? *occ : 0.0);
    PyObject* r = call_impl("add_atom_type_radial_function", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_atom(void* const* handler, char const* label,
                     double const* position, double const* vector_field,
                     int* error_code) {
    Gil g;
    PyObject* vf = vector_field ? list_from_doubles(vector_field, 3)
                                : Py_BuildValue("[ddd]", 0.0, 0.0, 0.0);
    PyObject* args = Py_BuildValue("(OsNN)", (PyObject*)*handler, label,
                                   list_from_doubles(position, 3), vf);
    PyObject* r = call_impl("add_atom", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_atom_position(void* const* handler, int const* ia,
                              double const* position, int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue("(OiN)", (PyObject*)*handler, *ia - 1,
                                   list_from_doubles(position, 3));
    PyObject* r = call_impl("set_atom_position", args);
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_initialize_context(void* const* handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("initialize_context",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_context_initialized(void* const* handler, bool* status,
                                int* error_code) {
    Gil g;
    PyObject* r = call_impl("context_initialized",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    *status = PyObject_IsTrue(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_create_kset_from_grid(void* const* handler, int const* k_grid,
                                  int const* k_shift,
                                  bool const* use_symmetry,
                                  void** kset_handler, int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(ONNO)", (PyObject*)*handler, list_from_ints(k_grid, 3),
        list_from_ints(k_shift, 3), *use_symmetry ? Py_True : Py_False);
    PyObject* r = call_impl("create_kset_from_grid", args);
    if (!r) { set_err(error_code, 1); return; }
    *kset_handler = r;
    set_err(error_code, 0);
}

void sirius_create_kset(void* const* handler, int const* num_kpoints,
                        double* kpoints, double const* kpoint_weights,
                        bool const* init_kset, void** kset_handler,
                        int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(ONNO)", (PyObject*)*handler,
        list_from_doubles(kpoints, 3 * *num_kpoints),
        list_from_doubles(kpoint_weights, *num_kpoints),
        *init_kset ? Py_True : Py_False);
    PyObject* r = call_impl("create_kset", args);
    if (!r) { set_err(error_code, 1); return; }
    *kset_handler = r;
    set_err(error_code, 0);
}

void sirius_create_ground_state(void* const* ks_handler, void** gs_handler,
                                int* error_code) {
    Gil g;
    PyObject* r = call_impl("create_ground_state",
                            Py_BuildValue("(O)", (PyObject*)*ks_handler));
    if (!r) { set_err(error_code, 1); return; }
    *gs_handler = r;
    set_err(error_code, 0);
}

void sirius_find_ground_state(void* const* gs_handler,
                              double const* density_tol,
                              double const* energy_tol,
                              double const* iter_solver_tol,
                              bool const* initial_guess,
                              int const* max_niter, bool const* save_state,
                              bool* converged, int* niter, double* rho_min,
                              int* error_code) {
    Gil g;
    PyObject* args = Py_BuildValue(
        "(OOOOOOO)", (PyObject*)*gs_handler,
        density_tol ? PyFloat_FromDouble(*density_tol) : Py_None,
        energy_tol ? PyFloat_FromDouble(*energy_tol) : Py_None,
        iter_solver_tol ? PyFloat_FromDouble(*iter_solver_tol) : Py_None,
        (!initial_guess || *initial_guess) ? Py_True : Py_False,
        max_niter ? PyLong_FromLong(*max_niter) : Py_None,
        (save_state && *save_state) ? Py_True : Py_False);
    PyObject* r = call_impl("find_ground_state", args);
    if (!r) { set_err(error_code, 1); return; }
    int conv = 0, nit = 0;
    double rmin = 0;
    PyArg_ParseTuple(r, "pid", &conv, &nit, &rmin);
    if (converged) *converged = conv;
    if (niter) *niter = nit;
    if (rho_min) *rho_min = rmin;
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_energy(void* const* gs_handler, char const* label,
                       double* energy, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_energy", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, label));
    if (!r) { set_err(error_code, 1); return; }
    *energy = PyFloat_AsDouble(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_forces(void* const* gs_handler, char const* label,
                       double* forces, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_forces", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, label));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, forces);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_stress_tensor(void* const* gs_handler, char const* label,
                              double* stress_tensor, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_stress_tensor", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, label));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, stress_tensor);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_num_kpoints(void* const* ks_handler, int* num_kpoints,
                            int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_num_kpoints",
                            Py_BuildValue("(O)", (PyObject*)*ks_handler));
    if (!r) { set_err(error_code, 1); return; }
    *num_kpoints = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_band_energies(void* const* ks_handler, int const* ik,
                              int const* ispn, double* band_energies,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_band_energies", Py_BuildValue(
        "(Oii)", (PyObject*)*ks_handler, *ik - 1, *ispn));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, band_energies);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_band_occupancies(void* const* ks_handler, int const* ik,
                                 int const* ispn, double* band_occupancies,
                                 int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_band_occupancies", Py_BuildValue(
        "(Oii)", (PyObject*)*ks_handler, *ik - 1, *ispn));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, band_occupancies);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_kpoint_properties(void* const* ks_handler, int const* ik,
                                  double* weight, double* coordinates,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_kpoint_properties", Py_BuildValue(
        "(Oi)", (PyObject*)*ks_handler, *ik - 1));
    if (!r) { set_err(error_code, 1); return; }
    PyObject* w = PyTuple_GetItem(r, 0);
    PyObject* c = PyTuple_GetItem(r, 1);
    *weight = PyFloat_AsDouble(w);
    doubles_from_seq(c, coordinates);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_save_state(void** gs_handler, const char* file_name,
                       int* error_code) {
    Gil g;
    PyObject* r = call_impl("save_state", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, file_name));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_load_state(void** gs_handler, const char* file_name,
                       int* error_code) {
    Gil g;
    PyObject* r = call_impl("load_state", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, file_name));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_free_object_handler(void** handler, int* error_code) {
    Gil g;
    Py_XDECREF((PyObject*)*handler);
    *handler = nullptr;
    set_err(error_code, 0);
}

}  // extern "C"
