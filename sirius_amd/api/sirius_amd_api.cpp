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


// ---- round-2 widening: introspection, arrays, SCF seams ----------------

void sirius_is_initialized(bool* status, int* error_code) {
    *status = impl_module != nullptr;
    set_err(error_code, 0);
}

void sirius_get_major_version(int* version) {
    Gil g;
    PyObject* r = call_impl("get_version", Py_BuildValue("(s)", "major"));
    *version = r ? (int)PyLong_AsLong(r) : -1;
    Py_XDECREF(r);
}

void sirius_get_minor_version(int* version) {
    Gil g;
    PyObject* r = call_impl("get_version", Py_BuildValue("(s)", "minor"));
    *version = r ? (int)PyLong_AsLong(r) : -1;
    Py_XDECREF(r);
}

void sirius_get_revision(int* version) {
    Gil g;
    PyObject* r = call_impl("get_version", Py_BuildValue("(s)", "revision"));
    *version = r ? (int)PyLong_AsLong(r) : -1;
    Py_XDECREF(r);
}

void sirius_get_num_atoms(void* const* gs_handler, int* num_atoms,
                          int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_num_atoms",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    if (!r) { set_err(error_code, 1); return; }
    *num_atoms = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_num_gvec(void* const* handler, int* num_gvec,
                         int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_num_gvec",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    *num_gvec = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_num_fft_grid_points(void* const* handler, int* n,
                                    int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_num_fft_grid_points",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    *n = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_fft_index(void* const* handler, int* fft_index,
                          int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_fft_index",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    Py_ssize_t n = PySequence_Length(r);
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject* it = PySequence_GetItem(r, i);
        fft_index[i] = (int)PyLong_AsLong(it);
        Py_DECREF(it);
    }
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_num_beta_projectors(void* const* handler, char const* label,
                                    int* num_bp, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_num_beta_projectors", Py_BuildValue(
        "(Os)", (PyObject*)*handler, label));
    if (!r) { set_err(error_code, 1); return; }
    *num_bp = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_gvec_arrays(void* const* handler, int* gvec,
                            double* gvec_cart, double* gvec_len,
                            int* index_by_gvec, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_gvec_arrays",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    PyObject* mil = PyTuple_GetItem(r, 0);
    PyObject* cart = PyTuple_GetItem(r, 1);
    PyObject* len = PyTuple_GetItem(r, 2);
    if (gvec) {
        Py_ssize_t n = PySequence_Length(mil);
        for (Py_ssize_t i = 0; i < n; i++) {
            PyObject* it = PySequence_GetItem(mil, i);
            gvec[i] = (int)PyLong_AsLong(it);
            Py_DECREF(it);
        }
    }
    if (gvec_cart) doubles_from_seq(cart, gvec_cart);
    if (gvec_len) doubles_from_seq(len, gvec_len);
    (void)index_by_gvec;  // box map not exported (native ordering applies)
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_max_num_gkvec(void* const* ks_handler, int* max_num_gkvec,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_max_num_gkvec",
                            Py_BuildValue("(O)", (PyObject*)*ks_handler));
    if (!r) { set_err(error_code, 1); return; }
    *max_num_gkvec = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_gkvec_arrays(void* const* ks_handler, int* ik, int* num_gkvec,
                             int* gvec_index, double* gkvec,
                             double* gkvec_cart, double* gkvec_len,
                             double* gkvec_tp, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_gkvec_arrays", Py_BuildValue(
        "(Oi)", (PyObject*)*ks_handler, *ik - 1));
    if (!r) { set_err(error_code, 1); return; }
    *num_gkvec = (int)PyLong_AsLong(PyTuple_GetItem(r, 0));
    PyObject* idx = PyTuple_GetItem(r, 1);
    if (gvec_index) {
        Py_ssize_t n = PySequence_Length(idx);
        for (Py_ssize_t i = 0; i < n; i++) {
            PyObject* it = PySequence_GetItem(idx, i);
            gvec_index[i] = (int)PyLong_AsLong(it);
            Py_DECREF(it);
        }
    }
    if (gkvec) doubles_from_seq(PyTuple_GetItem(r, 2), gkvec);
    if (gkvec_cart) doubles_from_seq(PyTuple_GetItem(r, 3), gkvec_cart);
    if (gkvec_len) doubles_from_seq(PyTuple_GetItem(r, 4), gkvec_len);
    if (gkvec_tp) doubles_from_seq(PyTuple_GetItem(r, 5), gkvec_tp);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_wave_functions(void* const* ks_handler, double const* vkl,
                               int const* spin, int const* num_gvec_loc,
                               int const* gvec_loc, double* evec,
                               int const* ld, int const* num_spin_comp,
                               int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_wave_functions", Py_BuildValue(
        "(O[ddd]i)", (PyObject*)*ks_handler, vkl[0], vkl[1], vkl[2],
        spin ? *spin - 1 : 0));
    if (!r) { set_err(error_code, 1); return; }
    int ngk = (int)PyLong_AsLong(PyTuple_GetItem(r, 0));
    int nb = (int)PyLong_AsLong(PyTuple_GetItem(r, 1));
    PyObject* flat = PyTuple_GetItem(r, 2);
    PyObject* fast = PySequence_Fast(flat, "seq");
    int stride = ld ? *ld : ngk;
    for (int ib = 0; ib < nb; ib++)
        for (int igp = 0; igp < ngk; igp++) {
            evec[2 * (ib * stride + igp)] = PyFloat_AsDouble(
                PySequence_Fast_GET_ITEM(fast, 2 * (ib * ngk + igp)));
            evec[2 * (ib * stride + igp) + 1] = PyFloat_AsDouble(
                PySequence_Fast_GET_ITEM(fast, 2 * (ib * ngk + igp) + 1));
        }
    Py_DECREF(fast);
    (void)num_gvec_loc; (void)gvec_loc; (void)num_spin_comp;
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_band_occupancies(void* const* ks_handler, int const* ik,
                                 int const* ispn,
                                 double const* band_occupancies,
                                 int const* num_bands, int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_band_occupancies", Py_BuildValue(
        "(OiiO)", (PyObject*)*ks_handler, *ik - 1, *ispn,
        list_from_doubles(band_occupancies, *num_bands)));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_generate_initial_density(void* const* gs_handler,
                                     int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_initial_density",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_generate_effective_potential(void* const* gs_handler,
                                         int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_effective_potential",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_generate_density(void* const* gs_handler,
                             bool const* add_core,
                             bool const* transform_to_rg,
                             bool const* paw_only, int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_density", Py_BuildValue(
        "(Oii)", (PyObject*)*gs_handler, add_core ? (int)*add_core : 0,
        transform_to_rg ? (int)*transform_to_rg : 0));
    (void)paw_only;
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_initialize_subspace(void* const* gs_handler,
                                void* const* ks_handler, int* error_code) {
    Gil g;
    (void)ks_handler;
    PyObject* r = call_impl("initialize_subspace",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_find_eigen_states(void* const* gs_handler,
                              void* const* ks_handler,
                              bool const* precompute_pw,
                              bool const* precompute_rf,
                              bool const* precompute_ri,
                              double const* iter_solver_tol,
                              int* error_code) {
    Gil g;
    (void)ks_handler; (void)precompute_rf; (void)precompute_ri;
    PyObject* r = call_impl("find_eigen_states", Py_BuildValue(
        "(Oid)", (PyObject*)*gs_handler,
        precompute_pw ? (int)*precompute_pw : 1,
        iter_solver_tol ? *iter_solver_tol : 1e-5));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_find_band_occupancies(void* const* ks_handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("find_band_occupancies",
                            Py_BuildValue("(O)", (PyObject*)*ks_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_periodic_function(void* const* gs_handler, char const* label,
                                  double* f_mt, int const* lmmax,
                                  int const* nrmtmax, int const* num_atoms,
                                  double* f_rg, int const* size_x,
                                  int const* size_y, int const* size_z,
                                  int const* offset_z, int* error_code) {
    Gil g;
    (void)f_mt; (void)lmmax; (void)nrmtmax; (void)num_atoms;
    (void)size_x; (void)size_y; (void)size_z; (void)offset_z;
    PyObject* r = call_impl("get_periodic_function", Py_BuildValue(
        "(Os)", (PyObject*)*gs_handler, label));
    if (!r) { set_err(error_code, 1); return; }
    if (f_rg) doubles_from_seq(r, f_rg);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_periodic_function(void* const* gs_handler, char const* label,
                                  double* f_mt, int const* lmmax,
                                  int const* nrmtmax, int const* num_atoms,
                                  double* f_rg, int const* size_x,
                                  int const* size_y, int const* size_z,
                                  int const* offset_z, int* error_code) {
    Gil g;
    (void)f_mt; (void)lmmax; (void)nrmtmax; (void)num_atoms; (void)offset_z;
    int n = (size_x && size_y && size_z) ? (*size_x) * (*size_y) * (*size_z)
                                         : 0;
    PyObject* r = call_impl("set_periodic_function", Py_BuildValue(
        "(OsO[iii])", (PyObject*)*gs_handler, label,
        list_from_doubles(f_rg, n), size_x ? *size_x : 0,
        size_y ? *size_y : 0, size_z ? *size_z : 0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_total_magnetization(void* const* gs_handler, double* mag,
                                    int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_total_magnetization",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, mag);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_atom_vector_field(void* const* handler, int const* ia,
                                  double const* vector_field,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_vector_field", Py_BuildValue(
        "(OiO)", (PyObject*)*handler, *ia - 1,
        list_from_doubles(vector_field, 3)));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_num_bands(void* const* handler, int* const num_bands,
                          int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_num_bands", Py_BuildValue(
        "(Oi)", (PyObject*)*handler, *num_bands));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_mpi_grid_dims(void* const* handler, int const* ndims,
                              int const* dims, int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_mpi_grid_dims", Py_BuildValue(
        "(OO)", (PyObject*)*handler, list_from_ints(dims, *ndims)));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_create_context_from_json(int fcomm, void** handler,
                                     char const* fname, int* error_code) {
    Gil g;
    (void)fcomm;
    PyObject* h = call_impl("create_context_from_json",
                            Py_BuildValue("(s)", fname));
    if (!h) { set_err(error_code, 1); return; }
    *handler = h;
    set_err(error_code, 0);
}

void sirius_update_ground_state(void** gs_handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("update_ground_state",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_print_info(void* const* handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("print_info",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_print_timers(bool* flatten, int* error_code) {
    Gil g;
    (void)flatten;
    PyObject* r = call_impl("print_timers", PyTuple_New(0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_option_get_number_of_sections(int* length, int* error_code) {
    Gil g;
    PyObject* r = call_impl("option_get_number_of_sections", PyTuple_New(0));
    if (!r) { set_err(error_code, 1); return; }
    *length = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_option_get_section_name(int elem, char* section_name,
                                    int section_name_length,
                                    int* error_code) {
    Gil g;
    PyObject* r = call_impl("option_get_section_name",
                            Py_BuildValue("(i)", elem - 1));
    if (!r) { set_err(error_code, 1); return; }
    const char* s = PyUnicode_AsUTF8(r);
    std::snprintf(section_name, section_name_length, "%s", s ? s : "");
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_option_get_section_length(char const* section, int* length,
                                      int* error_code) {
    Gil g;
    PyObject* r = call_impl("option_get_section_length",
                            Py_BuildValue("(s)", section));
    if (!r) { set_err(error_code, 1); return; }
    *length = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_option_get_info(char const* section, int elem, char* key_name,
                            int key_name_len, int* type, int* length,
                            int* enum_size, char* title, int title_len,
                            char* description, int description_len,
                            int* error_code) {
    Gil g;
    PyObject* r = call_impl("option_get_info", Py_BuildValue(
        "(si)", section, elem - 1));
    if (!r) { set_err(error_code, 1); return; }
    const char* key = PyUnicode_AsUTF8(PyTuple_GetItem(r, 0));
    std::snprintf(key_name, key_name_len, "%s", key ? key : "");
    *type = (int)PyLong_AsLong(PyTuple_GetItem(r, 1));
    *length = (int)PyLong_AsLong(PyTuple_GetItem(r, 2));
    if (enum_size) *enum_size = 0;
    if (title && title_len > 0) title[0] = 0;
    if (description && description_len > 0) description[0] = 0;
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_option_get(char const* section, char const* name,
                       int const* type, void* data_ptr,
                       int const* max_length, int const* enum_idx,
                       int* error_code) {
    Gil g;
    (void)enum_idx;
    PyObject* r = call_impl("option_get", Py_BuildValue("(ss)", section,
                                                        name));
    if (!r) { set_err(error_code, 1); return; }
    int code = (int)PyLong_AsLong(PyTuple_GetItem(r, 0));
    PyObject* v = PyTuple_GetItem(r, 1);
    int want = type ? *type : code;
    int cap = max_length ? *max_length : 1;
    switch (want) {
        case 1:  // int
            *(int*)data_ptr = (int)PyLong_AsLong(v);
            break;
        case 2:  // bool
            *(bool*)data_ptr = PyObject_IsTrue(v);
            break;
        case 4:  // double
            *(double*)data_ptr = PyFloat_AsDouble(v);
            break;
        case 3: {  // string
            const char* s = PyUnicode_AsUTF8(v);
            std::snprintf((char*)data_ptr, cap, "%s", s ? s : "");
            break;
        }
        case 7: {  // int array
            Py_ssize_t n = PySequence_Length(v);
            for (Py_ssize_t i = 0; i < n && i < cap; i++) {
                PyObject* it = PySequence_GetItem(v, i);
                ((int*)data_ptr)[i] = (int)PyLong_AsLong(it);
                Py_DECREF(it);
            }
            break;
        }
        case 9: {  // double array
            Py_ssize_t n = PySequence_Length(v);
            PyObject* fast = PySequence_Fast(v, "seq");
            for (Py_ssize_t i = 0; i < n && i < cap; i++)
                ((double*)data_ptr)[i] = PyFloat_AsDouble(
                    PySequence_Fast_GET_ITEM(fast, i));
            Py_DECREF(fast);
            break;
        }
        default:
            Py_DECREF(r);
            set_err(error_code, 2);
            return;
    }
    Py_DECREF(r);
    set_err(error_code, 0);
}


void sirius_set_atom_type_dion(void* const* handler, char const* label,
                               int const* num_beta, double* dion,
                               int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_type_dion", Py_BuildValue(
        "(OsOi)", (PyObject*)*handler, label,
        list_from_doubles(dion, (*num_beta) * (*num_beta)), *num_beta));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_atom_type_paw(void* const* handler, char const* label,
                              double const* core_energy,
                              double const* occupations, int const* num_occ,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_type_paw", Py_BuildValue(
        "(OsdOi)", (PyObject*)*handler, label, *core_energy,
        list_from_doubles(occupations, *num_occ), *num_occ));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_atom_type_configuration(void* const* handler,
                                        char const* label, int const* n,
                                        int const* l, int const* k,
                                        double const* occupancy,
                                        bool const* core, int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_type_configuration", Py_BuildValue(
        "(Osiiidi)", (PyObject*)*handler, label, *n, *l, *k, *occupancy,
        (int)*core));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_atom_type_aw_descriptor(void* const* handler,
                                        char const* label, int const* n,
                                        int const* l, double const* enu,
                                        int const* dme,
                                        bool const* auto_enu,
                                        int* error_code) {
    Gil g;
    PyObject* r = call_impl("add_atom_type_aw_descriptor", Py_BuildValue(
        "(Osiidii)", (PyObject*)*handler, label, *n, *l, *enu, *dme,
        (int)*auto_enu));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_atom_type_lo_descriptor(void* const* handler,
                                        char const* label, int const* ilo,
                                        int const* n, int const* l,
                                        double const* enu, int const* dme,
                                        bool const* auto_enu,
                                        int* error_code) {
    Gil g;
    PyObject* r = call_impl("add_atom_type_lo_descriptor", Py_BuildValue(
        "(Osiiidii)", (PyObject*)*handler, label, *ilo - 1, *n, *l, *enu,
        *dme, (int)*auto_enu));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_equivalent_atoms(void* const* handler,
                                 int* equivalent_atoms, int* error_code) {
    Gil g;
    // length = current atom count on the handler
    PyObject* na = call_impl("ctx_num_atoms",
                             Py_BuildValue("(O)", (PyObject*)*handler));
    if (!na) { set_err(error_code, 1); return; }
    int n = (int)PyLong_AsLong(na);
    Py_DECREF(na);
    PyObject* r = call_impl("set_equivalent_atoms", Py_BuildValue(
        "(OO)", (PyObject*)*handler, list_from_ints(equivalent_atoms, n)));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_fv_eigen_values(void* const* ks_handler, int const* ik,
                                double* fv_eval, int const* num_fv_states,
                                int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_fv_eigen_values", Py_BuildValue(
        "(Oii)", (PyObject*)*ks_handler, *ik - 1, *num_fv_states));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, fv_eval);
    Py_DECREF(r);
    set_err(error_code, 0);
}


void sirius_nlcg(void* const* gs_handler, void* const* ks_handler,
                 int* error_code) {
    Gil g;
    PyObject* r = call_impl("nlcg", Py_BuildValue(
        "(OO)", (PyObject*)*gs_handler, (PyObject*)*ks_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_nlcg_params(void* const* gs_handler, void* const* ks_handler,
                        double const* temp, char const* smearing,
                        double const* kappa, double const* tau,
                        double const* tol, int const* maxiter,
                        int const* restart, char const* processing_unit,
                        bool* converged, int* error_code) {
    Gil g;
    PyObject* r = call_impl("nlcg", Py_BuildValue(
        "(OOdsdddiis)", (PyObject*)*gs_handler, (PyObject*)*ks_handler,
        temp ? *temp : -1.0, smearing ? smearing : "",
        kappa ? *kappa : 0.3, tau ? *tau : 0.1, tol ? *tol : 1e-9,
        maxiter ? *maxiter : 300, restart ? *restart : 10,
        processing_unit ? processing_unit : ""));
    if (!r) { set_err(error_code, 1); return; }
    if (converged) *converged = PyObject_IsTrue(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}


void sirius_start_timer(char const* name, int* error_code) {
    Gil g;
    PyObject* r = call_impl("start_timer", Py_BuildValue("(s)", name));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_stop_timer(char const* name, int* error_code) {
    Gil g;
    PyObject* r = call_impl("stop_timer", Py_BuildValue("(s)", name));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_serialize_timers(char const* fname, int* error_code) {
    Gil g;
    PyObject* r = call_impl("serialize_timers", Py_BuildValue("(s)", fname));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_parameters(
    void* const* handler, int const* lmax_apw, int const* lmax_rho,
    int const* lmax_pot, int const* num_fv_states, int const* num_bands,
    int const* num_mag_dims, double const* pw_cutoff,
    double const* gk_cutoff, int const* fft_grid_size, int const* auto_rmt,
    bool const* gamma_point, bool const* use_symmetry,
    bool const* so_correction, char const* valence_rel,
    char const* core_rel, double const* iter_solver_tol_empty,
    char const* iter_solver_type, int const* verbosity,
    bool const* hubbard_correction, int const* hubbard_correction_kind,
    bool const* hubbard_full_orthogonalization,
    bool const* hubbard_constrained_calculation,
    char const* hubbard_orbitals, int const* sht_coverage,
    double const* min_occupancy, char const* smearing,
    double const* smearing_width, double const* spglib_tol,
    char const* electronic_structure_method, int* error_code) {
    Gil g;
    PyObject* d = PyDict_New();
    auto seti = [&](const char* k, int const* v) {
        if (v) PyDict_SetItemString(d, k, PyLong_FromLong(*v));
    };
    auto setd = [&](const char* k, double const* v) {
        if (v) PyDict_SetItemString(d, k, PyFloat_FromDouble(*v));
    };
    auto setb = [&](const char* k, bool const* v) {
        if (v) PyDict_SetItemString(d, k, PyBool_FromLong(*v));
    };
    auto sets = [&](const char* k, char const* v) {
        if (v && v[0]) PyDict_SetItemString(d, k, PyUnicode_FromString(v));
    };
    seti("lmax_apw", lmax_apw);
    seti("lmax_rho", lmax_rho);
    seti("lmax_pot", lmax_pot);
    seti("num_fv_states", num_fv_states);
    seti("num_bands", num_bands);
    seti("num_mag_dims", num_mag_dims);
    setd("pw_cutoff", pw_cutoff);
    setd("gk_cutoff", gk_cutoff);
    if (fft_grid_size)
        PyDict_SetItemString(d, "fft_grid_size",
                             list_from_ints(fft_grid_size, 3));
    seti("auto_rmt", auto_rmt);
    setb("gamma_point", gamma_point);
    setb("use_symmetry", use_symmetry);
    setb("so_correction", so_correction);
    sets("valence_rel", valence_rel);
    sets("core_rel", core_rel);
    setd("iter_solver_tol_empty", iter_solver_tol_empty);
    sets("iter_solver_type", iter_solver_type);
    seti("verbosity", verbosity);
    setb("hubbard_correction", hubbard_correction);
    seti("hubbard_correction_kind", hubbard_correction_kind);
    setb("hubbard_full_orthogonalization", hubbard_full_orthogonalization);
    setb("hubbard_constrained_calculation", hubbard_constrained_calculation);
    sets("hubbard_orbitals", hubbard_orbitals);
    seti("sht_coverage", sht_coverage);
    setd("min_occupancy", min_occupancy);
    sets("smearing", smearing);
    setd("smearing_width", smearing_width);
    setd("spglib_tol", spglib_tol);
    sets("electronic_structure_method", electronic_structure_method);
    PyObject* r = call_impl("set_parameters", Py_BuildValue(
        "(ON)", (PyObject*)*handler, d));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_update_context(void* const* handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("update_context",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_option_set(void* const* handler, char const* section,
                       char const* name, int const* type,
                       void const* data_ptr, int const* max_length,
                       bool const* append, int* error_code) {
    Gil g;
    PyObject* v = nullptr;
    int ty = type ? *type : 0;
    int n = max_length ? *max_length : 1;
    switch (ty) {
        case 1: v = PyLong_FromLong(*(int const*)data_ptr); break;
        case 2: v = PyBool_FromLong(*(bool const*)data_ptr); break;
        case 3: v = PyUnicode_FromString((char const*)data_ptr); break;
        case 4: v = PyFloat_FromDouble(*(double const*)data_ptr); break;
        case 7: v = list_from_ints((int const*)data_ptr, n); break;
        case 9: v = list_from_doubles((double const*)data_ptr, n); break;
        default: set_err(error_code, 2); return;
    }
    PyObject* r = call_impl("option_set", Py_BuildValue(
        "(OssNi)", (PyObject*)*handler, section, name, v,
        append ? (int)*append : 0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_dump_runtime_setup(void* const* handler, char* filename,
                               int* error_code) {
    Gil g;
    PyObject* r = call_impl("dump_runtime_setup", Py_BuildValue(
        "(Os)", (PyObject*)*handler, filename));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_kp_params_from_ctx(void* const* handler, int* k_grid,
                                   int* k_shift, bool* use_symmetry,
                                   int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_kp_params_from_ctx",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    PyObject* kg = PyTuple_GetItem(r, 0);
    PyObject* ks = PyTuple_GetItem(r, 1);
    for (int i = 0; i < 3; i++) {
        PyObject* a = PySequence_GetItem(kg, i);
        PyObject* b = PySequence_GetItem(ks, i);
        k_grid[i] = (int)PyLong_AsLong(a);
        k_shift[i] = (int)PyLong_AsLong(b);
        Py_DECREF(a);
        Py_DECREF(b);
    }
    *use_symmetry = PyObject_IsTrue(PyTuple_GetItem(r, 2));
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_scf_params_from_ctx(void* const* handler,
                                    double* density_tol, double* energy_tol,
                                    double* iter_solver_tol, int* max_niter,
                                    int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_scf_params_from_ctx",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    *density_tol = PyFloat_AsDouble(PyTuple_GetItem(r, 0));
    *energy_tol = PyFloat_AsDouble(PyTuple_GetItem(r, 1));
    *iter_solver_tol = PyFloat_AsDouble(PyTuple_GetItem(r, 2));
    *max_niter = (int)PyLong_AsLong(PyTuple_GetItem(r, 3));
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_pw_coeffs(void* const* gs_handler, char const* label,
                          void* pw_coeffs, int const* ngv, int* gvl,
                          int const* comm, int* error_code) {
    Gil g;
    (void)comm;
    int n = ngv ? *ngv : 0;
    PyObject* r = call_impl("get_pw_coeffs", Py_BuildValue(
        "(OsN)", (PyObject*)*gs_handler, label, list_from_ints(gvl, 3 * n)));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, (double*)pw_coeffs);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_pw_coeffs(void* const* gs_handler, char const* label,
                          void const* pw_coeffs, bool const* transform_to_rg,
                          int const* ngv, int* gvl, int const* comm,
                          int* error_code) {
    Gil g;
    (void)comm;
    int n = ngv ? *ngv : 0;
    PyObject* r = call_impl("set_pw_coeffs", Py_BuildValue(
        "(OsNNi)", (PyObject*)*gs_handler, label,
        list_from_doubles((double const*)pw_coeffs, 2 * n),
        list_from_ints(gvl, 3 * n),
        transform_to_rg ? (int)*transform_to_rg : 0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_fft_transform(void* const* gs_handler, char const* label,
                          int* direction, int* error_code) {
    Gil g;
    PyObject* r = call_impl("fft_transform", Py_BuildValue(
        "(Osi)", (PyObject*)*gs_handler, label, *direction));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_rg_values(void* const* gs_handler, char const* label,
                          int const* grid_dims, int const* local_box_origin,
                          int const* local_box_size, int const* fcomm,
                          double* values, bool const* transform_to_rg,
                          int* error_code) {
    Gil g;
    (void)grid_dims; (void)fcomm; (void)transform_to_rg;
    PyObject* r = call_impl("get_rg_values", Py_BuildValue(
        "(OsNN)", (PyObject*)*gs_handler, label,
        list_from_ints(local_box_origin, 3),
        list_from_ints(local_box_size, 3)));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, values);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_rg_values(void* const* gs_handler, char const* label,
                          int const* grid_dims, int const* local_box_origin,
                          int const* local_box_size, int const* fcomm,
                          double const* values, bool const* transform_to_pw,
                          int* error_code) {
    Gil g;
    (void)grid_dims; (void)fcomm;
    int n = local_box_size[0] * local_box_size[1] * local_box_size[2];
    PyObject* r = call_impl("set_rg_values", Py_BuildValue(
        "(OsNNNi)", (PyObject*)*gs_handler, label,
        list_from_ints(local_box_origin, 3),
        list_from_ints(local_box_size, 3), list_from_doubles(values, n),
        transform_to_pw ? (int)*transform_to_pw : 0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_generate_coulomb_potential(void* const* gs_handler,
                                       double* vh_el, int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_coulomb_potential",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    if (!r) { set_err(error_code, 1); return; }
    if (vh_el) doubles_from_seq(r, vh_el);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_generate_xc_potential(void* const* gs_handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_xc_potential",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}


void sirius_set_atom_type_hubbard(void* const* handler, char const* label,
                                  int const* l, int const* n,
                                  double const* occ, double const* U,
                                  double const* J, double const* alpha,
                                  double const* beta, double const* J0,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_type_hubbard", Py_BuildValue(
        "(Osiidddddd)", (PyObject*)*handler, label, *l, *n,
        occ ? *occ : 0.0, U ? *U : 0.0, J ? *J : 0.0,
        alpha ? *alpha : 0.0, beta ? *beta : 0.0, J0 ? *J0 : 0.0));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_hubbard_atom_pair(void* const* handler, int* const atom_pair,
                                  int* const translation, int* const n,
                                  int* const l, const double* const coupling,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("add_hubbard_atom_pair", Py_BuildValue(
        "(ONNNNd)", (PyObject*)*handler, list_from_ints(atom_pair, 2),
        list_from_ints(translation, 3), list_from_ints(n, 2),
        list_from_ints(l, 2), *coupling));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_add_hubbard_atom_constraint(void* const* handler,
                                        int* const atom_id, int* const n,
                                        int* const l, int* const lmax_at,
                                        const double* const occ,
                                        int* const orbital_order,
                                        int* const error_code) {
    Gil g;
    int mm = 2 * (*l) + 1;
    // the occupancy block is (nsp, mm, mm); nsp inferred Python-side —
    // pass the larger collinear size, extra values are ignored there
    PyObject* occl = list_from_doubles(occ, 2 * mm * mm);
    PyObject* ord = orbital_order ? list_from_ints(orbital_order, mm)
                                  : (Py_INCREF(Py_None), Py_None);
    PyObject* r = call_impl("add_hubbard_atom_constraint", Py_BuildValue(
        "(OiiiiNN)", (PyObject*)*handler, *atom_id, *n, *l,
        lmax_at ? *lmax_at : mm, occl, ord));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_kpoint_inner_comm(void* const* handler, int* fcomm,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_comm_handle", Py_BuildValue(
        "(Os)", (PyObject*)*handler, "inner"));
    if (!r) { set_err(error_code, 1); return; }
    *fcomm = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_kpoint_inter_comm(void* const* handler, int* fcomm,
                                  int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_comm_handle", Py_BuildValue(
        "(Os)", (PyObject*)*handler, "inter"));
    if (!r) { set_err(error_code, 1); return; }
    *fcomm = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_fft_comm(void* const* handler, int* fcomm, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_comm_handle", Py_BuildValue(
        "(Os)", (PyObject*)*handler, "fft"));
    if (!r) { set_err(error_code, 1); return; }
    *fcomm = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_energy_fermi(void* const* ks_handler, double* energy_fermi,
                             int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_energy_fermi", Py_BuildValue(
        "(Od)", (PyObject*)*ks_handler, *energy_fermi));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_check_scf_density(void* const* gs_handler, int* error_code) {
    Gil g;
    PyObject* r = call_impl("check_scf_density",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_step_function(void* const* handler, void* cfunig,
                              double* cfunrg, int* num_rg_points,
                              int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_step_function",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    if (cfunig) doubles_from_seq(PyTuple_GetItem(r, 0), (double*)cfunig);
    if (cfunrg) doubles_from_seq(PyTuple_GetItem(r, 1), cfunrg);
    if (num_rg_points)
        *num_rg_points = (int)PyLong_AsLong(PyTuple_GetItem(r, 2));
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_fv_eigen_vectors(void* const* ks_handler, int const* ik,
                                 void* fv_evec, int const* ld,
                                 int const* num_fv_states,
                                 int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_fv_eigen_vectors", Py_BuildValue(
        "(Oii)", (PyObject*)*ks_handler, *ik - 1, *num_fv_states));
    if (!r) { set_err(error_code, 1); return; }
    int nbasis = (int)PyLong_AsLong(PyTuple_GetItem(r, 0));
    PyObject* flat = PyTuple_GetItem(r, 1);
    PyObject* fast = PySequence_Fast(flat, "seq");
    double* out = (double*)fv_evec;
    int stride = ld ? *ld : nbasis;
    for (int j = 0; j < *num_fv_states; j++)
        for (int i = 0; i < nbasis; i++) {
            out[2 * (j * stride + i)] = PyFloat_AsDouble(
                PySequence_Fast_GET_ITEM(fast, 2 * (j * nbasis + i)));
            out[2 * (j * stride + i) + 1] = PyFloat_AsDouble(
                PySequence_Fast_GET_ITEM(fast, 2 * (j * nbasis + i) + 1));
        }
    Py_DECREF(fast);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_get_psi(void* const* ks_handler, int* ik, int* ispin,
                    void* psi, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_psi", Py_BuildValue(
        "(Oii)", (PyObject*)*ks_handler, *ik - 1, *ispin - 1));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(PyTuple_GetItem(r, 2), (double*)psi);
    Py_DECREF(r);
    set_err(error_code, 0);
}


void sirius_get_parameters(
    void* const* handler, int* lmax_apw, int* lmax_rho, int* lmax_pot,
    int* num_fv_states, int* num_bands, int* num_spins, int* num_mag_dims,
    double* pw_cutoff, double* gk_cutoff, int* fft_grid_size, int* auto_rmt,
    bool* gamma_point, bool* use_symmetry, bool* so_correction,
    double* iter_solver_tol, double* iter_solver_tol_empty, int* verbosity,
    bool* hubbard_correction, double* evp_work_count,
    int* num_loc_op_applied, int* num_sym_op,
    char* electronic_structure_method, int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_parameters",
                            Py_BuildValue("(O)", (PyObject*)*handler));
    if (!r) { set_err(error_code, 1); return; }
    auto gi = [&](const char* k, int* out) {
        if (out) {
            PyObject* v = PyDict_GetItemString(r, k);
            if (v) *out = (int)PyLong_AsLong(v);
        }
    };
    auto gd = [&](const char* k, double* out) {
        if (out) {
            PyObject* v = PyDict_GetItemString(r, k);
            if (v) *out = PyFloat_AsDouble(v);
        }
    };
    auto gb = [&](const char* k, bool* out) {
        if (out) {
            PyObject* v = PyDict_GetItemString(r, k);
            if (v) *out = PyObject_IsTrue(v);
        }
    };
    gi("lmax_apw", lmax_apw);
    gi("lmax_rho", lmax_rho);
    gi("lmax_pot", lmax_pot);
    gi("num_fv_states", num_fv_states);
    gi("num_bands", num_bands);
    gi("num_spins", num_spins);
    gi("num_mag_dims", num_mag_dims);
    gd("pw_cutoff", pw_cutoff);
    gd("gk_cutoff", gk_cutoff);
    if (fft_grid_size) {
        PyObject* v = PyDict_GetItemString(r, "fft_grid_size");
        for (int i = 0; v && i < 3; i++) {
            PyObject* it = PySequence_GetItem(v, i);
            fft_grid_size[i] = (int)PyLong_AsLong(it);
            Py_DECREF(it);
        }
    }
    gi("auto_rmt", auto_rmt);
    gb("gamma_point", gamma_point);
    gb("use_symmetry", use_symmetry);
    gb("so_correction", so_correction);
    gd("iter_solver_tol", iter_solver_tol);
    gd("iter_solver_tol_empty", iter_solver_tol_empty);
    gi("verbosity", verbosity);
    gb("hubbard_correction", hubbard_correction);
    gd("evp_work_count", evp_work_count);
    gi("num_loc_op_applied", num_loc_op_applied);
    gi("num_sym_op", num_sym_op);
    if (electronic_structure_method) {
        PyObject* v = PyDict_GetItemString(r,
                                           "electronic_structure_method");
        const char* sv = v ? PyUnicode_AsUTF8(v) : "";
        std::strcpy(electronic_structure_method, sv ? sv : "");
    }
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_initialize_kset(void* const* ks_handler, int* count,
                            int* error_code) {
    Gil g;
    PyObject* r = call_impl("initialize_kset",
                            Py_BuildValue("(O)", (PyObject*)*ks_handler));
    if (!r) { set_err(error_code, 1); return; }
    if (count) *count = (int)PyLong_AsLong(r);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_create_hamiltonian(void* const* gs_handler, void** H0_handler,
                               int* error_code) {
    Gil g;
    PyObject* r = call_impl("create_hamiltonian",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    if (!r) { set_err(error_code, 1); return; }
    *H0_handler = r;
    set_err(error_code, 0);
}

void sirius_diagonalize_hamiltonian(
    void* const* handler, void* const* gs_handler, void* const* H0_handler,
    double* const iter_solver_tol, int* const max_steps,
    int* converge_by_energy, bool* const exact_diagonalization,
    bool* converged, int* niter, int* error_code) {
    Gil g;
    (void)handler; (void)converge_by_energy;
    PyObject* r = call_impl("diagonalize_hamiltonian", Py_BuildValue(
        "(OOdii)", (PyObject*)*gs_handler, (PyObject*)*H0_handler,
        iter_solver_tol ? *iter_solver_tol : 1e-5,
        max_steps ? *max_steps : 20,
        exact_diagonalization ? (int)*exact_diagonalization : 0));
    if (!r) { set_err(error_code, 1); return; }
    if (converged) *converged = PyObject_IsTrue(PyTuple_GetItem(r, 0));
    if (niter) *niter = (int)PyLong_AsLong(PyTuple_GetItem(r, 1));
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_generate_d_operator_matrix(void* const* gs_handler,
                                       int* error_code) {
    Gil g;
    PyObject* r = call_impl("generate_d_operator_matrix",
                            Py_BuildValue("(O)", (PyObject*)*gs_handler));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_atom_type_radial_grid_inf(void* const* handler,
                                          char const* label,
                                          int const* num_radial_points,
                                          double const* radial_points,
                                          int* error_code) {
    Gil g;
    PyObject* r = call_impl("set_atom_type_radial_grid_inf", Py_BuildValue(
        "(OsN)", (PyObject*)*handler, label,
        list_from_doubles(radial_points, *num_radial_points)));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_gkvec(void* const* ks_handler, int* ik, double* gvec,
                      int* error_code) {
    Gil g;
    PyObject* r = call_impl("get_gkvec", Py_BuildValue(
        "(Oi)", (PyObject*)*ks_handler, *ik - 1));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(r, gvec);
    Py_DECREF(r);
    set_err(error_code, 0);
}

void sirius_set_local_occupation_matrix(void** handler, int const* ia,
                                        int const* n, int const* l,
                                        int const* spin, void* occ_mtrx,
                                        int const* ld, int* error_code) {
    Gil g;
    int mm = 2 * (*l) + 1;
    PyObject* r = call_impl("set_local_occupation_matrix", Py_BuildValue(
        "(OiiiiNi)", (PyObject*)*handler, *ia - 1, *n, *l, *spin - 1,
        list_from_doubles((double*)occ_mtrx, 2 * (*ld) * mm), *ld));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_set_nonlocal_occupation_matrix(void** handler,
                                           int const* atom_pair,
                                           int const* n, int const* l,
                                           int const* spin, int const* T,
                                           void* occ_mtrx, int const* ld1,
                                           int const* ld2,
                                           int* error_code) {
    Gil g;
    // 1-based atom indices from Fortran
    int ap[2] = {atom_pair[0] - 1, atom_pair[1] - 1};
    PyObject* r = call_impl("set_nonlocal_occupation_matrix", Py_BuildValue(
        "(ONNNiNNii)", (PyObject*)*handler, list_from_ints(ap, 2),
        list_from_ints(n, 2), list_from_ints(l, 2), *spin - 1,
        list_from_ints(T, 3),
        list_from_doubles((double*)occ_mtrx, 2 * (*ld1) * (*ld2)),
        *ld1, *ld2));
    set_err(error_code, r ? 0 : 1);
    Py_XDECREF(r);
}

void sirius_get_sv_eigen_vectors(void* const* ks_handler, int const* ik,
                                 void* sv_evec, int const* num_bands,
                                 int* error_code) {
    Gil g;
    (void)num_bands;
    PyObject* r = call_impl("get_sv_eigen_vectors", Py_BuildValue(
        "(Oi)", (PyObject*)*ks_handler, *ik - 1));
    if (!r) { set_err(error_code, 1); return; }
    doubles_from_seq(PyTuple_GetItem(r, 2), (double*)sv_evec);
    Py_DECREF(r);
    set_err(error_code, 0);
}

}  // extern "C"






