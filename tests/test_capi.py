"""C API (libsirius_amd.so) smoke test through ctypes.

Reference behavior: src/api/sirius_api.cpp handler model; the test
mirrors apps/tests/test_fortran_api.f90's flow (create context, import
parameters, ground state, energies) using the C ABI from ctypes (no
Fortran compiler in this stack — sirius.f90 is generated but compiled
by consumers).
"""

import ctypes
import json
import os
import subprocess
import sys

import pytest

API_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "sirius_amd", "api")
LIB = os.path.join(API_DIR, "libsirius_amd.so")


def _build():
    if os.path.exists(LIB) and os.path.getmtime(LIB) >= os.path.getmtime(
            os.path.join(API_DIR, "sirius_amd_api.cpp")):
        return
    inc = subprocess.run([sys.executable + "-config", "--includes"],
                         capture_output=True, text=True)
    if inc.returncode != 0:
        inc = subprocess.run(["python3-config", "--includes"],
                             capture_output=True, text=True)
    cmd = ["g++", "-O2", "-shared", "-fPIC", "-o", LIB,
           os.path.join(API_DIR, "sirius_amd_api.cpp")] \
        + inc.stdout.split() + ["-lpython3.10"]
    subprocess.run(cmd, check=True)


def test_c_api_ground_state(tmp_path):
    _build()
    lib = ctypes.CDLL(LIB)
    ec = ctypes.c_int(0)
    t = ctypes.c_bool(False)
    lib.sirius_initialize(ctypes.byref(t), ctypes.byref(ec))
    assert ec.value == 0

    h = ctypes.c_void_p()
    fk = ctypes.c_int()
    fb = ctypes.c_int()
    lib.sirius_create_context(0, ctypes.byref(h), ctypes.byref(fk),
                              ctypes.byref(fb), ctypes.byref(ec))
    assert ec.value == 0 and h.value

    deck = json.load(open("verification/test08/sirius.json"))
    # make atom file paths absolute for the API path
    base = os.path.abspath("verification/test08")
    for k, v in deck["unit_cell"]["atom_files"].items():
        deck["unit_cell"]["atom_files"][k] = os.path.join(base, v)
    lib.sirius_import_parameters(ctypes.byref(h),
                                 json.dumps(deck).encode(),
                                 ctypes.byref(ec))
    assert ec.value == 0

    st = ctypes.c_bool(True)
    lib.sirius_context_initialized(ctypes.byref(h), ctypes.byref(st),
                                   ctypes.byref(ec))
    assert ec.value == 0 and not st.value

    lib.sirius_initialize_context(ctypes.byref(h), ctypes.byref(ec))
    assert ec.value == 0
    lib.sirius_context_initialized(ctypes.byref(h), ctypes.byref(st),
                                   ctypes.byref(ec))
    assert st.value

    grid = (ctypes.c_int * 3)(2, 2, 2)
    shift = (ctypes.c_int * 3)(0, 0, 0)
    use_sym = ctypes.c_bool(True)
    ks = ctypes.c_void_p()
    lib.sirius_create_kset_from_grid(ctypes.byref(h), grid, shift,
                                     ctypes.byref(use_sym), ctypes.byref(ks),
                                     ctypes.byref(ec))
    assert ec.value == 0 and ks.value

    nk = ctypes.c_int()
    lib.sirius_get_num_kpoints(ctypes.byref(ks), ctypes.byref(nk),
                               ctypes.byref(ec))
    assert nk.value >= 1

    gs = ctypes.c_void_p()
    lib.sirius_create_ground_state(ctypes.byref(ks), ctypes.byref(gs),
                                   ctypes.byref(ec))
    assert ec.value == 0

    conv = ctypes.c_bool(False)
    niter = ctypes.c_int(0)
    rho_min = ctypes.c_double(0)
    maxit = ctypes.c_int(3)
    ig = ctypes.c_bool(True)
    sv = ctypes.c_bool(False)
    lib.sirius_find_ground_state(ctypes.byref(gs), None, None, None,
                                 ctypes.byref(ig), ctypes.byref(maxit),
                                 ctypes.byref(sv), ctypes.byref(conv),
                                 ctypes.byref(niter), ctypes.byref(rho_min),
                                 ctypes.byref(ec))
    assert ec.value == 0

    e = ctypes.c_double(0)
    lib.sirius_get_energy(ctypes.byref(gs), b"total", ctypes.byref(e),
                          ctypes.byref(ec))
    assert ec.value == 0
    assert e.value < 0 and e.value == e.value

    ef = ctypes.c_double(0)
    lib.sirius_get_energy(ctypes.byref(gs), b"fermi", ctypes.byref(ef),
                          ctypes.byref(ec))
    assert ec.value == 0

    # band energies of k-point 1 (1-based like the Fortran API)
    nb = 32
    be = (ctypes.c_double * nb)()
    ik = ctypes.c_int(1)
    ispn = ctypes.c_int(0)
    lib.sirius_get_band_energies(ctypes.byref(ks), ctypes.byref(ik),
                                 ctypes.byref(ispn), be, ctypes.byref(ec))
    assert ec.value == 0

    # state save via the API (HDF5)
    p = str(tmp_path / "sirius.h5").encode()
    lib.sirius_save_state(ctypes.byref(gs), p, ctypes.byref(ec))
    assert ec.value == 0 and os.path.exists(p)

    for hh in (gs, ks, h):
        lib.sirius_free_object_handler(ctypes.byref(hh), ctypes.byref(ec))
        assert ec.value == 0 and not hh.value
