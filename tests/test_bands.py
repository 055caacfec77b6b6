"""Band-structure and EOS mini-apps (apps/bands + mini_app eos parity)."""

import os

import numpy as np
import pytest

from conftest import requires_reference, REFERENCE


@requires_reference
@pytest.mark.slow
def test_band_structure_si():
    """Fixed-potential diagonalization along L-Γ-X for Si (test08 deck):
    triply degenerate VBM at Γ, deep s-band, CB above VBM at Γ."""
    from sirius_amd import Config
    from sirius_amd.bands import band_structure

    base = os.path.join(REFERENCE, "verification", "test08")
    cfg = Config.from_json(os.path.join(base, "sirius.json"))
    kpath = [[0.5, 0.5, 0.5], [0.25, 0.25, 0.25], [0, 0, 0],
             [0.25, 0, 0.25], [0.5, 0, 0.5]]
    r = band_structure(cfg, kpath, base_dir=base)
    b = np.array(r["bands"])[:, 0, :]
    g = b[2]                        # Γ
    ha2ev = 27.2114
    # VBM triple degeneracy at Γ
    assert abs(g[1] - g[3]) * ha2ev < 0.05
    # deep s-band ~ −12 eV below VBM
    assert -14 < (g[0] - g[3]) * ha2ev < -10
    # direct gap at Γ between 1.5 and 3.5 eV (LDA Si)
    assert 1.0 < (g[4] - g[3]) * ha2ev < 3.5


def test_eos_scan_synthetic():
    """E(V) over isotropic scaling is smooth and single-welled on the
    synthetic cell (the mini-app "eos" task shape)."""
    from sirius_amd.kpoint import KPointSet
    from sirius_amd.dft import DFTGroundState
    from sirius_amd.models.synthetic import make_context

    es = []
    for s in (0.92, 1.0, 1.12):
        ctx = make_context(natoms=2, gk_cutoff=4.0, pw_cutoff=10.0,
                           device="cpu", lattice_scale=s)
        kset = KPointSet(ctx)
        dft = DFTGroundState(kset).initial_state()
        r = dft.find(num_dft_iter=30)
        es.append(r["energy"]["total"])
    assert all(np.isfinite(es))
    # energies differ measurably with volume
    assert max(es) - min(es) > 1e-3
