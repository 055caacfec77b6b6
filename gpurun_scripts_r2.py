"""Scratch GPU A/B: Davidson expansion-block cap on si512 (not packaged)."""
import json
import time

import torch

from sirius_amd.models.synthetic import make_named_context
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState, diagonalize
from sirius_amd.hamiltonian import Hamiltonian0

outs = {}
for mb in (0, 384, 256):
    torch.manual_seed(0)
    ctx = make_named_context("si512", device="cuda:0")
    ctx.cfg.iterative_solver.set("max_block_size", mb)
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset).initial_state()
    dft.density.mixer_init(ctx.cfg.mixer)

    def step():
        h0 = Hamiltonian0(ctx, dft.potential, dft.density)
        diagonalize(ctx, h0, kset, 1e-4)
        kset.find_band_occupancies()
        dft.density.generate(kset, h0)
        dft.density.mix()
        dft.potential.generate(dft.density)

    for _ in range(2):
        step()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(3):
        step()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 3
    ev = kset.kpoints[0].eigvals[0]
    outs[mb] = {"s_per_iter": round(dt, 3), "ev0": float(ev[0]),
                "ev_last_occ": float(ev[1023])}
    print(mb, outs[mb], flush=True)
    del dft, kset, ctx
    torch.cuda.empty_cache()
print(json.dumps(outs))
