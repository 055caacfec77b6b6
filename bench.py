#!/usr/bin/env python3
"""Flagship benchmark: sec/SCF-iteration of the plane-wave DFT engine.

Config (BASELINE.json): Si N-atom supercell, norm-conserving PP-PW, LDA,
k-mesh distributed over GPUs (k-point parallelism over RCCL/xGMI).
Synthetic Si-like NC pseudopotential and random-init wavefunctions (no
network for real UPF files); fp64 (complex128) throughout — the
reference's working precision.

Contract: one timed step = one full SCF iteration (Davidson diagonalize
+ occupancies + density + mix + potential). Rank 0 prints ONE JSON line.

Run (single GPU):  python bench.py --gpus 1 --steps 4 --warmup 2
Multi-GPU:         python -m torch.distributed.run --nnodes=1
                   --nproc-per-node N --master-addr 127.0.0.1 bench.py ...
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

from sirius_amd.models.synthetic import make_context
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState
from sirius_amd.parallel import init_distributed, get_comm


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--natoms", type=int, default=64)
    ap.add_argument("--gk-cutoff", type=float, default=5.0)
    ap.add_argument("--pw-cutoff", type=float, default=14.0)
    ap.add_argument("--ngridk", type=int, nargs=3, default=[2, 2, 2])
    ap.add_argument("--device", type=str, default=None)
    args = ap.parse_args()

    comm = init_distributed()
    use_gpu = torch.cuda.is_available()
    if args.device:
        device = args.device
    else:
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}" if use_gpu else "cpu"
    if use_gpu and device.startswith("cuda"):
        torch.cuda.set_device(device)

    ctx = make_context(natoms=args.natoms, device=device,
                       gk_cutoff=args.gk_cutoff, pw_cutoff=args.pw_cutoff,
                       ngridk=tuple(args.ngridk))
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset)

    if comm.rank == 0:
        print(f"# natoms={ctx.unit_cell.num_atoms} nbands={ctx.num_bands} "
              f"nk={kset.num_kpoints} (local {len(kset.kpoints)}) "
              f"nGk~{kset.kpoints[0].num_gkvec if kset.kpoints else 0} "
              f"fine_dims={ctx.fft_fine.dims} device={device}", flush=True)

    dft.initial_state()

    # one SCF iteration as the timed step
    def scf_step():
        from sirius_amd.dft import diagonalize
        from sirius_amd.hamiltonian import Hamiltonian0

        h0 = Hamiltonian0(ctx, dft.potential, dft.density)
        diagonalize(ctx, h0, kset, 1e-4)
        kset.find_band_occupancies()
        dft.density.generate(kset, h0)
        dft.density.mix()
        dft.potential.generate(dft.density)

    dft.density.mixer_init(ctx.cfg.mixer)
    for _ in range(args.warmup):
        scf_step()

    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        scf_step()
    if use_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    dt = time.time() - t0

    # MAX over ranks
    dt = -comm.allreduce_scalar(-dt) if comm.active else dt
    ms_per_step = dt / args.steps * 1000.0
    sec_per_iter = dt / args.steps

    if comm.rank == 0:
        print(json.dumps({
            "metric": "sec/SCF-iteration",
            "value": sec_per_iter,
            "unit": "s",
            "n_gpus": comm.size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": f"Si{ctx.unit_cell.num_atoms}-NC-LDA",
                "natoms": ctx.unit_cell.num_atoms,
                "num_bands": ctx.num_bands,
                "ngridk": list(args.ngridk),
                "num_kpoints": kset.num_kpoints,
                "gk_cutoff": args.gk_cutoff,
                "pw_cutoff": args.pw_cutoff,
                "parallelism": f"kp{comm.size}",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
