#!/usr/bin/env python3
"""Flagship benchmark: sec/SCF-iteration of the plane-wave DFT engine.

Configs map to BASELINE.json's named configs (synthetic species — no
network for real UPF files — with the same shapes/cutoffs/SCF work):

  --model sto-uspp  (default)  BASELINE config 2: SrTiO3-shaped 5-atom
                    ultrasoft PP-PW cell, 4x4x4 k-mesh (augmentation +
                    Q-operator work in every timed step); 1 GPU headline,
                    k-point parallel at N>1.
  --model fe-paw    BASELINE config 4: Fe-bcc-shaped PAW, collinear spin,
                    12x12x12 k-mesh, k-point parallel over RCCL/xGMI.
  --model si512     BASELINE config 3: Si 512-atom supercell, NC, Γ-only.
  --model si2       BASELINE config 1: Si 2-atom diamond, NC, Γ-only.
  --model si64      round-1 trajectory config (Si 64-atom, 2x2x2 k).

fp64 (complex128) throughout — the reference's working precision.

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

import torch

from sirius_amd.models.synthetic import make_named_context
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState
from sirius_amd.parallel import init_distributed


MODEL_LABEL = {
    "si2": "Si2-NC-LDA-Gamma",
    "si64": "Si64-NC-LDA",
    "si512": "Si512-NC-LDA-Gamma",
    "sto-uspp": "SrTiO3-USPP-4x4x4k",
    "fe-paw": "Fe-bcc-PAW-collinear-12x12x12k",
    "nio-lapw": "NiO4-FP-LAPW-AFM-6x6x6k",
}


def bench_nio_lapw(args, comm):
    """BASELINE config 5 (partial coverage): NiO 4-atom FP-LAPW AFM on the
    real test16 species, 6x6x6 k default.  This engine's LAPW branch runs
    the collinear AFM LSDA problem (the reference raises for LAPW+Hubbard;
    non-collinear LAPW is a round-3 item) and orchestrates on the CPU —
    an honest but unaccelerated number; see NEXT.md."""
    if comm.size > 1:
        raise SystemExit("nio-lapw bench supports 1 rank (k-parallel LAPW "
                         "bench is a round-3 item)")
    from sirius_amd.lapw.engine import make_lapw_context, FPGroundState
    from sirius_amd.config import Config

    deck = json.load(open("verification/test16/sirius.json"))
    deck["parameters"]["ngridk"] = list(args.ngridk) if args.ngridk \
        else [6, 6, 6]
    deck["control"] = dict(deck.get("control", {}), verbosity=0)
    ctx = make_lapw_context(Config(deck), base_dir="verification/test16")
    kset = KPointSet(ctx)
    gs = FPGroundState(kset).initial_state()
    print(f"# model=nio-lapw natoms=4 nk={kset.num_kpoints} "
          f"nfv={ctx.cfg.parameters.num_fv_states} (cpu-orchestrated)",
          flush=True)
    for _ in range(args.warmup):
        gs.scf_iteration()
    t0 = time.time()
    for _ in range(args.steps):
        gs.scf_iteration()
    dt = time.time() - t0
    print(json.dumps({
        "metric": "sec/SCF-iteration",
        "value": dt / args.steps,
        "unit": "s",
        "n_gpus": comm.size,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000.0,
        "higher_is_better": False,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "fp64",
        "data": "test16 deck (staged reference verification data)",
        "config": {
            "model": MODEL_LABEL["nio-lapw"],
            "baseline_config": 5,
            "natoms": 4,
            "num_fv_states": int(ctx.cfg.parameters.num_fv_states),
            "num_mag_dims": 1,
            "ngridk": deck["parameters"]["ngridk"],
            "num_kpoints": kset.num_kpoints,
            "parallelism": "cpu-orchestrated (LAPW GPU path: NEXT.md)",
            "coverage": "collinear AFM LSDA; no Hubbard U (reference "
                        "raises for LAPW+U), no nc-magnetism yet",
        },
    }), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--model", type=str, default="sto-uspp",
                    choices=sorted(MODEL_LABEL))
    ap.add_argument("--gk-cutoff", type=float, default=None)
    ap.add_argument("--pw-cutoff", type=float, default=None)
    ap.add_argument("--ngridk", type=int, nargs=3, default=None)
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

    overrides = {}
    if args.gk_cutoff is not None:
        overrides["gk_cutoff"] = args.gk_cutoff
    if args.pw_cutoff is not None:
        overrides["pw_cutoff"] = args.pw_cutoff
    if args.ngridk is not None:
        overrides["ngridk"] = tuple(args.ngridk)

    if args.model == "nio-lapw":
        return bench_nio_lapw(args, comm)

    ctx = make_named_context(args.model, device=device, **overrides)
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset)

    if comm.rank == 0:
        print(f"# model={args.model} natoms={ctx.unit_cell.num_atoms} "
              f"nbands={ctx.num_bands} nspins={ctx.num_spins} "
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
                "model": MODEL_LABEL[args.model],
                "baseline_config": {
                    "si2": 1, "sto-uspp": 2, "si512": 3, "fe-paw": 4,
                    "si64": None}[args.model],
                "natoms": ctx.unit_cell.num_atoms,
                "num_bands": ctx.num_bands,
                "num_spins": ctx.num_spins,
                "ngridk": list(ctx.cfg.parameters.ngridk),
                "num_kpoints": kset.num_kpoints,
                "gk_cutoff": ctx.gk_cutoff,
                "pw_cutoff": ctx.pw_cutoff,
                "parallelism": f"kp{comm.size}",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
