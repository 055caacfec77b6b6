"""Command-line SCF application.

Reference behavior: apps/mini_app/sirius.scf.cpp — run a ground state from
a sirius.json deck, emit an output JSON with the energy breakdown,
convergence history and timings; `--test_against=output_ref.json`
self-checks |ΔE_tot| ≤ 1e-5 Ha (sirius.scf.cpp:309-341).

Usage:
    python -m sirius_amd.cli sirius.json [--device cpu|cuda]
        [--num-iter N] [--test-against output_ref.json] [--output out.json]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time


def main(argv=None):
    ap = argparse.ArgumentParser(prog="sirius_amd.scf")
    ap.add_argument("input", help="sirius.json input deck")
    ap.add_argument("--device", default=None)
    ap.add_argument("--num-iter", type=int, default=None)
    ap.add_argument("--test-against", default=None)
    ap.add_argument("--output", default=None)
    ap.add_argument("--override", action="append", default=[],
                    help="section.key=json_value config overrides")
    args = ap.parse_args(argv)

    from . import Config, SimulationContext, KPointSet, DFTGroundState

    t0 = time.time()
    cfg = Config.from_json(args.input)
    for ov in args.override:
        key, val = ov.split("=", 1)
        cfg.override(key, json.loads(val))
    base = os.path.dirname(os.path.abspath(args.input))
    ctx = SimulationContext(cfg, base_dir=base, device=args.device)
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset).initial_state()
    res = dft.find(num_dft_iter=args.num_iter)
    res["setup_and_scf_time"] = time.time() - t0

    out = {
        "ground_state": {
            "energy": res["energy"],
            "converged": res["converged"],
            "num_scf_iterations": res["num_scf_iterations"],
            "etot_history": res["etot_history"],
            "rms_history": [float(x) for x in res["rms_history"]],
            "scf_time": res["scf_time"],
            "efermi": res["efermi"],
            "magnetisation": {"total": [0.0, 0.0, res["magnetization"]]},
        },
        "context": {
            "num_bands": ctx.num_bands,
            "num_kpoints": kset.num_kpoints,
            "fft_grid": list(ctx.fft_fine.dims),
            "num_gvec": ctx.gvec_fine.num_gvec,
            "device": str(ctx.device),
        },
    }
    path = args.output or f"output_{int(time.time())}.json"
    with open(path, "w") as f:
        json.dump(out, f, indent=2)
    etot = res["energy"]["total"]
    print(f"total energy: {etot:.10f} Ha  (converged: {res['converged']}, "
          f"{res['num_scf_iterations']} iterations)")

    if args.test_against:
        ref = json.load(open(os.path.join(base, args.test_against)))
        eref = ref["ground_state"]["energy"]["total"]
        de = abs(etot - eref)
        ok = de < 1e-5
        print(f"test_against: |dE| = {de:.3e} Ha -> {'OK' if ok else 'FAIL'}")
        return 0 if ok else 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
