"""Command-line SCF application.

Reference behavior: apps/mini_app/sirius.scf.cpp — run a ground state from
a sirius.json deck, emit an output JSON with the energy breakdown,
convergence history and timings; `--test_against=output_ref.json`
self-checks |ΔE_tot| ≤ 1e-5 Ha (sirius.scf.cpp:309-341).

Usage:
    python -m sirius_amd.cli sirius.json [--device cpu|cuda]
        [--num-iter N] [--test-against output_ref.json] [--output out.json]
        [--forces] [--stress] [--relax | --vc-relax]
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
    ap.add_argument("--forces", action="store_true",
                    help="compute atomic forces after the SCF")
    ap.add_argument("--stress", action="store_true",
                    help="compute the stress tensor after the SCF")
    ap.add_argument("--relax", action="store_true",
                    help="relax atomic positions (fixed cell)")
    ap.add_argument("--vc-relax", action="store_true",
                    help="relax positions and cell")
    ap.add_argument("--save-state", default=None, metavar="FILE.h5|FILE.npz",
                    help="write the converged density/potential state")
    ap.add_argument("--restart", default=None, metavar="FILE.h5|FILE.npz",
                    help="start the SCF from a saved state")
    ap.add_argument("--verbosity", "-v", type=int, default=1)
    args = ap.parse_args(argv)

    from . import Config, SimulationContext, KPointSet, DFTGroundState

    t0 = time.time()
    cfg = Config.from_json(args.input)
    for ov in args.override:
        key, val = ov.split("=", 1)
        cfg.override(key, json.loads(val))
    base = os.path.dirname(os.path.abspath(args.input))
    if args.relax or args.vc_relax:
        from .relax import LatticeRelaxation

        rlx = LatticeRelaxation(cfg, base_dir=base, device=args.device,
                                variable_cell=args.vc_relax)
        rres = rlx.run()
        print(json.dumps(rres["history"][-1], indent=2, default=float))
        path = args.output or f"relax_{int(time.time())}.json"
        with open(path, "w") as f:
            json.dump(rres, f, indent=2, default=float)
        return 0 if rres["converged"] else 1

    if cfg.parameters.electronic_structure_method == "full_potential_lapwlo":
        # FP-LAPW branch (reference: the same sirius.scf entry point
        # dispatches on electronic_structure_method)
        from .lapw.engine import make_lapw_context, FPGroundState

        ctx = make_lapw_context(cfg, base_dir=base, device=args.device)
        ctx.cfg._data["control"]["verbosity"] = 0
        ctx.cfg.control.verbosity = 0
        kset = KPointSet(ctx)
        gs = FPGroundState(kset).initial_state()
        cb = None
        if args.verbosity >= 1:
            def cb(it, etot, rms):
                print(f"iter {it:3d}  Etot {etot:+.10f} Ha  rms {rms:.3e}",
                      flush=True)
        res = gs.find(num_dft_iter=args.num_iter, callback=cb)
        mtot, per = gs.density.total_magnetization() \
            if getattr(gs.density, "nmag", 0) else (0.0, [])
        out = {
            "ground_state": {
                "energy": res["energy"],
                "converged": res["converged"],
                "num_scf_iterations": res["num_scf_iterations"],
                "etot_history": res["etot_history"],
                "rms_history": [float(x) for x in res["rms_history"]],
                "scf_time": res["scf_time"],
                "efermi": res["efermi"],
                "magnetisation": {
                    "total": [0.0, 0.0, mtot],
                    "atoms": [[0.0, 0.0, m] for m in per]},
                "core_leakage": sum(a.core_leakage for a in gs.classes),
            },
            "context": {
                "num_bands": ctx.num_bands,
                "num_kpoints": kset.num_kpoints,
                "fft_grid": list(ctx.fft_fine.dims),
                "num_gvec": ctx.gvec_fine.num_gvec,
                "device": str(ctx.device),
            },
        }
        etot = res["energy"]["total"]
        print(f"total energy: {etot:.10f} Ha  "
              f"({res['num_scf_iterations']} SCF iterations, "
              f"{res['scf_time']:.1f} s)")
        rc = 0
        if args.test_against:
            with open(args.test_against) as f:
                refd = json.load(f)
            ref = refd["ground_state"]["energy"]["total"]
            de = abs(etot - ref)
            ok = de < 1e-5
            print(f"test_against: ref {ref:.10f}  |dE| = {de:.2e}  "
                  f"{'OK' if ok else 'FAIL'}")
            rc = 0 if ok else 1
        if args.output:
            with open(args.output, "w") as f:
                json.dump(out, f, indent=2, default=float)
        return rc

    ctx = SimulationContext(cfg, base_dir=base, device=args.device)
    kset = KPointSet(ctx)
    dft = DFTGroundState(kset).initial_state()
    if args.restart:
        if args.restart.endswith(".h5"):
            from .checkpoint import load_state_h5 as _load
        else:
            from .checkpoint import load_state as _load
        _load(args.restart, dft)
    cb = None
    if args.verbosity >= 1:
        def cb(it, etot, rms):
            print(f"iter {it:3d}  Etot {etot:+.10f} Ha  rms {rms:.3e}",
                  flush=True)
    res = dft.find(num_dft_iter=args.num_iter, callback=cb)
    res["setup_and_scf_time"] = time.time() - t0
    if args.save_state:
        if args.save_state.endswith(".h5"):
            from .checkpoint import save_state_h5 as _save
        else:
            from .checkpoint import save_state as _save
        _save(args.save_state, dft)

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
            "band_gap": res.get("band_gap", 0.0),
        },
        "counters": res.get("counters", {}),
        "context": {
            "num_bands": ctx.num_bands,
            "num_kpoints": kset.num_kpoints,
            "fft_grid": list(ctx.fft_fine.dims),
            "num_gvec": ctx.gvec_fine.num_gvec,
            "device": str(ctx.device),
        },
    }
    if args.forces:
        f = dft.forces()
        out["ground_state"]["forces"] = f["total"].tolist()
        if args.verbosity >= 1:
            print("total forces [Ha/bohr]:")
            for ia, row in enumerate(f["total"]):
                print(f"  atom {ia:3d}: "
                      + "  ".join(f"{x:+.8f}" for x in row))
    if args.stress:
        st = dft.stress()
        out["ground_state"]["stress"] = st["total"].T.tolist()
        if args.verbosity >= 1:
            print("stress tensor [Ha/bohr^3]:")
            for row in st["total"].T:
                print("  " + "  ".join(f"{x:+.8f}" for x in row))
            p3 = -sum(st["total"][i][i] for i in range(3)) / 3.0
            # au2kbar = 2.94210119e5 (reference src/geometry/stress.cpp:557)
            print(f"  pressure: {p3 * 2.94210119e5:+.4f} kbar")
    path = args.output or f"output_{int(time.time())}.json"
    with open(path, "w") as f:
        json.dump(out, f, indent=2)
    etot = res["energy"]["total"]
    print(f"total energy: {etot:.10f} Ha  (converged: {res['converged']}, "
          f"{res['num_scf_iterations']} iterations)")
    if args.verbosity >= 2 or os.environ.get("SIRIUS_AMD_PRINT_PERFORMANCE"):
        # rt_graph-style timer tree (reference: control.print_timers /
        # SIRIUS_PRINT_PERFORMANCE env toggles)
        from .utils.profiler import profiler

        print(profiler.report())

    if args.test_against:
        ref = json.load(open(os.path.join(base, args.test_against)))
        eref = ref["ground_state"]["energy"]["total"]
        de = abs(etot - eref)
        ok = de < 1e-5
        print(f"test_against: |dE| = {de:.3e} Ha -> {'OK' if ok else 'FAIL'}")
        import numpy as np
        if args.forces and "forces" in ref["ground_state"]:
            df = np.abs(np.array(out["ground_state"]["forces"])
                        - np.array(ref["ground_state"]["forces"])).max()
            okf = df < 1e-5
            ok = ok and okf
            print(f"test_against forces: max|dF| = {df:.3e} -> "
                  f"{'OK' if okf else 'FAIL'}")
        if args.stress and "stress" in ref["ground_state"]:
            ds = np.abs(np.array(out["ground_state"]["stress"])
                        - np.array(ref["ground_state"]["stress"])).max()
            oks = ds < 1e-6
            ok = ok and oks
            print(f"test_against stress: max|ds| = {ds:.3e} -> "
                  f"{'OK' if oks else 'FAIL'}")
        return 0 if ok else 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
