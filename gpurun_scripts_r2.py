# GPU validation round 3: converged gamma vs plain on device
import torch, time, json
from sirius_amd.models.synthetic import make_synthetic_config, make_synthetic_cell
from sirius_amd.context import SimulationContext
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState

dev = "cuda:0"
out = {}
for tag, gamma in (("plain", False), ("gamma", True)):
    cfg, _ = make_synthetic_config(natoms=64, gk_cutoff=5.0, pw_cutoff=14.0, ngridk=(1,1,1))
    cfg._data["parameters"]["gamma_point"] = gamma; cfg.parameters.gamma_point = gamma
    ctx = SimulationContext(cfg, unit_cell=make_synthetic_cell(64), device=dev)
    dft = DFTGroundState(KPointSet(ctx)).initial_state()
    t0=time.time(); r = dft.find(num_dft_iter=30); dt=time.time()-t0
    out[tag] = {"etot": r["energy"]["total"], "iters": r["num_scf_iterations"], "t": dt}
print(json.dumps(out, indent=1))
