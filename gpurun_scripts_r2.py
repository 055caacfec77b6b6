import json
from sirius_amd.models.synthetic import make_synthetic_config, make_synthetic_cell
from sirius_amd.context import SimulationContext
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState
outs = {}
for gamma in (False, True):
    cfg, _ = make_synthetic_config(natoms=8, gk_cutoff=4.0, pw_cutoff=10.0, ngridk=(1,1,1))
    cfg._data["parameters"]["gamma_point"] = gamma
    cfg.parameters.gamma_point = gamma
    ctx = SimulationContext(cfg, unit_cell=make_synthetic_cell(8), device="cuda:0")
    dft = DFTGroundState(KPointSet(ctx)).initial_state()
    hist = []
    r = dft.find(num_dft_iter=25, callback=lambda it,e,rm: hist.append((it, e, rm)))
    outs[str(gamma)] = {"etot": r["energy"]["total"], "hist": [(h[0], round(h[1],8), float(h[2])) for h in hist[-6:]]}
print(json.dumps(outs, indent=1))
