# GPU validation: gamma path, fp32, beta_phase kernel on device
import torch, time, json
from sirius_amd.models.synthetic import make_named_context, make_synthetic_config, make_synthetic_cell
from sirius_amd.context import SimulationContext
from sirius_amd.kpoint import KPointSet
from sirius_amd.dft import DFTGroundState

dev = "cuda:0"
out = {}
# 1) si64 gamma vs plain on GPU
cfg, _ = make_synthetic_config(natoms=64, gk_cutoff=5.0, pw_cutoff=14.0, ngridk=(1,1,1))
ctx = SimulationContext(cfg, unit_cell=make_synthetic_cell(64), device=dev)
dft = DFTGroundState(KPointSet(ctx)).initial_state()
t0=time.time(); r0 = dft.find(num_dft_iter=6); t_plain = time.time()-t0
cfg2, _ = make_synthetic_config(natoms=64, gk_cutoff=5.0, pw_cutoff=14.0, ngridk=(1,1,1))
cfg2._data["parameters"]["gamma_point"] = True; cfg2.parameters.gamma_point = True
ctx2 = SimulationContext(cfg2, unit_cell=make_synthetic_cell(64), device=dev)
dft2 = DFTGroundState(KPointSet(ctx2)).initial_state()
t0=time.time(); r1 = dft2.find(num_dft_iter=6); t_gamma = time.time()-t0
out["si64_plain"] = {"etot": r0["energy"]["total"], "t": t_plain}
out["si64_gamma"] = {"etot": r1["energy"]["total"], "t": t_gamma}
# 2) fp32 on GPU
cfg3, _ = make_synthetic_config(natoms=64, gk_cutoff=5.0, pw_cutoff=14.0, ngridk=(1,1,1))
cfg3._data["parameters"]["precision_wf"] = "fp32"; cfg3.parameters.precision_wf = "fp32"
ctx3 = SimulationContext(cfg3, unit_cell=make_synthetic_cell(64), device=dev)
dft3 = DFTGroundState(KPointSet(ctx3)).initial_state()
t0=time.time(); r2 = dft3.find(num_dft_iter=6); t_fp32 = time.time()-t0
out["si64_fp32"] = {"etot": r2["energy"]["total"], "t": t_fp32}
print(json.dumps(out, indent=1))
