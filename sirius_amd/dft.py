"""SCF ground-state driver.

Reference behavior: src/dft/dft_ground_state.cpp —
initial_state (:23): atomic-density superposition, potential, LCAO
subspace init; find (:179): SCF loop = build H0 → diagonalize (Davidson)
→ Fermi occupancies → new density → mix → adaptive solver tolerance
(:253-262) → new potential → scf-correction energy (:320-323) → total
energy (energy.cpp:142-165) → convergence on |ΔE| and RMS (:349-356).
"""

from __future__ import annotations

import math
import time

import numpy as np
import torch

from .core import la
from .core import ylm as ylm_mod
from .core.radial import RadialIntegrals
from .davidson import davidson
from .density import Density
from .hamiltonian import Hamiltonian0
from .kpoint import KPointSet
from .parallel import get_comm
from .potential import Potential
from .utils.profiler import profiler


def atomic_orbitals(ctx, kp) -> torch.Tensor:
    """LCAO trial orbitals χ_lm(G+k) for all atoms [n_ao, nGk]
    (reference: initialize_subspace.hpp:27 via Radial_integrals_atomic_wf)."""
    uc = ctx.unit_cell
    g = kp.gkvec
    glen = g.gk_len
    _, theta, phi = ylm_mod.spherical_coords(g.gkvec_cart)
    lmax = max((w.l for at in uc.atom_types.values() for w in at.atomic_wfs), default=0)
    rl = ylm_mod.rlm(lmax, theta, phi)
    cols_t = {}
    for lab, at in uc.atom_types.items():
        cols = []
        for w in at.atomic_wfs:
            f = RadialIntegrals.sbessel_transform(w.l, at.r, w.f_r, glen, rpow=1)
            z = (-1j) ** w.l * (4 * math.pi / math.sqrt(uc.omega))
            for m in range(-w.l, w.l + 1):
                cols.append(z * rl[:, ylm_mod.lm_index(w.l, m)] * f)
        cols_t[lab] = np.stack(cols, axis=1) if cols else np.zeros((len(glen), 0),
                                                                   dtype=np.complex128)
    blocks = []
    mk = (g.miller + g.k_frac).astype(np.float64)
    for lab, tau in uc.atoms:
        if cols_t[lab].shape[1] == 0:
            continue
        phase = np.exp(-2j * math.pi * (mk @ tau))
        blocks.append((cols_t[lab] * phase[:, None]).T)  # [nao_a, nGk]
    if not blocks:
        return torch.zeros(0, g.num_gvec, dtype=ctx.dtype, device=ctx.device)
    return torch.from_numpy(np.concatenate(blocks, axis=0)).to(ctx.device)


def initialize_subspace(ctx, kp, hk):
    """LCAO + random trial basis, one Rayleigh-Ritz step
    (initialize_subspace.hpp:27-150)."""
    nb = ctx.num_bands
    init = ctx.cfg.iterative_solver.init_subspace
    phi = atomic_orbitals(ctx, kp) if init == "lcao" else \
        torch.zeros(0, kp.num_gkvec, dtype=ctx.dtype, device=ctx.device)
    n_ao = phi.shape[0] * ctx.num_spinors
    if n_ao < nb:
        gen = torch.Generator(device="cpu").manual_seed(12345 + kp.gkvec.num_gvec)
        nrnd = (nb - n_ao + 1) // ctx.num_spinors
        if bool(ctx.cfg.parameters.gamma_point) \
                and float(np.abs(kp.k_frac).max()) < 1e-12:
            # Γ-trick: random REAL-space fields give exactly
            # conjugate-symmetric c(-G) = c*(G) trial vectors
            chunks = []
            step = max(1, kp.fft._band_chunk(nrnd))
            for i in range(0, nrnd, step):
                nn_ = min(step, nrnd - i)
                rr = torch.randn(nn_, *kp.fft.dims, generator=gen,
                                 dtype=torch.float64).to(ctx.device)
                chunks.append(kp.fft.to_pw(rr.to(ctx.dtype)))
            rnd = torch.cat(chunks, dim=0)
        else:
            rnd = torch.randn(nrnd, kp.num_gkvec, 2,
                              generator=gen, dtype=torch.float64)
            rnd = torch.view_as_complex(rnd.contiguous()).to(ctx.device)
        # damp high-G components for smoother start
        damp = 1.0 / (1.0 + kp.gkvec.gk2_t)
        rnd = rnd * damp
        phi = torch.cat([phi, rnd], dim=0)

    if ctx.nc_magnetism:
        # spinor trial: each scalar orbital as (φ,0) and (0,φ)
        n0 = phi.shape[0]
        z = torch.zeros_like(phi)
        phi = torch.cat([torch.cat([phi, z], dim=1),
                         torch.cat([z, phi], dim=1)], dim=0)

    from .davidson import _ortho_joint
    hphi, sphi = hk.apply_h_s(phi)
    phi, hphi, sphi = _ortho_joint(phi, hphi, sphi, None, None, None)
    H = la.inner(phi, hphi)
    H = 0.5 * (H + H.conj().T)
    evals, Z = la.eigh(H)
    nsel = min(nb, phi.shape[0])
    psi = la.transform(Z[:, :nsel], phi)
    if nsel < nb:
        raise RuntimeError("not enough trial orbitals for requested bands")
    for ispn in range(ctx.num_spin_steps):
        kp.psi[ispn] = psi
        kp.eigvals[ispn] = evals[:nb].real.cpu().numpy()


def _gamma_neg_index(kp):
    """Index of -G for every G of the k=0 sphere (cached on the kp)."""
    idx = getattr(kp, "_gamma_neg", None)
    if idx is None:
        key = {tuple(m): i for i, m in enumerate(kp.gkvec.miller)}
        neg = np.array([key[tuple(-m)] for m in kp.gkvec.miller],
                       dtype=np.int64)
        idx = torch.from_numpy(neg).to(kp.psi.device)
        kp._gamma_neg = idx
    return idx


def diagonalize_exact(ctx, h0: Hamiltonian0, kset: KPointSet) -> bool:
    """Exact dense diagonalization (iterative_solver.type = "exact";
    reference diagonalize_pp_exact, diagonalize_pp.hpp:24): build the
    full PW Hamiltonian/overlap H(G,G') = ½|G+k|²δ + V_eff(G−G') +
    Σ β D β†, S = I + β Q β† and solve the generalized problem.  A
    correctness tool for small basis sets."""
    for kp in kset:
        hk = h0(kp)
        ngk = kp.num_gkvec
        if ngk > 6000:
            raise RuntimeError(f"exact solver: basis too large ({ngk})")
        if ctx.nc_magnetism:
            raise NotImplementedError("exact solver for spinors TODO")
        # V_eff(G-G') from the coarse sphere
        veff_pw = {}
        m_c = ctx.gvec_coarse.miller
        lutd = {tuple(mm): i for i, mm in enumerate(m_c)}
        mgk = kp.gkvec.miller
        d = mgk[:, None, :] - mgk[None, :, :]
        idx = np.empty((ngk, ngk), dtype=np.int64)
        ok = np.ones((ngk, ngk), dtype=bool)
        for i in range(ngk):
            for j in range(ngk):
                t = lutd.get(tuple(d[i, j]))
                if t is None:
                    ok[i, j] = False
                    idx[i, j] = 0
                else:
                    idx[i, j] = t
        for ispn in range(ctx.num_spin_steps):
            vc = ctx.fft_coarse.to_pw(
                hk.h0.veff_r_coarse[ispn].to(ctx.dtype)).cpu().numpy()
            V = vc[idx] * ok
            H = torch.from_numpy(V).to(ctx.device)
            H += torch.diag(hk.ekin.to(H.dtype))
            S = torch.eye(ngk, dtype=H.dtype, device=ctx.device)
            if hk.bp.num_beta_total:
                B = hk.bp.beta_t                    # [nbf, ngk]
                H += B.T @ hk.D[ispn].to(H.dtype) @ B.conj()
                if hk.Q is not None:
                    S += B.T @ hk.Q.to(H.dtype) @ B.conj()
            from scipy.linalg import eigh as seigh
            w, v = seigh(H.cpu().numpy(), S.cpu().numpy(),
                         subset_by_index=[0, ctx.num_bands - 1])
            kp.eigvals[ispn] = w
            kp.psi[ispn] = torch.from_numpy(
                np.ascontiguousarray(v.T)).to(ctx.device).to(ctx.dtype)
    return True


def _solve_kps_threaded(ctx, kps, solve_fn, nthreads):
    """Run the independent per-k-point Davidson solves on a small thread
    pool, one HIP stream per worker.  Each k-point is handled by exactly
    one worker; per-kp state (kp.fft staging buffer, kp.psi) is only ever
    touched by that worker within the region, and every stream is
    host-synchronized before returning, so downstream serial code
    (density generate) sees completed results.

    Kept as an opt-in experiment (control.num_kpoint_threads > 1):
    measured on sto-uspp (64 k, 1 MI355X) the per-k loop is GIL-bound —
    thousands of small launches per SCF iteration — and 2/4 workers run
    1.2x/2.1x SLOWER than serial while producing bit-identical energies.
    The GPU-side fix for this regime is batching k-points into single
    tensor ops, not host threads (NEXT.md)."""
    import itertools
    from concurrent.futures import ThreadPoolExecutor

    streams = getattr(ctx, "_kp_streams", None)
    if streams is None or len(streams) < nthreads:
        streams = [torch.cuda.Stream(device=ctx.device)
                   for _ in range(nthreads)]
        ctx._kp_streams = streams
    nxt = itertools.count()          # CPython: __next__ is atomic
    outs = [None] * len(kps)

    def run_shard(wid):
        s = streams[wid]
        s.wait_stream(torch.cuda.default_stream(ctx.device))
        with torch.cuda.stream(s):
            while True:
                i = next(nxt)
                if i >= len(kps):
                    break
                outs[i] = solve_fn(kps[i])
        s.synchronize()

    import sys
    from .core import la
    saved_nt = torch.get_num_threads()
    saved_si = sys.getswitchinterval()
    try:
        # pin the (global) intra-op pool once instead of per-eigh toggling
        # from 4 threads, and shorten the GIL switch interval: the workers
        # interleave many short C-extension calls
        torch.set_num_threads(la._HOST_SOLVE_THREADS)
        la.PIN_HOST_THREADS = True
        sys.setswitchinterval(0.0005)
        with ThreadPoolExecutor(max_workers=nthreads) as ex:
            futs = [ex.submit(run_shard, w) for w in range(nthreads)]
            for f in futs:
                f.result()
    finally:
        la.PIN_HOST_THREADS = False
        sys.setswitchinterval(saved_si)
        torch.set_num_threads(saved_nt)
    return outs


def diagonalize(ctx, h0: Hamiltonian0, kset: KPointSet, itsol_tol: float,
                wf_dtype=None) -> bool:
    """Davidson for all local k-points/spins (reference diagonalize.hpp).

    wf_dtype=torch.complex64 runs the fp32 wave-function mode
    (reference precision_wf, dft_ground_state.cpp:269-304): trial
    vectors, H/S application and the subspace algebra all in single
    precision; converged psi is stored back in fp64."""
    itso = ctx.cfg.iterative_solver
    if str(itso.type) == "exact":
        return diagonalize_exact(ctx, h0, kset)
    empy_tol = max(itsol_tol * itso.tolerance_ratio, itso.empty_states_tolerance)
    fp32 = wf_dtype == torch.complex64
    use_gamma = bool(ctx.cfg.parameters.gamma_point) \
        and not ctx.nc_magnetism and ctx.hubbard is None
    bc = getattr(ctx, "band_comm", None)
    band_par = bc is not None and bc.active

    def _solve_kp(kp):
        hk = h0(kp)
        o_diag = hk.o_diag()
        if fp32:
            o_diag = o_diag.to(torch.float32)
        conv, nit, work = True, 0, 0
        for ispn in range(ctx.num_spin_steps):
            h_diag = hk.h_diag(ispn)
            psi0 = kp.psi[ispn]
            if fp32:
                h_diag = h_diag.to(torch.float32)
                psi0 = psi0.to(torch.complex64)
            if band_par:
                # band-parallel H/S application: each band-group rank
                # applies its slice of the trial block, results are
                # allgathered (replicated storage, split
                # compute — the FFT-bound apply dominates)
                def apply_fn(p, s=ispn):
                    n = p.shape[0]
                    cnt = [n // bc.size + (1 if r < n % bc.size else 0)
                           for r in range(bc.size)]
                    o = int(np.cumsum([0] + cnt)[bc.rank])
                    h_loc, s_loc = hk.apply_h_s(p[o:o + cnt[bc.rank]], s)
                    h = bc.allgather_rows(h_loc, cnt)
                    sg = bc.allgather_rows(s_loc, cnt) \
                        if s_loc is not None else None
                    return h, sg
            else:
                def apply_fn(p, s=ispn):
                    return hk.apply_h_s(p, s)
            res = davidson(
                apply_fn,
                psi0, h_diag, o_diag, occ=kp.occ[ispn],
                tol_occ=itsol_tol, tol_empty=empy_tol,
                num_steps=itso.num_steps, subspace_size=itso.subspace_size,
                max_block=int(itso.get("max_block_size", 0)),
                min_occupancy=itso.min_occupancy, extra_ortho=itso.extra_ortho,
                locking=bool(itso.locking),
                early_restart=float(itso.early_restart),
                gamma=use_gamma and float(np.abs(kp.k_frac).max()) < 1e-12,
                gamma_neg=_gamma_neg_index(kp)
                if use_gamma and float(np.abs(kp.k_frac).max()) < 1e-12
                else None)
            kp.psi[ispn] = res.psi.to(torch.complex128) if fp32 else res.psi
            kp.eigvals[ispn] = res.eval
            nit += res.niter
            work += res.evp_work
            conv = conv and res.converged
        return conv, nit, work

    kps = list(kset)
    # auto = serial: measured on sto-uspp/MI355X the loop is GIL-bound and
    # threads only add contention (see control.num_kpoint_threads)
    nthreads = max(1, int(ctx.cfg.control.num_kpoint_threads or 0))
    nthreads = min(nthreads, max(1, len(kps)))
    if nthreads > 1 and str(ctx.device) != "cpu" and not band_par:
        outs = _solve_kps_threaded(ctx, kps, _solve_kp, nthreads)
    else:
        outs = [_solve_kp(kp) for kp in kps]
    all_conv = True
    for conv, nit, work in outs:
        ctx.counters["num_itsol_steps"] += nit
        ctx.counters["band_evp_work_count"] += work
        all_conv = all_conv and conv
    comm = get_comm()
    if comm.active:
        all_conv = bool(comm.allreduce_scalar(float(all_conv)) == comm.size)
    return all_conv


class DFTGroundState:
    def __init__(self, kset: KPointSet):
        self.kset = kset
        self.ctx = kset.ctx
        self.density = Density(self.ctx)
        self.potential = Potential(self.ctx)
        self.scf_correction_energy = 0.0

    # -- energies (energy.cpp) --------------------------------------------

    def energy_potential(self, rho_r: torch.Tensor, mag_r=None) -> float:
        """∫ρ V_eff + ∫m·B (+ PAW/hubbard terms when present;
        reference energy.cpp:251-257)."""
        e = self.ctx.integrate_rg_fine(rho_r * self.potential.veff_r)
        if self.ctx.nc_magnetism and self.potential.bvec_r is not None:
            for i in range(3):
                e += self.ctx.integrate_rg_fine(
                    self.density.magv_r[i] * self.potential.bvec_r[i])
        elif mag_r is not None and self.potential.bz_r is not None:
            e += self.ctx.integrate_rg_fine(mag_r * self.potential.bz_r)
        # NOTE: the reference's energy_potential also adds hubbard_energy
        # (energy.cpp:251-257), but both SCF-correction evaluations use the
        # SAME occupation matrix, so the Hubbard term cancels in e2-e1 and
        # is deliberately left out here.
        return e

    def total_energy_components(self) -> dict:
        d = {}
        d["valence_eval_sum"] = self.kset.valence_eval_sum()
        d["vxc"] = self.potential.energy_vxc(self.density)
        d["bxc"] = self.potential.energy_bxc(self.density)
        d["vha"] = self.potential.energy_vha
        d["exc"] = self.potential.energy_exc(self.density)
        d["vloc"] = self.potential.energy_vloc(self.density)
        d["ewald"] = self.potential.ewald
        d["entropy"] = self.kset.entropy_sum()
        d["scf_correction"] = self.scf_correction_energy
        d["fermi"] = self.kset.energy_fermi
        hub = self.ctx.hubbard
        d["hubbard_energy"] = hub.energy() if hub else 0.0
        d["hubbard_one_el"] = hub.one_electron_energy() if hub else 0.0
        paw = self.ctx.paw
        d["PAW_total_energy"] = paw.total_energy() if paw else 0.0
        d["PAW_one_elec"] = paw.one_elec_energy(self.density) if paw else 0.0
        return d

    def total_energy(self) -> float:
        """KS total energy, PP branch (energy.cpp:152-157) + scf correction."""
        d = self.total_energy_components()
        return (d["valence_eval_sum"] - d["vxc"] - d["bxc"] - d["PAW_one_elec"]
                - 0.5 * d["vha"] + d["exc"] + d["PAW_total_energy"] + d["ewald"]
                + d["hubbard_energy"] - d["hubbard_one_el"]
                + d["scf_correction"])

    # -- driver ------------------------------------------------------------

    def initial_state(self):
        self.density.initial_density()
        if self.ctx.hubbard is not None:
            self.ctx.hubbard.initial_occupation()
        self.potential.generate(self.density)
        self.potential.generate_paw(self.density)
        h0 = Hamiltonian0(self.ctx, self.potential)
        for kp in self.kset:
            initialize_subspace(self.ctx, kp, h0(kp))
        self.kset.find_band_occupancies()
        return self

    def find(self, density_tol=None, energy_tol=None, itsol_tol=None,
             num_dft_iter=None, callback=None) -> dict:
        ctx = self.ctx
        p = ctx.cfg.parameters
        itso = ctx.cfg.iterative_solver
        density_tol = density_tol if density_tol is not None else p.density_tol
        energy_tol = energy_tol if energy_tol is not None else p.energy_tol
        itsol_tol = itsol_tol if itsol_tol is not None else itso.energy_tolerance
        num_dft_iter = num_dft_iter if num_dft_iter is not None else p.num_dft_iter

        self.density.mixer_init(ctx.cfg.mixer)
        eold = 0.0
        etot_hist, rms_hist = [], []
        num_iter = -1
        t0 = time.time()

        # fp32 wave-function mode with runtime fp64 promotion
        # (reference precision_wf + fp32_to_fp64_rms,
        # dft_ground_state.cpp:269-304)
        import torch as _torch
        wf_dtype = _torch.complex64 \
            if str(getattr(p, "precision_wf", "fp64")) == "fp32" else None
        fp32_rms = float(ctx.cfg.settings.fp32_to_fp64_rms) or \
            max(10.0 * density_tol, 1e-5)

        for it in range(num_dft_iter):
            with profiler("scf_iteration"):
                with profiler("Hamiltonian0"):
                    h0 = Hamiltonian0(ctx, self.potential, self.density)
                    self.h0 = h0  # kept for post-SCF forces (D at diag-time V)
                with profiler("diagonalize"):
                    bands_converged = diagonalize(ctx, h0, self.kset,
                                                  itsol_tol,
                                                  wf_dtype=wf_dtype)
                with profiler("occupancies"):
                    self.kset.find_band_occupancies()
                with profiler("density"):
                    self.density.generate(self.kset, h0)

            e1 = self.energy_potential(self.density.rho_r, self.density.mag_r)
            rho1_r = self.density.rho_r.clone()
            mag1_r = self.density.mag_r.clone() if self.density.mag_r is not None else None
            magv1_r = [m.clone() for m in self.density.magv_r] \
                if ctx.nc_magnetism else None

            with profiler("mix"):
                rms = self.density.mix()

            if wf_dtype is not None and rms < fp32_rms:
                # promote to fp64 for the remaining iterations
                wf_dtype = None
                if ctx.cfg.control.verbosity >= 1:
                    print(f"[scf] switching wave functions fp32 -> fp64 "
                          f"at iteration {it} (rms {rms:.2e})", flush=True)

            tol = rms
            tol = min(itso.tolerance_scale[0] * tol,
                      itso.tolerance_scale[1] * itsol_tol)
            itsol_tol = max(itso.min_tolerance, tol)
            itsol_converged = tol <= itso.min_tolerance

            paw = ctx.paw
            dm_fresh = {k: v.clone() for k, v in
                        self.density.density_matrix.items()}                 if (paw and self.density.density_matrix) else None
            if paw:
                class _D:
                    density_matrix = dm_fresh
                e1 += paw.one_elec_energy(_D)

            with profiler("potential"):
                self.potential.generate(self.density)
                self.potential.generate_paw(self.density)

            e2 = self.ctx.integrate_rg_fine(rho1_r * self.potential.veff_r)
            if magv1_r is not None and self.potential.bvec_r is not None:
                # noncollinear: the full vector m·B enters both potential
                # energies (energy.cpp:251-257) — z alone leaves a spurious
                # residual in e2−e1 at self-consistency
                for i in range(3):
                    e2 += self.ctx.integrate_rg_fine(
                        magv1_r[i] * self.potential.bvec_r[i])
            elif mag1_r is not None and self.potential.bz_r is not None:
                e2 += self.ctx.integrate_rg_fine(mag1_r * self.potential.bz_r)
            if paw:
                e2 += paw.one_elec_energy(_D)
            self.scf_correction_energy = e2 - e1

            etot = self.total_energy()
            etot_hist.append(etot)
            rms_hist.append(rms)
            if callback:
                callback(it, etot, rms)

            conv = (abs(eold - etot) < energy_tol) and bands_converged \
                and itsol_converged and (rms < density_tol)
            if conv:
                num_iter = it
                break
            eold = etot

        out = {
            "timers": profiler.to_dict(),
            "converged": num_iter >= 0,
            "num_scf_iterations": num_iter,
            "energy": self.total_energy_components() | {
                "total": self.total_energy(),
                "free": self.total_energy() + self.kset.entropy_sum(),
            },
            "etot_history": etot_hist,
            "rms_history": rms_hist,
            "scf_time": time.time() - t0,
            "efermi": self.kset.energy_fermi,
            "magnetization": self.density.total_magnetization()
            if ctx.num_mag_dims else 0.0,
            "counters": dict(ctx.counters),
            "band_gap": self.kset.band_gap,
        }
        return out

    def forces(self, add_scf_corr: bool = True) -> dict:
        """Post-SCF atomic forces (reference: Force::calc_forces_total;
        the non-local term uses the D matrices of the LAST diagonalization
        while vloc/us/core/scf_corr use the final potential — matching the
        reference's sequencing, where scf_corr compensates exactly that
        potential lag)."""
        from .forces import Forces

        f = Forces(self.ctx, self.density, self.potential, self.kset, self.h0)
        return f.calc_forces_total(add_scf_corr=add_scf_corr)

    def stress(self) -> dict:
        """Post-SCF stress tensor (reference: Stress::calc_stress_total)."""
        from .stress import Stress

        st = Stress(self.ctx, self.density, self.potential, self.kset,
                    self.h0)
        return st.calc_stress_total()
