"""Exchange-correlation functionals, evaluated on torch tensors (device-resident).

Reference behavior: src/potential/xc.cpp:421 (+xc_functional_base.hpp) which
delegates to libxc. libxc is not in this stack; the functionals the
verification decks use (XC_LDA_X, XC_LDA_C_PZ, XC_LDA_C_PW, XC_GGA_X_PBE,
XC_GGA_C_PBE) are implemented directly here as fp64 torch expressions so
they run fused on the GPU real-space grid.

Interface per functional: lda(rho) -> (eps, vrho); gga(rho, sigma) ->
(eps, vrho, vsigma), with eps the energy density per particle (libxc zk
convention), sigma = |grad rho|^2.
"""

from __future__ import annotations

import math

import torch

_THIRD = 1.0 / 3.0


def _safe_rho(rho: torch.Tensor) -> torch.Tensor:
    return torch.clamp(rho, min=1e-24)


# -- LDA exchange (Slater) --------------------------------------------------

def lda_x(rho: torch.Tensor):
    """XC_LDA_X, unpolarized: eps_x = -(3/4)(3/pi)^{1/3} rho^{1/3}."""
    rho = _safe_rho(rho)
    cx = (3.0 / 4.0) * (3.0 / math.pi) ** _THIRD
    r13 = rho ** _THIRD
    eps = -cx * r13
    vrho = -(4.0 / 3.0) * cx * r13
    return eps, vrho


# -- LDA correlation: Perdew-Zunger 81 -------------------------------------

_PZ_U = dict(A=0.0311, B=-0.048, C=0.0020, D=-0.0116,
             gamma=-0.1423, beta1=1.0529, beta2=0.3334)


def lda_c_pz(rho: torch.Tensor):
    """XC_LDA_C_PZ, unpolarized."""
    rho = _safe_rho(rho)
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    p = _PZ_U
    lo = rs < 1.0
    lnrs = torch.log(rs)
    eps_lo = p["A"] * lnrs + p["B"] + p["C"] * rs * lnrs + p["D"] * rs
    # v = eps - rs/3 deps/drs
    deps_lo = p["A"] / rs + p["C"] * (lnrs + 1.0) + p["D"]
    sq = torch.sqrt(rs)
    den = 1.0 + p["beta1"] * sq + p["beta2"] * rs
    eps_hi = p["gamma"] / den
    deps_hi = -p["gamma"] * (0.5 * p["beta1"] / sq + p["beta2"]) / den**2
    eps = torch.where(lo, eps_lo, eps_hi)
    deps = torch.where(lo, deps_lo, deps_hi)
    vrho = eps - rs / 3.0 * deps
    return eps, vrho


# -- LDA correlation: Perdew-Wang 92 ----------------------------------------

def _pw_g(rs, A, a1, b1, b2, b3, b4):
    sq = torch.sqrt(rs)
    q0 = -2.0 * A * (1.0 + a1 * rs)
    q1 = 2.0 * A * (b1 * sq + b2 * rs + b3 * rs * sq + b4 * rs * rs)
    lg = torch.log(1.0 + 1.0 / q1)
    g = q0 * lg
    # dg/drs
    dq0 = -2.0 * A * a1
    dq1 = A * (b1 / sq + 2.0 * b2 + 3.0 * b3 * sq + 4.0 * b4 * rs)
    dg = dq0 * lg - q0 * dq1 / (q1 * (q1 + 1.0))
    return g, dg


def lda_c_pw(rho: torch.Tensor):
    """XC_LDA_C_PW, unpolarized."""
    rho = _safe_rho(rho)
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    eps, deps = _pw_g(rs, 0.0310907, 0.21370, 7.5957, 3.5876, 1.6382, 0.49294)
    vrho = eps - rs / 3.0 * deps
    return eps, vrho


def _vwn_eps(x, A, b, c, x0):
    """VWN interpolation e(x), de/dx with x = sqrt(rs)."""
    Q = math.sqrt(4 * c - b * b)
    X = x * x + b * x + c
    X0 = x0 * x0 + b * x0 + c
    t = 2 * x + b
    atn = torch.atan(Q / t)
    e = A * (torch.log(x * x / X) + 2 * b / Q * atn
             - b * x0 / X0 * (torch.log((x - x0) ** 2 / X)
                              + 2 * (b + 2 * x0) / Q * atn))
    de = A * (2.0 / x - t / X - 4 * b / (Q * Q + t * t)
              - b * x0 / X0 * (2.0 / (x - x0) - t / X
                               - 4 * (b + 2 * x0) / (Q * Q + t * t)))
    return e, de


def lda_c_vwn(rho: torch.Tensor):
    """XC_LDA_C_VWN (VWN5), unpolarized (libxc parameters:
    A=0.0310907, b=3.72744, c=12.9352, x0=-0.10498)."""
    rho = _safe_rho(rho)
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    x = torch.sqrt(rs)
    e, de = _vwn_eps(x, 0.0310907, 3.72744, 12.9352, -0.10498)
    # v = e + rho de/drho = e - (rs/3) de/drs = e - (x/6) de/dx
    vrho = e - x / 6.0 * de
    return e, vrho


def lda_c_vwn_spin(ru: torch.Tensor, rd: torch.Tensor):
    """XC_LDA_C_VWN spin-polarized (VWN5 interpolation with the spin
    stiffness alpha_c; standard libxc scheme)."""
    rho = _safe_rho(ru + rd)
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    z = (ru - rd) / rho
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    x = torch.sqrt(rs)
    eP, deP = _vwn_eps(x, 0.0310907, 3.72744, 12.9352, -0.10498)
    eF, deF = _vwn_eps(x, 0.01554535, 7.06042, 18.0578, -0.32500)
    eA, deA = _vwn_eps(x, -1.0 / (6.0 * math.pi ** 2), 1.13107, 13.0045,
                       -0.00475840)
    fz = _fzeta(z)
    dfz = _dfzeta(z)
    d2f0 = 4.0 / (9.0 * (2.0 ** (1.0 / 3) - 1.0))
    z4 = z ** 4
    g = fz / d2f0 * (1 - z4)
    e = eP + eA * g + (eF - eP) * fz * z4
    de_dx = deP + deA * g + (deF - deP) * fz * z4
    de_dz = eA / d2f0 * (dfz * (1 - z4) - fz * 4 * z ** 3) \
        + (eF - eP) * (dfz * z4 + fz * 4 * z ** 3)
    # v_sigma = e + rho de/drho_sigma; de/drho_s = (de/drs)(drs/drho) + (de/dz)(dz/drho_s)
    common = e - x / 6.0 * de_dx
    vu = common + de_dz * (1.0 - z)
    vd = common - de_dz * (1.0 + z)
    return e, vu, vd


# -- PBE ---------------------------------------------------------------------

_PBE_KAPPA = 0.8040
_PBE_MU = 0.2195149727645171
_PBE_BETA = 0.06672455060314922
_PBE_GAMMA = (1.0 - math.log(2.0)) / math.pi**2


def gga_x_pbe(rho: torch.Tensor, sigma: torch.Tensor, mu: float = _PBE_MU,
              kappa: float = _PBE_KAPPA, rpbe: bool = False):
    """XC_GGA_X_PBE family, unpolarized: PBE (default), PBEsol
    (mu=10/81), revPBE (kappa=1.245, XC_GGA_X_PBE_R) and RPBE
    (Hammer-Hansen-Norskov exponential form, XC_GGA_X_RPBE).
    Returns (eps, vrho, vsigma)."""
    rho = _safe_rho(rho)
    sigma = torch.clamp(sigma, min=1e-40)
    kf = (3.0 * math.pi**2 * rho) ** _THIRD
    # s^2 = sigma / (2 kf rho)^2
    s2 = sigma / (2.0 * kf * rho) ** 2
    if rpbe:
        ex = torch.exp(-mu * s2 / kappa)
        fx = 1.0 + kappa * (1.0 - ex)
        dfx_ds2 = mu * ex
    else:
        fdenom = 1.0 + mu * s2 / kappa
        fx = 1.0 + kappa - kappa / fdenom
        dfx_ds2 = mu / fdenom**2
    cx = (3.0 / 4.0) * (3.0 / math.pi) ** _THIRD
    eps_unif = -cx * rho ** _THIRD
    eps = eps_unif * fx
    # d(rho*eps)/drho and d(rho*eps)/dsigma
    # rho*eps = eps_unif(rho)*rho*fx(s2); s2 ~ sigma * rho^{-8/3} * const
    c_s2 = 1.0 / (4.0 * (3.0 * math.pi**2) ** (2.0 / 3.0))
    # s2 = c_s2 * sigma * rho^{-8/3}
    ds2_drho = -(8.0 / 3.0) * c_s2 * sigma * rho ** (-11.0 / 3.0)
    ds2_dsigma = c_s2 * rho ** (-8.0 / 3.0)
    d_rho_eps_unif = -(4.0 / 3.0) * cx * rho ** _THIRD  # d(rho eps_unif)/drho
    vrho = d_rho_eps_unif * fx + eps_unif * rho * dfx_ds2 * ds2_drho
    vsigma = eps_unif * rho * dfx_ds2 * ds2_dsigma
    return eps, vrho, vsigma


def gga_c_pbe(rho: torch.Tensor, sigma: torch.Tensor, beta: float = _PBE_BETA):
    """XC_GGA_C_PBE (or PBEsol with beta=0.046), unpolarized.
    Returns (eps, vrho, vsigma)."""
    rho = _safe_rho(rho)
    sigma = torch.clamp(sigma, min=1e-40)
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    ec, dec = _pw_g(rs, 0.0310907, 0.21370, 7.5957, 3.5876, 1.6382, 0.49294)
    gamma = _PBE_GAMMA
    kf = (3.0 * math.pi**2 * rho) ** _THIRD
    ks = torch.sqrt(4.0 * kf / math.pi)
    # t^2 = sigma / (2 ks rho)^2
    t2 = sigma / (2.0 * ks * rho) ** 2
    expo = torch.exp(-ec / gamma)
    A = beta / gamma / (expo - 1.0 + 1e-30)
    At2 = A * t2
    num = 1.0 + At2
    den = 1.0 + At2 + At2 * At2
    arg = 1.0 + beta / gamma * t2 * num / den
    H = gamma * torch.log(arg)
    eps = ec + H

    # derivatives via autograd-free chain rule
    # dH/dt2 and dH/dA
    dnum_dt2 = A
    dden_dt2 = A + 2.0 * A * At2
    q = t2 * num / den
    dq_dt2 = (num + t2 * dnum_dt2) / den - t2 * num * dden_dt2 / den**2
    dH_darg = gamma / arg
    dH_dt2 = dH_darg * beta / gamma * dq_dt2
    dnum_dA = t2
    dden_dA = t2 + 2.0 * t2 * At2
    dq_dA = t2 * (dnum_dA * den - num * dden_dA) / den**2
    dH_dA = dH_darg * beta / gamma * dq_dA
    dA_dec = A * expo / (gamma * (expo - 1.0 + 1e-30))
    # t2 = c_t2 * sigma * rho^{-7/3} with c_t2 = pi/16 * (3 pi^2)^{-1/3}... derive:
    # (2 ks rho)^2 = 4 ks^2 rho^2 = 16 kf/pi rho^2 = (16/pi)(3pi^2)^{1/3} rho^{7/3}
    c_t2 = math.pi / 16.0 * (3.0 * math.pi**2) ** (-1.0 / 3.0)
    dt2_drho = -(7.0 / 3.0) * c_t2 * sigma * rho ** (-10.0 / 3.0)
    dt2_dsigma = c_t2 * rho ** (-7.0 / 3.0)
    dec_drho = dec * (-rs / (3.0 * rho))  # drs/drho = -rs/(3 rho)
    deps_drho = dec_drho + dH_dt2 * dt2_drho + dH_dA * dA_dec * dec_drho
    deps_dsigma = dH_dt2 * dt2_dsigma
    vrho = eps + rho * deps_drho
    vsigma = rho * deps_dsigma
    return eps, vrho, vsigma


# -- spin-polarized variants (libxc-compatible conventions) -----------------
# eps = energy density per particle of the total density; E = ∫ n·eps.

_PZ_P = dict(A=0.01555, B=-0.0269, C=0.0007, D=-0.0048,
             gamma=-0.0843, beta1=1.3981, beta2=0.2611)


def _fzeta(z):
    c = 1.0 / (2.0 ** (4.0 / 3.0) - 2.0)
    return c * ((1 + z) ** (4.0 / 3.0) + (1 - z) ** (4.0 / 3.0) - 2.0)


def _dfzeta(z):
    c = 1.0 / (2.0 ** (4.0 / 3.0) - 2.0)
    return c * (4.0 / 3.0) * ((1 + z) ** (1.0 / 3.0) - (1 - z) ** (1.0 / 3.0))


def lda_x_spin(ru: torch.Tensor, rd: torch.Tensor):
    """Exchange by exact spin scaling: E_x[n↑,n↓] = ½(E_x[2n↑]+E_x[2n↓])."""
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    n = ru + rd
    cx = (3.0 / 4.0) * (3.0 / math.pi) ** _THIRD
    eu = -cx * (2.0 * ru) ** _THIRD
    ed = -cx * (2.0 * rd) ** _THIRD
    eps = (ru * eu + rd * ed) / n
    vu = -(4.0 / 3.0) * cx * (2.0 * ru) ** _THIRD
    vd = -(4.0 / 3.0) * cx * (2.0 * rd) ** _THIRD
    return eps, vu, vd


def _pz_branch(rs, p):
    lo = rs < 1.0
    lnrs = torch.log(rs)
    eps_lo = p["A"] * lnrs + p["B"] + p["C"] * rs * lnrs + p["D"] * rs
    deps_lo = p["A"] / rs + p["C"] * (lnrs + 1.0) + p["D"]
    sq = torch.sqrt(rs)
    den = 1.0 + p["beta1"] * sq + p["beta2"] * rs
    eps_hi = p["gamma"] / den
    deps_hi = -p["gamma"] * (0.5 * p["beta1"] / sq + p["beta2"]) / den**2
    return torch.where(lo, eps_lo, eps_hi), torch.where(lo, deps_lo, deps_hi)


def lda_c_pz_spin(ru: torch.Tensor, rd: torch.Tensor):
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    n = ru + rd
    z = torch.clamp((ru - rd) / n, -1.0 + 1e-15, 1.0 - 1e-15)
    rs = (3.0 / (4.0 * math.pi * n)) ** _THIRD
    eU, dU = _pz_branch(rs, _PZ_U)
    eP, dP = _pz_branch(rs, _PZ_P)
    f = _fzeta(z)
    df = _dfzeta(z)
    eps = eU + f * (eP - eU)
    deps_drs = dU + f * (dP - dU)
    deps_dz = df * (eP - eU)
    common = eps - rs / 3.0 * deps_drs
    vu = common + deps_dz * (1.0 - z)
    vd = common - deps_dz * (1.0 + z)
    return eps, vu, vd


def lda_c_pw_spin(ru: torch.Tensor, rd: torch.Tensor):
    """Perdew-Wang 92 with full ζ interpolation."""
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    n = ru + rd
    z = torch.clamp((ru - rd) / n, -1.0 + 1e-15, 1.0 - 1e-15)
    rs = (3.0 / (4.0 * math.pi * n)) ** _THIRD
    ecU, dU = _pw_g(rs, 0.0310907, 0.21370, 7.5957, 3.5876, 1.6382, 0.49294)
    ecP, dP = _pw_g(rs, 0.01554535, 0.20548, 14.1189, 6.1977, 3.3662, 0.62517)
    mac, dmac = _pw_g(rs, 0.0168869, 0.11125, 10.357, 3.6231, 0.88026, 0.49671)
    ac, dac = -mac, -dmac  # spin stiffness (PW fit returns -alpha_c)
    f = _fzeta(z)
    df = _dfzeta(z)
    fdd0 = 4.0 / (9.0 * (2.0 ** (1.0 / 3.0) - 1.0))
    z4 = z**4
    eps = ecU + ac * f / fdd0 * (1.0 - z4) + (ecP - ecU) * f * z4
    deps_drs = dU + dac * f / fdd0 * (1.0 - z4) + (dP - dU) * f * z4
    deps_dz = ac / fdd0 * (df * (1.0 - z4) - 4.0 * z**3 * f) + \
        (ecP - ecU) * (df * z4 + 4.0 * z**3 * f)
    common = eps - rs / 3.0 * deps_drs
    vu = common + deps_dz * (1.0 - z)
    vd = common - deps_dz * (1.0 + z)
    return eps, vu, vd


def gga_x_pbe_spin(ru, rd, s_uu, s_dd, mu: float = _PBE_MU,
                   kappa: float = _PBE_KAPPA, rpbe: bool = False):
    """PBE-family exchange via exact spin scaling. Returns
    (eps, vu, vd, vs_uu, vs_dd); vsigma_ud = 0 for exchange."""
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    n = ru + rd
    e_u, v_u, vs_u = gga_x_pbe(2.0 * ru, 4.0 * s_uu, mu=mu, kappa=kappa,
                               rpbe=rpbe)
    e_d, v_d, vs_d = gga_x_pbe(2.0 * rd, 4.0 * s_dd, mu=mu, kappa=kappa,
                               rpbe=rpbe)
    eps = (ru * e_u + rd * e_d) / n
    return eps, v_u, v_d, 2.0 * vs_u, 2.0 * vs_d


def gga_c_pbe_spin(ru, rd, sigma, beta_pbe: float = _PBE_BETA):
    """PBE correlation, spin-polarized; sigma = |∇(n↑+n↓)|².
    Returns (eps, vu, vd, vsigma) with vsigma = ∂(n·eps)/∂σ (same for all
    σ components: vsigma_uu = vsigma, vsigma_ud = 2·vsigma, vsigma_dd = vsigma
    in libxc terms — handled by the caller using total-gradient form)."""
    ru = _safe_rho(ru)
    rd = _safe_rho(rd)
    n = ru + rd
    sigma = torch.clamp(sigma, min=1e-40)
    z = torch.clamp((ru - rd) / n, -1.0 + 1e-12, 1.0 - 1e-12)
    rs = (3.0 / (4.0 * math.pi * n)) ** _THIRD
    phi = 0.5 * ((1 + z) ** (2.0 / 3.0) + (1 - z) ** (2.0 / 3.0))
    beta, gamma = beta_pbe, _PBE_GAMMA

    # need ec(rs, z) and its derivatives — reuse lda_c_pw_spin pieces
    ecU, dU = _pw_g(rs, 0.0310907, 0.21370, 7.5957, 3.5876, 1.6382, 0.49294)
    ecP, dP = _pw_g(rs, 0.01554535, 0.20548, 14.1189, 6.1977, 3.3662, 0.62517)
    mac, dmac = _pw_g(rs, 0.0168869, 0.11125, 10.357, 3.6231, 0.88026, 0.49671)
    ac, dac = -mac, -dmac
    f = _fzeta(z)
    df = _dfzeta(z)
    fdd0 = 4.0 / (9.0 * (2.0 ** (1.0 / 3.0) - 1.0))
    z4 = z**4
    ec = ecU + ac * f / fdd0 * (1.0 - z4) + (ecP - ecU) * f * z4
    dec_drs = dU + dac * f / fdd0 * (1.0 - z4) + (dP - dU) * f * z4
    dec_dz = ac / fdd0 * (df * (1.0 - z4) - 4.0 * z**3 * f) + \
        (ecP - ecU) * (df * z4 + 4.0 * z**3 * f)
    dphi_dz = ((1 + z) ** (-1.0 / 3.0) - (1 - z) ** (-1.0 / 3.0)) / 3.0

    kf = (3.0 * math.pi**2 * n) ** _THIRD
    ks = torch.sqrt(4.0 * kf / math.pi)
    t2 = sigma / (2.0 * phi * ks * n) ** 2
    g3 = phi**3
    expo = torch.exp(-ec / (gamma * g3))
    A = beta / gamma / (expo - 1.0 + 1e-30)
    At2 = A * t2
    num = 1.0 + At2
    den = 1.0 + At2 + At2 * At2
    arg = 1.0 + beta / gamma * t2 * num / den
    H = gamma * g3 * torch.log(arg)
    eps = ec + H

    # partials of H wrt t2, A, phi, ec
    q = t2 * num / den
    dq_dt2 = (num + t2 * A) / den - t2 * num * (A + 2.0 * A * At2) / den**2
    dq_dA = t2 * (t2 * den - num * (t2 + 2.0 * t2 * At2)) / den**2
    dH_darg = gamma * g3 / arg
    dH_dt2 = dH_darg * beta / gamma * dq_dt2
    dH_dA = dH_darg * beta / gamma * dq_dA
    dA_dec = A * expo / (gamma * g3 * (expo - 1.0 + 1e-30))
    # d expo/dphi = expo * 3 ec/(gamma g3 phi) => dA/dphi = -A/(expo-1)*dexpo_dphi
    dexpo_dphi = expo * 3.0 * ec / (gamma * g3 * phi)
    dA_dphi = -A * dexpo_dphi / (expo - 1.0 + 1e-30)
    dH_dphi = 3.0 * H / phi + dH_dA * dA_dphi - dH_dt2 * 2.0 * t2 / phi
    dH_dec = dH_dA * dA_dec

    c_t2 = math.pi / 16.0 * (3.0 * math.pi**2) ** (-1.0 / 3.0)
    # t2 = c_t2 * sigma * n^{-7/3} / phi^2
    dt2_dn = -(7.0 / 3.0) * c_t2 * sigma * n ** (-10.0 / 3.0) / phi**2
    dt2_dsigma = c_t2 * n ** (-7.0 / 3.0) / phi**2
    drs_dn = -rs / (3.0 * n)

    deps_dn_at_z = (dec_drs + dH_dec * dec_drs) * drs_dn + dH_dt2 * dt2_dn
    deps_dz_tot = dec_dz + dH_dec * dec_dz + dH_dphi * dphi_dz
    common = eps + n * deps_dn_at_z
    vu = common + deps_dz_tot * (1.0 - z)
    vd = common - deps_dz_tot * (1.0 + z)
    vsigma = n * dH_dt2 * dt2_dsigma
    return eps, vu, vd, vsigma


_PBESOL_MU = 10.0 / 81.0
_PBESOL_BETA = 0.046


def gga_x_pbesol(rho, sigma):
    return gga_x_pbe(rho, sigma, mu=_PBESOL_MU)


def gga_c_pbesol(rho, sigma):
    return gga_c_pbe(rho, sigma, beta=_PBESOL_BETA)


_LDA = {"XC_LDA_X": lda_x, "XC_LDA_C_PZ": lda_c_pz, "XC_LDA_C_PW": lda_c_pw,
        "XC_LDA_C_VWN": lda_c_vwn}
def gga_x_revpbe(rho, sigma):
    """XC_GGA_X_PBE_R (revPBE, Zhang-Yang kappa=1.245)."""
    return gga_x_pbe(rho, sigma, kappa=1.245)


def gga_x_rpbe(rho, sigma):
    """XC_GGA_X_RPBE (Hammer-Hansen-Norskov)."""
    return gga_x_pbe(rho, sigma, rpbe=True)


_GGA = {"XC_GGA_X_PBE": gga_x_pbe, "XC_GGA_C_PBE": gga_c_pbe,
        "XC_GGA_X_PBE_SOL": gga_x_pbesol, "XC_GGA_C_PBE_SOL": gga_c_pbesol,
        "XC_GGA_X_PBE_R": gga_x_revpbe, "XC_GGA_X_RPBE": gga_x_rpbe}
_LDA_SPIN = {"XC_LDA_X": lda_x_spin, "XC_LDA_C_PZ": lda_c_pz_spin,
             "XC_LDA_C_PW": lda_c_pw_spin, "XC_LDA_C_VWN": lda_c_vwn_spin}


def evaluate_spin(names: list[str], ru: torch.Tensor, rd: torch.Tensor,
                  s_uu=None, s_dd=None, s_tot=None):
    """Collinear spin evaluation. Returns (eps, vu, vd, vs_uu, vs_dd, vs_tot)
    where GGA exchange uses per-spin gradients (vs_uu/vs_dd) and GGA
    correlation the total gradient (vs_tot)."""
    eps = torch.zeros_like(ru)
    vu = torch.zeros_like(ru)
    vd = torch.zeros_like(ru)
    vs_uu = torch.zeros_like(ru) if s_uu is not None else None
    vs_dd = torch.zeros_like(ru) if s_uu is not None else None
    vs_tot = torch.zeros_like(ru) if s_uu is not None else None
    n = _safe_rho(ru + rd)
    for name in names:
        if name in _LDA_SPIN:
            e, a, b = _LDA_SPIN[name](ru, rd)
            eps = eps + e
            vu = vu + a
            vd = vd + b
        elif name in ("XC_GGA_X_PBE", "XC_GGA_X_PBE_SOL",
                      "XC_GGA_X_PBE_R", "XC_GGA_X_RPBE"):
            mu = _PBESOL_MU if name == "XC_GGA_X_PBE_SOL" else _PBE_MU
            kap = 1.245 if name == "XC_GGA_X_PBE_R" else _PBE_KAPPA
            e, a, b, su, sd = gga_x_pbe_spin(
                ru, rd, s_uu, s_dd, mu=mu, kappa=kap,
                rpbe=name == "XC_GGA_X_RPBE")
            eps = eps + e
            vu = vu + a
            vd = vd + b
            vs_uu = vs_uu + su
            vs_dd = vs_dd + sd
        elif name in ("XC_GGA_C_PBE", "XC_GGA_C_PBE_SOL"):
            bt = _PBE_BETA if name == "XC_GGA_C_PBE" else _PBESOL_BETA
            e, a, b, st = gga_c_pbe_spin(ru, rd, s_tot, beta_pbe=bt)
            eps = eps + e
            vu = vu + a
            vd = vd + b
            vs_tot = vs_tot + st
        else:
            raise ValueError(f"unsupported spin xc functional: {name}")
    return eps, vu, vd, vs_uu, vs_dd, vs_tot


def is_gga(names: list[str]) -> bool:
    return any(n in _GGA for n in names)


def evaluate(names: list[str], rho: torch.Tensor, sigma: torch.Tensor | None = None):
    """Sum of functionals. Returns (eps, vrho, vsigma|None)."""
    eps = torch.zeros_like(rho)
    vrho = torch.zeros_like(rho)
    vsigma = torch.zeros_like(rho) if is_gga(names) else None
    for n in names:
        if n in _LDA:
            e, v = _LDA[n](rho)
            eps = eps + e
            vrho = vrho + v
        elif n in _GGA:
            e, v, vs = _GGA[n](rho, sigma)
            eps = eps + e
            vrho = vrho + v
            vsigma = vsigma + vs
        else:
            raise ValueError(f"unsupported xc functional: {n}")
    return eps, vrho, vsigma
