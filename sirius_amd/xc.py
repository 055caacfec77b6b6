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


# -- PBE ---------------------------------------------------------------------

_PBE_KAPPA = 0.8040
_PBE_MU = 0.2195149727645171
_PBE_BETA = 0.06672455060314922
_PBE_GAMMA = (1.0 - math.log(2.0)) / math.pi**2


def gga_x_pbe(rho: torch.Tensor, sigma: torch.Tensor):
    """XC_GGA_X_PBE, unpolarized. Returns (eps, vrho, vsigma)."""
    rho = _safe_rho(rho)
    sigma = torch.clamp(sigma, min=1e-40)
    kf = (3.0 * math.pi**2 * rho) ** _THIRD
    # s^2 = sigma / (2 kf rho)^2
    s2 = sigma / (2.0 * kf * rho) ** 2
    kappa, mu = _PBE_KAPPA, _PBE_MU
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


def gga_c_pbe(rho: torch.Tensor, sigma: torch.Tensor):
    """XC_GGA_C_PBE, unpolarized. Returns (eps, vrho, vsigma)."""
    rho = _safe_rho(rho)
    sigma = torch.clamp(sigma, min=1e-40)
    rs = (3.0 / (4.0 * math.pi * rho)) ** _THIRD
    ec, dec = _pw_g(rs, 0.0310907, 0.21370, 7.5957, 3.5876, 1.6382, 0.49294)
    beta, gamma = _PBE_BETA, _PBE_GAMMA
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


_LDA = {"XC_LDA_X": lda_x, "XC_LDA_C_PZ": lda_c_pz, "XC_LDA_C_PW": lda_c_pw}
_GGA = {"XC_GGA_X_PBE": gga_x_pbe, "XC_GGA_C_PBE": gga_c_pbe}


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
