"""Smearing functions and Fermi-level search.

Reference behavior: src/dft/smearing.{hpp,cpp}:29-45 (gaussian,
fermi_dirac, cold, methfessel_paxton) and
K_point_set::find_band_occupancies (src/k_point/k_point_set.cpp:286) —
bisection on the chemical potential so that sum of occupancies equals the
number of valence electrons. Argument convention: x = mu - eps.
"""

from __future__ import annotations

import math

import numpy as np
from scipy.special import erf


def occupancy(kind: str, x: np.ndarray, w: float) -> np.ndarray:
    t = x / w
    if kind == "gaussian":
        return 0.5 * (1.0 + erf(t))
    if kind == "fermi_dirac":
        return 1.0 - 1.0 / (1.0 + np.exp(np.clip(t, -200, 200)))
    if kind == "cold":
        z = t - 1.0 / math.sqrt(2.0)
        return 0.5 * (1.0 + erf(z)) + np.where(z * z > 200, 0.0,
                                               np.exp(-np.clip(z * z, None, 200)) / math.sqrt(2 * math.pi))
    if kind == "methfessel_paxton":  # order 1
        # Reference: smearing.cpp methfessel_paxton::occupancy (z = -x/w):
        # f = 0.5(1-erf(z)) + A1 H1(z) exp(-z^2), A1 = -1/(4 sqrt(pi)),
        # H1(z) = 2z.  With t = x/w = -z this is
        # f = 0.5(1+erf(t)) + t exp(-t^2)/(2 sqrt(pi)).
        g = 0.5 * (1.0 + erf(t))
        f = g + t * np.exp(-np.clip(t * t, None, 200)) / (2.0 * math.sqrt(math.pi))
        return np.where(f < 1e-30, 0.0, f)
    raise ValueError(f"unknown smearing: {kind}")


def entropy(kind: str, x: np.ndarray, w: float) -> np.ndarray:
    """Per-state entropy term (reference smearing.cpp:36-125 conventions;
    the total-energy 'entropy_sum' is Σ_k w_k max_occ Σ_n S(mu-eps))."""
    t = x / w
    if kind == "gaussian":
        return -np.exp(-np.clip(t * t, None, 200)) * w / (2.0 * math.sqrt(math.pi))
    if kind == "fermi_dirac":
        f = 1.0 / (1.0 + np.exp(np.clip(t, -200, 200)))
        f = np.clip(f, 1e-300, 1 - 1e-16)
        s = w * ((1 - f) * np.log1p(-f) + f * np.log(f))
        return np.where(np.abs(f - 1.0) * np.abs(f) < 1e-16, 0.0, s)
    if kind == "cold":
        z = t - 1.0 / math.sqrt(2.0)
        z2 = np.clip(z * z, None, 200)
        return -np.exp(-z2) * (w - math.sqrt(2.0) * w * t) / (2.0 * math.sqrt(math.pi))
    if kind == "methfessel_paxton":
        # Exact port of the reference series (smearing.cpp
        # methfessel_paxton::entropy, n=1 — QE w1gauss form).  NOTE: the
        # reference does NOT multiply by the width here (unlike its
        # gaussian/FD entropies); we reproduce that behavior for parity.
        arg = np.clip(t * t, None, 200.0)
        S = -0.5 * np.exp(-arg) / math.sqrt(math.pi)
        hd = np.zeros_like(t)
        hp = np.exp(-arg)
        ni = 0
        a = 1.0 / math.sqrt(math.pi)
        for i in range(1, 2):  # n = 1
            hd = 2.0 * t * hp - 2.0 * ni * hd
            ni += 1
            hpm1 = hp
            hp = 2.0 * t * hd - 2.0 * ni * hp
            ni += 1
            a = -a / (i + 4.0)
            S = S - a * (0.5 * hp + ni * hpm1)
        return S
    raise ValueError(f"unknown smearing: {kind}")


def find_fermi(eigvals: np.ndarray, weights: np.ndarray, n_electrons: float,
               kind: str, width: float, max_occ: float) -> float:
    """Bisection for mu: Σ_k w_k max_occ Σ_n f((mu-e_nk)/w) = n_electrons.

    eigvals [nk, nbnd] (or [nk*nspin, nbnd] with per-row weights).
    """
    lo = eigvals.min() - 10.0 * width - 1.0
    hi = eigvals.max() + 10.0 * width + 1.0

    def count(mu):
        f = occupancy(kind, mu - eigvals, width)
        return float((weights[:, None] * f).sum() * max_occ)

    if count(hi) < n_electrons - 1e-10:
        raise RuntimeError("not enough bands to hold all electrons")
    for _ in range(200):
        mu = 0.5 * (lo + hi)
        c = count(mu)
        if abs(c - n_electrons) < 1e-13:
            break
        if c > n_electrons:
            hi = mu
        else:
            lo = mu
    return 0.5 * (lo + hi)
