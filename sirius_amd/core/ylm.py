"""Real and complex spherical harmonics.

Reference behavior: src/core/sht/sht.hpp (sf::spherical_harmonics for
R_lm/Y_lm; index lm = l(l+1)+m). Vectorized over point batches; a CDNA4
HIP kernel (reference GPU twin: spherical_harmonics.cu) takes over on
device for the LAPW muffin-tin path.
"""

from __future__ import annotations

import numpy as np
from scipy.special import lpmv


def lmmax(lmax: int) -> int:
    return (lmax + 1) ** 2


def lm_index(l: int, m: int) -> int:
    return l * l + l + m


def spherical_coords(v: np.ndarray):
    """Cartesian [N,3] -> (r, theta, phi) arrays."""
    r = np.linalg.norm(v, axis=-1)
    safe = np.where(r > 1e-12, r, 1.0)
    ct = np.clip(v[..., 2] / safe, -1.0, 1.0)
    theta = np.arccos(ct)
    phi = np.arctan2(v[..., 1], v[..., 0])
    theta = np.where(r > 1e-12, theta, 0.0)
    phi = np.where(r > 1e-12, phi, 0.0)
    return r, theta, phi


def rlm(lmax: int, theta: np.ndarray, phi: np.ndarray) -> np.ndarray:
    """Real spherical harmonics R_lm(θ,φ), shape [N, lmmax].

    Convention (matches reference sf::spherical_harmonics real branch):
      m=0:  K_l0 P_l^0(cosθ)
      m>0:  √2 K_lm cos(mφ) P_l^m(cosθ)
      m<0:  √2 K_l|m| sin(|m|φ) P_l^|m|(cosθ)
    with Condon-Shortley phase inside P_l^m (as in scipy lpmv) and
    K_lm = sqrt((2l+1)/4π (l-|m|)!/(l+|m|)!).
    """
    ct = np.cos(theta)
    n = theta.shape[0]
    out = np.empty((n, lmmax(lmax)), dtype=np.float64)
    from math import factorial, pi, sqrt

    for l in range(lmax + 1):
        for m in range(0, l + 1):
            k = sqrt((2 * l + 1) / (4 * pi) * factorial(l - m) / factorial(l + m))
            p = lpmv(m, l, ct)
            if m == 0:
                out[:, lm_index(l, 0)] = k * p
            else:
                out[:, lm_index(l, m)] = sqrt(2.0) * k * np.cos(m * phi) * p
                out[:, lm_index(l, -m)] = sqrt(2.0) * k * np.sin(m * phi) * p
    return out


def ylm(lmax: int, theta: np.ndarray, phi: np.ndarray) -> np.ndarray:
    """Complex spherical harmonics Y_lm, shape [N, lmmax] (CS phase)."""
    ct = np.cos(theta)
    n = theta.shape[0]
    out = np.empty((n, lmmax(lmax)), dtype=np.complex128)
    from math import factorial, pi, sqrt

    for l in range(lmax + 1):
        for m in range(0, l + 1):
            k = sqrt((2 * l + 1) / (4 * pi) * factorial(l - m) / factorial(l + m))
            p = lpmv(m, l, ct)
            e = np.exp(1j * m * phi)
            out[:, lm_index(l, m)] = k * p * e
            if m > 0:
                out[:, lm_index(l, -m)] = (-1) ** m * np.conj(out[:, lm_index(l, m)])
    return out


def rlm_and_cart_grad(lmax: int, gc):
    """R_lm(Ĝ) [nG, lmmax] and the true Cartesian gradient
    ∂R_lm(Ĝ)/∂G_x [nG, 3, lmmax] by central differences
    (sf::dRlm_dr analogue, divide_by_r=true semantics; rows with
    |G| < 1e-10 get zero gradient)."""
    import numpy as _np

    r = _np.linalg.norm(gc, axis=1)
    _, th, ph = spherical_coords(gc)
    rl = rlm(lmax, th, ph)
    grad = _np.zeros((len(gc), 3, rl.shape[1]))
    h = _np.maximum(1e-6 * r, 1e-9)
    ok = r > 1e-10
    for x in range(3):
        gp = gc.copy()
        gp[:, x] += h
        gm = gc.copy()
        gm[:, x] -= h
        _, tp, pp = spherical_coords(gp)
        _, tm, pm = spherical_coords(gm)
        d = (rlm(lmax, tp, pp) - rlm(lmax, tm, pm)) / (2.0 * h)[:, None]
        grad[ok, x, :] = d[ok]
    return rl, grad
