"""Density mixers.

Reference behavior: src/mixer/mixer.hpp:275-420 (variadic Mixer base:
set_input → residual = input − last_output → rms → mix_impl),
anderson_mixer.hpp:54 (Anderson with history Gram least squares and beta
rescaling), broyden2_mixer.hpp:88, linear.

Here a mixed "function" is a dict of named torch tensors (complex or
real); each name carries an inner-product callable and all components sum
into the mixing Gram matrix exactly as the reference's multi-function
mixer does. RMS = sqrt(Σ_components inner(res,res)/global_size), matching
mixer.hpp:update_rms with normalize=true.
"""

from __future__ import annotations

import torch


def dot_re(x: torch.Tensor, y: torch.Tensor) -> float:
    """Re <x|y> over all elements."""
    if torch.is_complex(x):
        return float(torch.vdot(x.reshape(-1), y.reshape(-1)).real)
    return float(torch.dot(x.reshape(-1), y.reshape(-1)))


class Component:
    """One mixed quantity: inner product weight and normalization."""

    def __init__(self, name: str, inner=None, global_size: float = 1.0):
        self.name = name
        self.inner = inner or dot_re
        self.global_size = global_size


class Mixer:
    def __init__(self, components: list[Component], max_history: int = 8,
                 beta: float = 0.7, beta0: float = 0.15,
                 beta_scaling_factor: float = 1.0):
        self.components = {c.name: c for c in components}
        self.max_history = max_history
        self.beta = beta
        self.beta0 = beta0
        self.beta_scaling_factor = beta_scaling_factor
        self.step = 0
        self.output_history: list[dict | None] = [None] * max_history
        self.residual_history: list[dict | None] = [None] * max_history
        self.rmse_history = [0.0] * max_history
        self.input: dict | None = None

    # -- helpers over dict-of-tensors ------------------------------------

    def _copy(self, x: dict) -> dict:
        return {k: v.clone() for k, v in x.items()}

    def _axpy(self, a: float, x: dict, y: dict):
        for k in y:
            y[k] += a * x[k]

    def _scale(self, a: float, x: dict):
        for k in x:
            x[k] *= a

    def _inner(self, x: dict, y: dict, normalize: bool) -> float:
        s = 0.0
        for k in x:
            c = self.components[k]
            v = c.inner(x[k], y[k])
            if normalize:
                v /= c.global_size
            s += v
        return s

    def _idx(self, step: int) -> int:
        return step % self.max_history

    # -- public ----------------------------------------------------------

    def initialize(self, init_value: dict):
        self.output_history[0] = self._copy(init_value)
        self.input = self._copy(init_value)

    def set_input(self, value: dict):
        self.input = self._copy(value)

    def get_output(self) -> dict:
        return self._copy(self.output_history[self._idx(self.step)])

    def mix(self, rms_min: float = 1e-16) -> float:
        idx = self._idx(self.step)
        res = self._copy(self.input)
        self._axpy(-1.0, self.output_history[idx], res)
        self.residual_history[idx] = res
        rmse = self._inner(res, res, normalize=True) ** 0.5
        self.rmse_history[idx] = rmse
        if rmse < rms_min:
            return rmse
        self.mix_impl()
        self.step += 1
        return rmse


class Linear(Mixer):
    def mix_impl(self):
        idx = self._idx(self.step)
        nxt = self._copy(self.output_history[idx])
        self._axpy(self.beta, self.residual_history[idx], nxt)
        self.output_history[self._idx(self.step + 1)] = nxt


class Anderson(Mixer):
    """Anderson mixing (reference: anderson_mixer.hpp:77-168)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.history_size = 0
        n = self.max_history - 1
        self.S = torch.zeros(n, n, dtype=torch.float64)

    def mix_impl(self):
        idx = self._idx(self.step)
        idx_prev = self._idx(self.step - 1)
        hs = self.history_size

        # adaptive beta rescaling
        if self.step > self.max_history:
            rmse_avg = sum(self.rmse_history) / len(self.rmse_history)
            if self.rmse_history[idx] > rmse_avg:
                self.beta = max(self.beta0, self.beta * self.beta_scaling_factor)

        nxt = self._copy(self.output_history[idx])
        self._axpy(self.beta, self.residual_history[idx], nxt)

        if hs > 0:
            # residual[prev] <- residual[step] - residual[prev]; same for output
            self._scale(-1.0, self.residual_history[idx_prev])
            self._axpy(1.0, self.residual_history[idx], self.residual_history[idx_prev])
            self._scale(-1.0, self.output_history[idx_prev])
            self._axpy(1.0, self.output_history[idx], self.output_history[idx_prev])

            for i in range(hs):
                j = self._idx(self.step - i - 1)
                v = self._inner(self.residual_history[j], self.residual_history[idx_prev],
                                normalize=False)
                self.S[hs - 1, hs - i - 1] = v
                self.S[hs - i - 1, hs - 1] = v

            h = torch.zeros(hs, dtype=torch.float64)
            for i in range(1, hs + 1):
                j = self._idx(self.step - i)
                h[hs - i] = self._inner(self.residual_history[j], self.residual_history[idx],
                                        normalize=False)
            try:
                hsol = torch.linalg.solve(self.S[:hs, :hs], h)
                for i in range(1, hs + 1):
                    j = self._idx(self.step - i)
                    self._axpy(-self.beta * float(hsol[hs - i]), self.residual_history[j], nxt)
                    self._axpy(-float(hsol[hs - i]), self.output_history[j], nxt)
            except Exception:
                self.history_size = 0
                hs = 0

        if self.history_size == self.max_history - 1:
            self.S[: hs - 1, : hs - 1] = self.S[1:hs, 1:hs].clone()

        self.output_history[self._idx(self.step + 1)] = nxt
        self.history_size = min(self.history_size + 1, self.max_history - 1)


class Broyden2(Mixer):
    """Second Broyden method (rank-one product expansion over the residual
    history; replicates broyden2_mixer.hpp:105-177 including the Gram
    bookkeeping and the full-history shift)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        n = self.max_history
        self.S = torch.zeros(n, n, dtype=torch.float64)

    def mix_impl(self):
        idx = self._idx(self.step)
        n = min(self.step, self.max_history - 1)
        for i in range(n + 1):
            j = self._idx(self.step - i)
            v = self._inner(self.residual_history[j],
                            self.residual_history[idx], normalize=False)
            self.S[n - i, n] = v
            self.S[n, n - i] = v
        gamma = torch.zeros(self.max_history, dtype=torch.float64)
        S = self.S
        for i in range(1, n + 1):
            g = S[n - i, n] - S[n - i + 1, n]
            for j in range(1, i):
                g += float((-S[n - i + 1, n - j + 1] + S[n - i + 1, n - j]
                            + S[n - i, n - j + 1] - S[n - i, n - j])
                           * gamma[n - j])
            denom = (S[n - i + 1, n - i + 1] - S[n - i + 1, n - i]
                     - S[n - i, n - i + 1] + S[n - i, n - i])
            gamma[n - i] = g / denom

        # reference seeds the accumulator with the CURRENT output
        # (mixer copy(output_history[idx], input_), broyden2_mixer.hpp:142)
        nxt = self._copy(self.output_history[idx])
        if n > 0:
            j = self._idx(self.step - n)
            self._axpy(-self.beta * float(gamma[0]),
                       self.residual_history[j], nxt)
            self._axpy(-float(gamma[0]), self.output_history[j], nxt)
            for i in range(1, n):
                coeff = float(gamma[n - i - 1] - gamma[n - i])
                j = self._idx(self.step - i)
                self._axpy(self.beta * coeff, self.residual_history[j], nxt)
                self._axpy(coeff, self.output_history[j], nxt)
            j = idx
            self._axpy(self.beta * (float(gamma[n - 1]) + 1.0),
                       self.residual_history[j], nxt)
            self._axpy(float(gamma[n - 1]), self.output_history[j], nxt)
        else:
            self._axpy(self.beta, self.residual_history[idx], nxt)
        self.output_history[self._idx(self.step + 1)] = nxt
        if n == self.max_history - 1:
            self.S[:n, :n] = self.S[1:n + 1, 1:n + 1].clone()


class AndersonStable(Mixer):
    """Anderson with an orthonormal (QR) residual-difference history
    (anderson_stable_mixer.hpp:50-222): modified Gram-Schmidt twice, the
    Anderson solve through the triangular R, and Givens-rotation eviction
    of the oldest column when the history is full."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.history_size = 0
        n = self.max_history - 1
        self.R = torch.zeros(n, n, dtype=torch.float64)

    def _rotate(self, c: float, sn: float, x: dict, y: dict):
        for k in x:
            xi = x[k].clone()
            x[k].mul_(c).add_(sn * y[k])
            y[k].mul_(c).sub_(sn * xi)

    def mix_impl(self):
        idx = self._idx(self.step)
        idx_prev = self._idx(self.step - 1)
        hs = self.history_size

        nxt = self._copy(self.output_history[idx])
        self._axpy(self.beta, self.residual_history[idx], nxt)

        if hs > 0:
            # Δf in the previous slot (f_n − f_{n−1}); same for Δx
            self._scale(-1.0, self.residual_history[idx_prev])
            self._axpy(1.0, self.residual_history[idx],
                       self.residual_history[idx_prev])
            self._scale(-1.0, self.output_history[idx_prev])
            self._axpy(1.0, self.output_history[idx],
                       self.output_history[idx_prev])
            # modified Gram-Schmidt against the existing Q columns, twice
            for _pass in range(2):
                for i in range(1, hs):
                    j = self._idx(self.step - i - 1)
                    sz = self._inner(self.residual_history[j],
                                     self.residual_history[idx_prev],
                                     normalize=False)
                    if _pass == 0:
                        self.R[hs - 1 - i, hs - 1] = sz
                    else:
                        self.R[hs - 1 - i, hs - 1] += sz
                    self._axpy(-sz, self.residual_history[j],
                               self.residual_history[idx_prev])
            nrm2 = self._inner(self.residual_history[idx_prev],
                               self.residual_history[idx_prev],
                               normalize=False)
            if nrm2 > 0:
                sz = nrm2 ** 0.5
                self.R[hs - 1, hs - 1] = sz
                self._scale(1.0 / sz, self.residual_history[idx_prev])
                # h = Qᵀ f_n ; k = R⁻¹ h
                h = torch.zeros(hs, dtype=torch.float64)
                for i in range(1, hs + 1):
                    j = self._idx(self.step - i)
                    h[hs - i] = self._inner(self.residual_history[j],
                                            self.residual_history[idx],
                                            normalize=False)
                k = h.clone()
                for j in range(hs - 1, -1, -1):
                    k[j] /= self.R[j, j]
                    for i in range(j - 1, -1, -1):
                        k[i] -= self.R[i, j] * k[j]
                for i in range(1, hs + 1):
                    j = self._idx(self.step - i)
                    self._axpy(-self.beta * float(h[hs - i]),
                               self.residual_history[j], nxt)
                    self._axpy(-float(k[hs - i]),
                               self.output_history[j], nxt)
            else:
                self.history_size = 0
                hs = 0

        if self.history_size == self.max_history - 1:
            # Givens eviction of the oldest column
            for row in range(1, hs):
                a, b = float(self.R[row - 1, row]), float(self.R[row, row])
                nrm = (a * a + b * b) ** 0.5
                c = a / nrm if nrm > 0 else 1.0
                sn = b / nrm if nrm > 0 else 0.0
                self.R[row - 1, row] = nrm
                self.R[row, row] = 0.0
                for col in range(row + 1, hs):
                    r1, r2 = float(self.R[row - 1, col]), float(self.R[row, col])
                    self.R[row - 1, col] = c * r1 + sn * r2
                    self.R[row, col] = -sn * r1 + c * r2
                i1 = self._idx(self.step - hs + row - 1)
                i2 = self._idx(self.step - hs + row)
                self._rotate(c, sn, self.residual_history[i1],
                             self.residual_history[i2])
            for i in range(1, hs):
                i1 = self._idx(self.step - i - 1)
                i2 = self._idx(self.step - i)
                self.residual_history[i1], self.residual_history[i2] = \
                    self.residual_history[i2], self.residual_history[i1]
            for col in range(hs - 1):
                for row in range(col + 1):
                    self.R[row, col] = self.R[row, col + 1]

        self.output_history[self._idx(self.step + 1)] = nxt
        self.history_size = min(self.history_size + 1, self.max_history - 1)


def make_mixer(cfg_mixer, components: list[Component]) -> Mixer:
    kind = cfg_mixer.type
    kw = dict(max_history=cfg_mixer.max_history, beta=cfg_mixer.beta,
              beta0=cfg_mixer.beta0, beta_scaling_factor=cfg_mixer.beta_scaling_factor)
    if kind == "linear":
        return Linear(components, **kw)
    if kind in ("anderson", "broyden1"):
        # broyden1 falls back to anderson (same family)
        return Anderson(components, **kw)
    if kind == "anderson_stable":
        return AndersonStable(components, **kw)
    if kind == "broyden2":
        return Broyden2(components, **kw)
    raise ValueError(f"unknown mixer type {kind}")
