"""Observation likelihoods for the Laplace-approximation estimators.

R&W Algorithms 3.1 (posterior mode) and 5.1 (approximate evidence +
gradient) only touch the likelihood through four quantities at the latent
f: log p(y|f) and its first three derivatives in f.  The reference
hardcodes the logistic link (``classification/GaussianProcessClassifier.scala:74-129``);
factoring it out adds count-data (Poisson) GP regression for free — a
model family the reference does not have.

All methods are elementwise over [E, k] tensors.
"""

from __future__ import annotations

import torch


class Likelihood:
    """log p(y | f) and its derivatives w.r.t. f (elementwise)."""

    def log_lik(self, f: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def d1(self, f: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """d/df log p — the Newton gradient term."""
        raise NotImplementedError

    def w(self, f: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """-d^2/df^2 log p  (must be positive: log-concave likelihoods)."""
        raise NotImplementedError

    def d3(self, f: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """dW/df = -d^3/df^3 log p — the Algorithm 5.1 s2 ingredient.

        NOTE the sign convention: the evidence pipeline (and the
        reference, ``classification/GaussianProcessClassifier.scala:118``)
        consume the derivative of the Newton WEIGHT W = -d2 log p, not the
        raw third log-likelihood derivative."""
        raise NotImplementedError

    def d3_evidence(self, f_eval: torch.Tensor, f_final: torch.Tensor,
                    y: torch.Tensor) -> torch.Tensor:
        """d3 as the evidence pass consumes it.  Default: at the final
        latent.  The logistic override reproduces the reference's exact
        state bookkeeping, which mixes the last Newton EVALUATION point
        (for pi) with the final latent (for exp(-f)) —
        ``classification/GaussianProcessClassifier.scala:113-128`` computes
        5.1 from loop-scope variables after the in-place f update."""
        return self.d3(f_final, y)

    def validate_targets(self, y: torch.Tensor) -> bool:
        return True


class LogisticLikelihood(Likelihood):
    """Bernoulli with logistic link, y in {0, 1} (the reference's GPC).

    The expressions reproduce the existing code paths bit-for-bit (e.g. d3
    keeps the pi^2 * exp(-f) form rather than the algebraically equal
    pi (1-pi) (1-2 pi))."""

    def log_lik(self, f, y):
        return torch.nn.functional.logsigmoid((2.0 * y - 1.0) * f)

    def d1(self, f, y):
        return y - torch.sigmoid(f)

    def w(self, f, y):
        pi = torch.sigmoid(f)
        return pi * (1.0 - pi)

    def d3(self, f, y):
        pi = torch.sigmoid(f)
        return -(2.0 * pi - 1.0) * pi * pi * torch.exp(-f)

    def d3_evidence(self, f_eval, f_final, y):
        pi = torch.sigmoid(f_eval)
        return -(2.0 * pi - 1.0) * pi * pi * torch.exp(-f_final)

    def validate_targets(self, y):
        return bool(torch.isin(y, torch.tensor([0.0, 1.0], dtype=y.dtype,
                                               device=y.device)).all())


class ProbitLikelihood(Likelihood):
    """Bernoulli with probit link, y in {0, 1}:  p(y=1|f) = Phi(f).

    With t = 2y - 1 and z = t f (R&W §3.9 probit expressions):
      log p = log Phi(z)
      d1    = t h(z),            h(z) = phi(z)/Phi(z)  (inverse Mills ratio)
      W     = h(z) (h(z) + z)        in (0, 1), log-concave
      dW/df = t [ h + (z + 2h) h' ],  h' = -(z h + h^2)
    Everything is computed through ``log_ndtr`` so deep tails (|f| large)
    never underflow Phi."""

    _LOG_SQRT_2PI = 0.9189385332046727    # log sqrt(2 pi)

    def _tzh(self, f, y):
        t = 2.0 * y - 1.0
        z = t * f
        log_phi = -0.5 * z * z - self._LOG_SQRT_2PI
        h = torch.exp(log_phi - torch.special.log_ndtr(z))
        return t, z, h

    def log_lik(self, f, y):
        return torch.special.log_ndtr((2.0 * y - 1.0) * f)

    def d1(self, f, y):
        t, _, h = self._tzh(f, y)
        return t * h

    def w(self, f, y):
        _, z, h = self._tzh(f, y)
        return h * (h + z)

    def d3(self, f, y):
        t, z, h = self._tzh(f, y)
        hp = -(z * h + h * h)
        return t * (h + (z + 2.0 * h) * hp)

    def validate_targets(self, y):
        return bool(torch.isin(y, torch.tensor([0.0, 1.0], dtype=y.dtype,
                                               device=y.device)).all())


class PoissonLikelihood(Likelihood):
    """Poisson counts with log link: y | f ~ Poisson(exp(f)), y in {0,1,2,..}.

    log p = y f - e^f - log y!;  d1 = y - e^f;  W = e^f;  dW/df = e^f.
    ``fmax`` clips the latent inside exp() so a wild Newton candidate
    cannot overflow fp32/fp64 (the step-halving then rejects it)."""

    def __init__(self, fmax: float = 30.0):
        self.fmax = float(fmax)

    def _lam(self, f):
        return torch.exp(f.clamp(max=self.fmax))

    def log_lik(self, f, y):
        return y * f - self._lam(f) - torch.lgamma(y + 1.0)

    def d1(self, f, y):
        return y - self._lam(f)

    def w(self, f, y):
        return self._lam(f)

    def d3(self, f, y):
        return self._lam(f)            # dW/df = d(e^f)/df = e^f

    def validate_targets(self, y):
        return bool(((y >= 0) & (y == y.round())).all())
