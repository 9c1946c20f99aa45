"""Action distributions.

Parity surface with /root/reference/stoix/networks/distributions.py
(AffineTanhTransformedDistribution :19-94, ClippedBeta :97-113,
DiscreteValuedTfpDistribution :116-208, MultiDiscreteActionDistribution
:211-243) plus the distrax primitives the reference uses (Categorical,
EpsilonGreedy, MultivariateNormalDiag). Thin torch implementations with an
explicit ``mode()`` (greedy eval acting, reference evaluator.py:48-67) and
reparameterised ``rsample`` where the algorithm needs pathwise gradients
(SAC).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

Tensor = torch.Tensor


class Distribution:
    def sample(self, generator: Optional[torch.Generator] = None) -> Tensor:
        raise NotImplementedError

    def log_prob(self, value: Tensor) -> Tensor:
        raise NotImplementedError

    def entropy(self) -> Tensor:
        raise NotImplementedError

    def mode(self) -> Tensor:
        raise NotImplementedError


class Categorical(Distribution):
    def __init__(self, logits: Tensor):
        self.logits = logits - logits.logsumexp(dim=-1, keepdim=True)

    def sample(self, generator=None) -> Tensor:
        # Gumbel-max: graph-capturable, generator-controllable
        u = torch.rand(self.logits.shape, device=self.logits.device, generator=generator)
        g = -torch.log((-torch.log(u.clamp(min=1e-10))).clamp(min=1e-10))
        return (self.logits + g).argmax(dim=-1)

    def log_prob(self, value: Tensor) -> Tensor:
        return self.logits.gather(-1, value.long().unsqueeze(-1)).squeeze(-1)

    def entropy(self) -> Tensor:
        p = self.logits.exp()
        return -(p * self.logits).sum(-1)

    def mode(self) -> Tensor:
        return self.logits.argmax(dim=-1)

    def kl_divergence(self, other: "Categorical") -> Tensor:
        p = self.logits.exp()
        return (p * (self.logits - other.logits)).sum(-1)

    @property
    def probs(self) -> Tensor:
        return self.logits.exp()


class EpsilonGreedy(Distribution):
    """Epsilon-greedy over Q-values (the reference bakes this into
    DiscreteQNetworkHead, heads.py:202-217)."""

    def __init__(self, preferences: Tensor, epsilon: float):
        self.preferences = preferences
        self.epsilon = epsilon

    def _probs(self) -> Tensor:
        a = self.preferences.argmax(dim=-1, keepdim=True)
        n = self.preferences.shape[-1]
        greedy = torch.zeros_like(self.preferences).scatter_(-1, a, 1.0)
        return (1 - self.epsilon) * greedy + self.epsilon / n

    def sample(self, generator=None) -> Tensor:
        p = self._probs()
        u = torch.rand(p.shape, device=p.device, generator=generator)
        g = -torch.log((-torch.log(u.clamp(min=1e-10))).clamp(min=1e-10))
        return (p.clamp(min=1e-10).log() + g).argmax(dim=-1)

    def log_prob(self, value: Tensor) -> Tensor:
        p = self._probs()
        return p.gather(-1, value.long().unsqueeze(-1)).squeeze(-1).clamp(min=1e-10).log()

    def entropy(self) -> Tensor:
        p = self._probs()
        return -(p * p.clamp(min=1e-10).log()).sum(-1)

    def mode(self) -> Tensor:
        return self.preferences.argmax(dim=-1)


class Normal(Distribution):
    def __init__(self, loc: Tensor, scale: Tensor):
        self.loc = loc
        self.scale = scale

    def sample(self, generator=None) -> Tensor:
        eps = torch.randn(self.loc.shape, device=self.loc.device, generator=generator)
        return self.loc + self.scale * eps

    def rsample(self, generator=None) -> Tensor:
        eps = torch.randn(self.loc.shape, device=self.loc.device, generator=generator)
        return self.loc + self.scale * eps

    def log_prob(self, value: Tensor) -> Tensor:
        var = self.scale**2
        return -((value - self.loc) ** 2) / (2 * var) - self.scale.log() - 0.5 * math.log(2 * math.pi)

    def entropy(self) -> Tensor:
        return 0.5 + 0.5 * math.log(2 * math.pi) + self.scale.log()

    def mode(self) -> Tensor:
        return self.loc


class MultivariateNormalDiag(Distribution):
    """Diagonal Gaussian; log_prob/entropy summed over the event dim
    (reference heads.py:101-114)."""

    def __init__(self, loc: Tensor, scale_diag: Tensor):
        self._n = Normal(loc, scale_diag)

    def sample(self, generator=None) -> Tensor:
        return self._n.sample(generator)

    def rsample(self, generator=None) -> Tensor:
        return self._n.rsample(generator)

    def log_prob(self, value: Tensor) -> Tensor:
        return self._n.log_prob(value).sum(-1)

    def entropy(self) -> Tensor:
        return self._n.entropy().sum(-1)

    def mode(self) -> Tensor:
        return self._n.loc


class AffineTanhTransformedDistribution(Distribution):
    """tanh-squashed Gaussian affinely mapped to [minimum, maximum], with the
    boundary log-prob clipped for numerical stability (reference
    distributions.py:19-94). Event dims summed (independent per-dim base)."""

    def __init__(
        self,
        loc: Tensor,
        scale: Tensor,
        minimum: float | Tensor,
        maximum: float | Tensor,
        epsilon: float = 1e-3,
    ):
        self.base = Normal(loc, scale)
        # Keep python-float bounds as scalars: creating device tensors here
        # would issue a pageable H2D copy on every forward, which is illegal
        # inside hip-graph capture (and needless work outside it).
        if isinstance(minimum, torch.Tensor) or isinstance(maximum, torch.Tensor):
            mn = torch.as_tensor(minimum, dtype=loc.dtype, device=loc.device)
            mx = torch.as_tensor(maximum, dtype=loc.dtype, device=loc.device)
            self.scale_affine = (mx - mn) / 2.0
            self.shift_affine = (mx + mn) / 2.0
            self._log_scale = torch.log(self.scale_affine)
        else:
            self.scale_affine = (float(maximum) - float(minimum)) / 2.0
            self.shift_affine = (float(maximum) + float(minimum)) / 2.0
            self._log_scale = math.log(self.scale_affine)
        self.eps = epsilon

    def _forward(self, u: Tensor) -> Tensor:
        return torch.tanh(u) * self.scale_affine + self.shift_affine

    def _inverse(self, a: Tensor) -> Tensor:
        y = (a - self.shift_affine) / self.scale_affine
        y = y.clamp(-1.0 + self.eps, 1.0 - self.eps)
        return torch.atanh(y)

    def sample(self, generator=None) -> Tensor:
        return self._forward(self.base.sample(generator))

    def rsample(self, generator=None) -> Tensor:
        return self._forward(self.base.rsample(generator))

    def sample_and_log_prob(self, generator=None) -> Tuple[Tensor, Tensor]:
        u = self.base.rsample(generator)
        a = self._forward(u)
        lp = self._log_prob_from_u(u)
        return a, lp

    def _log_prob_from_u(self, u: Tensor) -> Tensor:
        # log |d tanh/du| = log(1 - tanh(u)^2) = 2*(log2 - u - softplus(-2u))
        log_det = 2.0 * (math.log(2.0) - u - F.softplus(-2.0 * u)) + self._log_scale
        return (self.base.log_prob(u) - log_det).sum(-1)

    def log_prob(self, value: Tensor) -> Tensor:
        """Reference boundary semantics (distributions.py:55-80): inside
        the clipping range use the change-of-variables density; AT/BEYOND
        the range use the log of the AVERAGE density of the probability
        MASS squashed into the boundary strip —
        ``log_cdf(u_min)/log_sf(u_max) - log(eps)``. The mass form is
        bounded (~0 for a saturated base), unlike the clamped-atanh
        density (which reaches -900s when |loc| drifts and detonates the
        PPO ratio: exp(new - (-900)) = inf; found by the swing-up stress
        test)."""
        # thresholds in ACTION units: [min+eps, max-eps] (reference :58-59)
        lo = self.shift_affine - self.scale_affine
        hi = self.shift_affine + self.scale_affine
        min_th = lo + self.eps
        max_th = hi - self.eps
        y_min = (min_th - self.shift_affine) / self.scale_affine
        y_max = (max_th - self.shift_affine) / self.scale_affine
        if isinstance(y_min, Tensor):
            u_min = torch.atanh(y_min)
            u_max = torch.atanh(y_max)
        else:
            u_min = math.atanh(y_min)
            u_max = math.atanh(y_max)
        loc, scale = self.base.loc, self.base.scale
        log_eps = (
            torch.log(self.eps) if isinstance(self.eps, Tensor) else math.log(self.eps)
        )
        # log_cdf / log_survival of the base Normal at the thresholds
        log_prob_left = torch.special.log_ndtr((u_min - loc) / scale) - log_eps
        log_prob_right = torch.special.log_ndtr((loc - u_max) / scale) - log_eps
        v = value.clamp(min_th, max_th)
        u = torch.atanh(((v - self.shift_affine) / self.scale_affine).clamp(-1.0 + 1e-6, 1.0 - 1e-6))
        log_det = 2.0 * (math.log(2.0) - u - F.softplus(-2.0 * u)) + self._log_scale
        inside = self.base.log_prob(u) - log_det
        per_dim = torch.where(
            v <= min_th, log_prob_left,
            torch.where(v >= max_th, log_prob_right, inside),
        )
        return per_dim.sum(-1)

    def entropy(self, num_samples: int = 1, generator=None) -> Tensor:
        """Monte-Carlo entropy estimate (exact entropy of a tanh-Gaussian has
        no closed form; the reference estimates it the same way)."""
        u = self.base.rsample(generator)
        return -self._log_prob_from_u(u)

    def mode(self) -> Tensor:
        return self._forward(self.base.loc)


class ClippedBeta(Distribution):
    """Beta distribution with samples clipped away from {0,1}
    (reference distributions.py:97-113), affinely mapped to [minimum, maximum]."""

    def __init__(self, alpha: Tensor, beta: Tensor, minimum: float = 0.0, maximum: float = 1.0, epsilon: float = 1e-6):
        self.alpha = alpha
        self.beta = beta
        self.mn = minimum
        self.mx = maximum
        self.eps = epsilon
        self._d = torch.distributions.Beta(alpha, beta)

    def sample(self, generator=None) -> Tensor:
        x = self._d.rsample().clamp(self.eps, 1 - self.eps)
        return x * (self.mx - self.mn) + self.mn

    rsample = sample

    def _to_unit(self, value: Tensor) -> Tensor:
        return ((value - self.mn) / (self.mx - self.mn)).clamp(self.eps, 1 - self.eps)

    def log_prob(self, value: Tensor) -> Tensor:
        x = self._to_unit(value)
        return (self._d.log_prob(x) - math.log(self.mx - self.mn)).sum(-1)

    def entropy(self) -> Tensor:
        return (self._d.entropy() + math.log(self.mx - self.mn)).sum(-1)

    def mode(self) -> Tensor:
        a, b = self.alpha, self.beta
        m = torch.where(
            (a > 1) & (b > 1),
            (a - 1) / (a + b - 2).clamp(min=1e-6),
            (a >= b).to(a.dtype),
        )
        return m * (self.mx - self.mn) + self.mn


class DiscreteValuedDistribution(Categorical):
    """Categorical over a fixed support of scalar values
    (reference distributions.py:116-208; used by DiscreteValuedTfpHead /
    D4PG critic)."""

    def __init__(self, logits: Tensor, values: Tensor):
        super().__init__(logits)
        self.values = values  # [num_atoms]

    def mean(self) -> Tensor:
        return (self.probs * self.values).sum(-1)

    def mode(self) -> Tensor:
        idx = self.logits.argmax(dim=-1)
        return self.values[idx]


class MultiDiscreteDistribution(Distribution):
    """Independent categoricals per action dimension
    (reference distributions.py:211-243)."""

    def __init__(self, flat_logits: Tensor, num_values: list):
        self.dists = []
        off = 0
        for n in num_values:
            self.dists.append(Categorical(flat_logits[..., off : off + n]))
            off += n

    def sample(self, generator=None) -> Tensor:
        return torch.stack([d.sample(generator) for d in self.dists], dim=-1)

    def log_prob(self, value: Tensor) -> Tensor:
        return sum(d.log_prob(value[..., i]) for i, d in enumerate(self.dists))

    def entropy(self) -> Tensor:
        return sum(d.entropy() for d in self.dists)

    def mode(self) -> Tensor:
        return torch.stack([d.mode() for d in self.dists], dim=-1)
