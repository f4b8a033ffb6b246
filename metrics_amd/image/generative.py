"""KID, InceptionScore, MiFID, LPIPS, PerceptualPathLength.

Parity: torchmetrics ``image/{kid,inception,mifid,lpip,perceptual_path_length}.py``.
All take user-supplied networks (no downloadable weights offline); the metric
math (polynomial-kernel MMD, KL-based IS, memorization-penalized FID, path
length statistics) is complete.
"""
from __future__ import annotations

from typing import Any, Callable, List, Optional, Tuple

import torch
from torch import Tensor
from torch.nn import Module

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.image.fid import NoTrainModule, _compute_fid


def _feature_net_or_raise(name: str, net: Optional[Module]) -> NoTrainModule:
    if net is None:
        raise ModuleNotFoundError(
            f"{name} needs a feature network. The reference downloads pretrained weights at runtime, which is"
            " impossible in this offline environment — pass any nn.Module via `feature_extractor=`."
        )
    return NoTrainModule(net)


def poly_mmd(f_real: Tensor, f_fake: Tensor, degree: int = 3, gamma: Optional[float] = None, coef: float = 1.0) -> Tensor:
    """Polynomial-kernel MMD^2 between two feature sets."""
    if gamma is None:
        gamma = 1.0 / f_real.shape[1]
    k_11 = (gamma * f_real @ f_real.t() + coef) ** degree
    k_22 = (gamma * f_fake @ f_fake.t() + coef) ** degree
    k_12 = (gamma * f_real @ f_fake.t() + coef) ** degree

    m = f_real.shape[0]
    diag_11 = k_11.diagonal().sum()
    diag_22 = k_22.diagonal().sum()
    term1 = (k_11.sum() - diag_11) / (m * (m - 1))
    term2 = (k_22.sum() - diag_22) / (m * (m - 1))
    term3 = 2 * k_12.mean()
    return term1 + term2 - term3


class KernelInceptionDistance(Metric):
    """KID: polynomial-kernel MMD between real and fake features, over subsets."""

    is_differentiable = False
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    real_features: List[Tensor]
    fake_features: List[Tensor]

    def __init__(
        self,
        feature: Any = 2048,
        subsets: int = 100,
        subset_size: int = 1000,
        degree: int = 3,
        gamma: Optional[float] = None,
        coef: float = 1.0,
        reset_real_features: bool = True,
        normalize: bool = False,
        feature_extractor: Optional[Module] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if feature_extractor is None and isinstance(feature, Module):
            feature_extractor = feature
        self.inception = _feature_net_or_raise("KernelInceptionDistance", feature_extractor)
        if not (isinstance(subsets, int) and subsets > 0):
            raise ValueError("Argument `subsets` expected to be integer larger than 0")
        self.subsets = subsets
        if not (isinstance(subset_size, int) and subset_size > 0):
            raise ValueError("Argument `subset_size` expected to be integer larger than 0")
        self.subset_size = subset_size
        self.degree = degree
        self.gamma = gamma
        self.coef = coef
        self.reset_real_features = reset_real_features
        self.normalize = normalize

        self.add_state("real_features", [], dist_reduce_fx=None)
        self.add_state("fake_features", [], dist_reduce_fx=None)

    def update(self, imgs: Tensor, real: bool) -> None:
        """Extract and store features."""
        features = self.inception(imgs)
        if real:
            self.real_features.append(features)
        else:
            self.fake_features.append(features)

    def compute(self) -> Tuple[Tensor, Tensor]:
        """(mean, std) of KID over random subsets."""
        real_features = dim_zero_cat(self.real_features)
        fake_features = dim_zero_cat(self.fake_features)
        n_samples_real = real_features.shape[0]
        if n_samples_real < self.subset_size:
            raise ValueError("Argument `subset_size` should be smaller than the number of samples")
        n_samples_fake = fake_features.shape[0]
        if n_samples_fake < self.subset_size:
            raise ValueError("Argument `subset_size` should be smaller than the number of samples")

        kid_scores_ = []
        for _ in range(self.subsets):
            perm = torch.randperm(n_samples_real)
            f_real = real_features[perm[: self.subset_size]]
            perm = torch.randperm(n_samples_fake)
            f_fake = fake_features[perm[: self.subset_size]]
            o = poly_mmd(f_real, f_fake, self.degree, self.gamma, self.coef)
            kid_scores_.append(o)
        kid_scores = torch.stack(kid_scores_)
        return kid_scores.mean(), kid_scores.std(unbiased=False)

    def reset(self) -> None:
        if not self.reset_real_features:
            real_features = self.real_features
            super().reset()
            self.real_features = real_features
        else:
            super().reset()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class InceptionScore(Metric):
    """IS: exp(E_x KL(p(y|x) || p(y))) over classifier logits of generated images."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False
    plot_lower_bound: float = 0.0

    features: List[Tensor]

    def __init__(
        self,
        feature: Any = "logits_unbiased",
        splits: int = 10,
        normalize: bool = False,
        feature_extractor: Optional[Module] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if feature_extractor is None and isinstance(feature, Module):
            feature_extractor = feature
        self.inception = _feature_net_or_raise("InceptionScore", feature_extractor)
        self.splits = splits
        self.normalize = normalize
        self.add_state("features", [], dist_reduce_fx=None)

    def update(self, imgs: Tensor) -> None:
        """Extract classifier logits."""
        features = self.inception(imgs)
        self.features.append(features)

    def compute(self) -> Tuple[Tensor, Tensor]:
        """(mean, std) of the inception score over splits."""
        features = dim_zero_cat(self.features)
        # random permute the features
        idx = torch.randperm(features.shape[0])
        features = features[idx]

        prob = features.softmax(dim=1)
        log_prob = features.log_softmax(dim=1)

        prob = prob.chunk(self.splits, dim=0)
        log_prob = log_prob.chunk(self.splits, dim=0)

        mean_prob = [p.mean(dim=0, keepdim=True) for p in prob]
        kl_ = [p * (log_p - m_p.log()) for p, log_p, m_p in zip(prob, log_prob, mean_prob)]
        kl_ = [k.sum(dim=1).mean().exp() for k in kl_]
        kl = torch.stack(kl_)
        return kl.mean(), kl.std()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MemorizationInformedFrechetInceptionDistance(Metric):
    """MiFID: FID scaled by a memorization penalty based on cosine distances."""

    is_differentiable = False
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    real_features: List[Tensor]
    fake_features: List[Tensor]

    def __init__(
        self,
        feature: Any = 2048,
        reset_real_features: bool = True,
        cosine_distance_eps: float = 0.1,
        normalize: bool = False,
        feature_extractor: Optional[Module] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if feature_extractor is None and isinstance(feature, Module):
            feature_extractor = feature
        self.inception = _feature_net_or_raise("MemorizationInformedFrechetInceptionDistance", feature_extractor)
        if not isinstance(reset_real_features, bool):
            raise ValueError("Argument `reset_real_features` expected to be a bool")
        self.reset_real_features = reset_real_features
        if not (isinstance(cosine_distance_eps, float) and 1 > cosine_distance_eps > 0):
            raise ValueError("Argument `cosine_distance_eps` expected to be a float greater than 0 and less than 1")
        self.cosine_distance_eps = cosine_distance_eps
        self.normalize = normalize
        self.add_state("real_features", [], dist_reduce_fx=None)
        self.add_state("fake_features", [], dist_reduce_fx=None)

    def update(self, imgs: Tensor, real: bool) -> None:
        """Extract and store features."""
        features = self.inception(imgs)
        if real:
            self.real_features.append(features)
        else:
            self.fake_features.append(features)

    def compute(self) -> Tensor:
        """FID x memorization penalty."""
        real = dim_zero_cat(self.real_features).double()
        fake = dim_zero_cat(self.fake_features).double()

        mu1, sigma1 = real.mean(0), torch.cov(real.t())
        mu2, sigma2 = fake.mean(0), torch.cov(fake.t())
        fid = _compute_fid(mu1, sigma1, mu2, sigma2)

        # memorization distance: mean over fake of min cosine distance to real
        real_n = real / real.norm(dim=1, keepdim=True)
        fake_n = fake / fake.norm(dim=1, keepdim=True)
        d = 1 - (fake_n @ real_n.t()).abs()
        mean_min_d = d.min(dim=1).values.mean()
        m_dist = mean_min_d if mean_min_d < self.cosine_distance_eps else torch.ones_like(mean_min_d)
        return (fid / (m_dist + 1e-15)).float()

    def reset(self) -> None:
        """Reset states; cached real features survive when reset_real_features=False."""
        if not self.reset_real_features:
            value = self._defaults.pop("real_features")
            super().reset()
            self._defaults["real_features"] = value
        else:
            super().reset()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class LearnedPerceptualImagePatchSimilarity(Metric):
    """LPIPS: perceptual distance via a user-supplied comparator network.

    ``net`` must be a callable (img1, img2) -> per-sample distance (the
    reference bundles pretrained alex/vgg/squeeze nets; offline, pass yours).
    """

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    sum_scores: Tensor
    total: Tensor

    def __init__(
        self,
        net_type: str = "alex",
        reduction: str = "mean",
        normalize: bool = False,
        net: Optional[Callable] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if net is None:
            raise ModuleNotFoundError(
                "LearnedPerceptualImagePatchSimilarity needs a comparator network: pass `net=` a callable"
                " (img1, img2) -> per-sample distances. Pretrained alex/vgg weights cannot be downloaded offline."
            )
        self.net = net
        if reduction not in ("mean", "sum"):
            raise ValueError(f"Argument `reduction` must be one of 'mean'/'sum' but got {reduction}")
        self.reduction = reduction
        if not isinstance(normalize, bool):
            raise ValueError(f"Argument `normalize` should be a bool but got {normalize}")
        self.normalize = normalize
        self.add_state("sum_scores", torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("total", torch.tensor(0.0), dist_reduce_fx="sum")

    def update(self, img1: Tensor, img2: Tensor) -> None:
        """Accumulate perceptual distances."""
        loss = self.net(img1, img2).squeeze()
        self.sum_scores += loss.sum()
        self.total += img1.shape[0]

    def compute(self) -> Tensor:
        if self.reduction == "mean":
            return self.sum_scores / self.total
        return self.sum_scores

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class PerceptualPathLength(Metric):
    """PPL: LPIPS distance statistics along latent interpolations of a generator."""

    is_differentiable = False
    higher_is_better = False
    full_state_update = False

    def __init__(
        self,
        num_samples: int = 10_000,
        conditional: bool = False,
        batch_size: int = 128,
        interpolation_method: str = "lerp",
        epsilon: float = 1e-4,
        resize: Optional[int] = 64,
        lower_discard: Optional[float] = 0.01,
        upper_discard: Optional[float] = 0.99,
        sim_net: Any = "vgg",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if sim_net is None or isinstance(sim_net, str):
            # the reference's default 'vgg' downloads LPIPS weights; offline
            # that path cannot work — require a user-supplied network
            raise ModuleNotFoundError(
                "PerceptualPathLength needs a similarity network (LPIPS-style): pass `sim_net=`."
            )
        self.sim_net = sim_net
        self.num_samples = num_samples
        self.conditional = conditional
        self.batch_size = batch_size
        if interpolation_method not in ("lerp", "slerp_any", "slerp_unit"):
            raise ValueError(f"Interpolation method {interpolation_method} not supported")
        self.interpolation_method = interpolation_method
        self.epsilon = epsilon
        self.resize = resize
        self.lower_discard = lower_discard
        self.upper_discard = upper_discard
        self.add_state("distances", [], dist_reduce_fx=None)

    @staticmethod
    def _interpolate(a: Tensor, b: Tensor, t: float, method: str) -> Tensor:
        if method == "lerp":
            return a + (b - a) * t
        # slerp
        a_n = a / a.norm(dim=-1, keepdim=True)
        b_n = b / b.norm(dim=-1, keepdim=True)
        omega = torch.acos((a_n * b_n).sum(-1, keepdim=True).clamp(-1, 1))
        so = torch.sin(omega)
        out = (torch.sin((1 - t) * omega) / so) * a + (torch.sin(t * omega) / so) * b
        if method == "slerp_unit":
            out = out / out.norm(dim=-1, keepdim=True)
        return out

    def update(self, generator: Any) -> None:
        """Sample latent pairs, generate image pairs at t and t+eps, measure distances."""
        if not hasattr(generator, "sample"):
            raise NotImplementedError("The generator must expose `sample(num_samples)` returning latents")
        n_done = 0
        while n_done < self.num_samples:
            n = min(self.batch_size, self.num_samples - n_done)
            z0 = generator.sample(n)
            z1 = generator.sample(n)
            t = torch.rand(1).item()
            za = self._interpolate(z0, z1, t, self.interpolation_method)
            zb = self._interpolate(z0, z1, t + self.epsilon, self.interpolation_method)
            img_a = generator(za)
            img_b = generator(zb)
            d = self.sim_net(img_a, img_b).squeeze() / self.epsilon**2
            self.distances.append(d.reshape(-1))
            n_done += n

    def compute(self) -> Tuple[Tensor, Tensor, Tensor]:
        """(mean, std, median) of the filtered path-length distances."""
        distances = dim_zero_cat(self.distances)
        lower = torch.quantile(distances, self.lower_discard) if self.lower_discard is not None else distances.min()
        upper = torch.quantile(distances, self.upper_discard) if self.upper_discard is not None else distances.max()
        distances = distances[(distances >= lower) & (distances <= upper)]
        return distances.mean(), distances.std(), distances.median()
