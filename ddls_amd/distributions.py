"""Sampling distributions (reference: ``ddls/distributions/``)."""
from __future__ import annotations

from typing import Dict, List, Optional, Union

import numpy as np


class Distribution:
    def sample(self, size=None, replace=True):
        raise NotImplementedError


class Fixed(Distribution):
    """Reference ``distributions/fixed.py:7``."""

    def __init__(self, val):
        self.val = val

    def sample(self, size=None, replace=True):
        return self.val


class Uniform(Distribution):
    """Uniform in [min_val, max_val], optionally rounded to ``decimals``
    (reference ``distributions/uniform.py:7``)."""

    def __init__(self, min_val, max_val, decimals: Optional[int] = None):
        self.min_val, self.max_val, self.decimals = min_val, max_val, decimals

    def sample(self, size=None, replace=True):
        v = np.random.uniform(low=self.min_val, high=self.max_val, size=size)
        if self.decimals is not None:
            v = np.around(v, decimals=self.decimals)
        if size is None:
            return float(v)
        return v


class Exponential(Distribution):
    """Exponential interarrival times -> Poisson arrival process
    (BASELINE.json configs[4])."""

    def __init__(self, mean: float):
        self.mean = mean

    def sample(self, size=None, replace=True):
        v = np.random.exponential(scale=self.mean, size=size)
        if size is None:
            return float(v)
        return v


class ProbabilityMassFunction(Distribution):
    """Discrete PMF mapping value -> probability
    (reference ``distributions/probability_mass_function.py:7``)."""

    def __init__(self, probability_mass_function: Dict[float, float]):
        self.vals = np.array([float(k) for k in probability_mass_function.keys()])
        probs = np.array([float(v) for v in probability_mass_function.values()])
        self.probs = probs / probs.sum()

    def sample(self, size=None, replace=True):
        v = np.random.choice(self.vals, p=self.probs, size=size, replace=replace)
        if size is None:
            return float(v)
        return v


class CustomSkewNorm(Distribution):
    """Skew-normal shaped PMF on [min_val, max_val]
    (reference ``distributions/custom_skew_norm.py:11``)."""

    def __init__(self, skewness, min_val, max_val, decimals: int = 10,
                 accuracy_factor: int = 10, num_bins: int = 1000,
                 cutoff_mode: str = "resample"):
        from scipy.stats import skewnorm
        self.min_val, self.max_val = min_val, max_val
        if decimals > 0:
            interval = 1 / (10 ** decimals)
        elif decimals < 0:
            interval = 10 ** abs(decimals)
        else:
            interval = 1
        size = int(max(50000, ((max_val - min_val) / interval) * accuracy_factor))
        vals = skewnorm.rvs(a=skewness, loc=max_val, size=size)
        vals -= np.min(vals)
        vals /= np.max(vals)
        vals *= max_val
        if cutoff_mode == "min_val":
            vals = np.where(vals > min_val, vals, min_val)
        elif cutoff_mode == "resample":
            vals = np.where(vals > min_val, vals,
                            np.random.uniform(low=min_val, high=max_val, size=len(vals)))
        else:
            raise ValueError(f"Unrecognised cutoff_mode {cutoff_mode}")
        hist_counts, bin_vals = np.histogram(vals, bins=num_bins)
        self.random_var_vals = np.around(bin_vals[1:], decimals=decimals)
        self.random_var_probs = hist_counts / hist_counts.sum()

    def sample(self, size=None, replace=True):
        v = np.random.choice(self.random_var_vals, p=self.random_var_probs,
                             size=size, replace=replace)
        if size is None:
            return float(v)
        return v


class ListOfDistributions(Distribution):
    """Samples one of a list of distributions
    (reference ``distributions/list_of_distributions.py:9``)."""

    def __init__(self, name_to_cls_to_kwargs: Dict[str, Dict[str, dict]]):
        from .utils.registry import get_class_from_path
        self.dists: List[Distribution] = []
        for _name, cls_to_kwargs in name_to_cls_to_kwargs.items():
            for cls_path, kwargs in cls_to_kwargs.items():
                self.dists.append(get_class_from_path(cls_path)(**kwargs))

    def sample(self, size=None, replace=True):
        idx = np.random.randint(low=0, high=len(self.dists))
        return self.dists[idx]


def distribution_from_config(cfg: Union[Distribution, dict]) -> Distribution:
    """Instantiate a Distribution from a ``{_target_: path, **kwargs}`` dict
    (reference ``jobs_generator.py:125-130``)."""
    if isinstance(cfg, Distribution):
        return cfg
    if not isinstance(cfg, dict):
        raise TypeError(f"Cannot build Distribution from {cfg!r}")
    if "_target_" not in cfg:
        raise ValueError("Distribution dict config requires a _target_ key")
    from .utils.registry import get_class_from_path
    kwargs = {k: v for k, v in cfg.items() if k != "_target_"}
    return get_class_from_path(cfg["_target_"])(**kwargs)
