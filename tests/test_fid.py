"""FID harness correctness: Frechet distance against closed forms."""

import numpy as np
import torch

from gansformer_amd.metrics.fid import (RandomConvFeatures, compute_fid,
                                        frechet_distance)


def test_frechet_identical_is_zero():
    rng = np.random.RandomState(0)
    mu = rng.randn(8)
    a = rng.randn(64, 8)
    sigma = np.cov(a, rowvar=False)
    assert abs(frechet_distance(mu, sigma, mu, sigma)) < 1e-6


def test_frechet_mean_shift_closed_form():
    d = 6
    mu1, mu2 = np.zeros(d), np.full(d, 2.0)
    sigma = np.eye(d)
    # FID = |mu1-mu2|^2 when covariances equal
    assert abs(frechet_distance(mu1, sigma, mu2, sigma) - d * 4.0) < 1e-6


def test_frechet_isotropic_closed_form():
    d = 5
    s1, s2 = 2.0, 0.5
    # diag covs: Tr(S1+S2-2 sqrt(S1 S2)) = d*(s1+s2-2*sqrt(s1*s2))
    expect = d * (s1 + s2 - 2 * np.sqrt(s1 * s2))
    got = frechet_distance(np.zeros(d), np.eye(d) * s1,
                           np.zeros(d), np.eye(d) * s2)
    assert abs(got - expect) < 1e-6


def test_compute_fid_sanity():
    """Same distribution -> small FID; different -> larger."""
    torch.manual_seed(0)
    dev = torch.device("cpu")
    ex = RandomConvFeatures(feature_dim=32, seed=7).eval()

    def reals(n):
        g = torch.Generator().manual_seed(1)
        return torch.rand(n, 3, 32, 32, generator=g) * 2 - 1

    def fakes_same(n):
        g = torch.Generator().manual_seed(2)
        return torch.rand(n, 3, 32, 32, generator=g) * 2 - 1

    def fakes_diff(n):
        g = torch.Generator().manual_seed(3)
        return torch.rand(n, 3, 32, 32, generator=g) * 0.2 - 0.9

    fid_same, stats = compute_fid(fakes_same, reals, 64, 16, dev, extractor=ex)
    fid_diff, _ = compute_fid(fakes_diff, reals, 64, 16, dev, extractor=ex,
                              real_stats_cache=stats)
    assert fid_diff > fid_same


def test_random_features_deterministic():
    a = RandomConvFeatures(feature_dim=16, seed=5)
    b = RandomConvFeatures(feature_dim=16, seed=5)
    x = torch.randn(2, 3, 32, 32)
    assert torch.allclose(a(x), b(x))
