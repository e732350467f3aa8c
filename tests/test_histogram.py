"""Histogrammer tests vs numpy (oracle style of reference
test/test_histogram.py)."""

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.field import Field, Call


def test_histogrammer(grid_shape=(16, 16, 16)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    num_bins = 32
    torch.manual_seed(0)
    f = torch.rand(grid_shape, dtype=torch.float64)

    F = Field("f", offset=0)
    hist = ps.Histogrammer(
        decomp, {"h": (F * num_bins, 1)}, num_bins, np.float64,
        halo_shape=0)
    out = hist(f=f)["h"]

    expect, _ = np.histogram(f.numpy().reshape(-1),
                             bins=np.arange(num_bins + 1) / num_bins)
    assert np.array_equal(out, expect.astype(np.float64))


def test_weighted_histogram(grid_shape=(8, 8, 8)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    num_bins = 16
    torch.manual_seed(1)
    f = torch.rand(grid_shape, dtype=torch.float64)
    F = Field("f", offset=0)
    hist = ps.Histogrammer(
        decomp, {"w": (Call("round", (F * (num_bins - 1),)), F**2)},
        num_bins, np.float64, halo_shape=0)
    out = hist(f=f)["w"]

    b = np.round(f.numpy() * (num_bins - 1)).astype(int)
    expect = np.zeros(num_bins)
    np.add.at(expect, b.reshape(-1), (f.numpy()**2).reshape(-1))
    assert np.allclose(out, expect, rtol=1e-12)


def test_field_histogrammer(grid_shape=(16, 16, 16)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    torch.manual_seed(2)
    f = torch.rand((2,) + grid_shape, dtype=torch.float64) + 0.5
    fh = ps.FieldHistogrammer(decomp, 50, np.float64, halo_shape=0,
                              rank_shape=grid_shape)
    out = fh(f)
    assert out["linear"].shape == (2, 50)
    # every site lands in some bin
    assert np.allclose(out["linear"].sum(axis=-1),
                       float(np.prod(grid_shape)))
    assert np.allclose(out["log"].sum(axis=-1),
                       float(np.prod(grid_shape)))
