"""ElementWiseMap vs numpy oracle (style of reference
test/test_elementwise.py)."""

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.field import Field, var


def test_elementwise_map(grid_shape=(16, 16, 16)):
    f = Field("f", offset=0, shape=(2,))
    g = Field("g", offset=0)
    c = var("c")
    ew = ps.ElementWiseMap({g: c * f[0]**2 + f[1] / 3},
                           halo_shape=0, rank_shape=grid_shape)
    rng = np.random.default_rng(3)
    fx = torch.as_tensor(rng.random((2,) + grid_shape))
    gx = torch.zeros(grid_shape, dtype=torch.float64)
    ew(f=fx, g=gx, c=1.7)
    want = 1.7 * fx[0].numpy()**2 + fx[1].numpy() / 3
    assert np.allclose(gx.numpy(), want)


def test_elementwise_tmp_instructions(grid_shape=(8, 8, 8)):
    f = Field("f", offset=0)
    g = Field("g", offset=0)
    t = var("tmp0")
    ew = ps.ElementWiseMap({g: t * t},
                           tmp_instructions={t: f + 1.0},
                           halo_shape=0, rank_shape=grid_shape)
    fx = torch.rand(grid_shape, dtype=torch.float64)
    gx = torch.zeros(grid_shape, dtype=torch.float64)
    ew(f=fx, g=gx)
    assert torch.allclose(gx, (fx + 1.0)**2)


def test_elementwise_in_place_sequential(grid_shape=(8, 8, 8)):
    """Statements execute in dict order with per-site semantics."""
    f = Field("f", offset=0)
    g = Field("g", offset=0)
    ew = ps.ElementWiseMap({g: 2 * f, f: f + 1},
                           halo_shape=0, rank_shape=grid_shape)
    fx = torch.rand(grid_shape, dtype=torch.float64)
    f0 = fx.clone()
    gx = torch.zeros(grid_shape, dtype=torch.float64)
    ew(f=fx, g=gx)
    assert torch.allclose(gx, 2 * f0)
    assert torch.allclose(fx, f0 + 1)
