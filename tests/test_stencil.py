"""Stencil kernels vs sliced-array numpy oracle (style of reference
test/test_stencil.py:72-97)."""

import numpy as np
import pytest
import torch

import pystella_amd as ps
from pystella_amd.field import Field, shift_fields


@pytest.mark.parametrize("h", [1, 2])
def test_stencil_vs_sliced_oracle(h, grid_shape=(16, 16, 16)):
    f = Field("f", offset="h")
    g = Field("g", offset=0)
    # asymmetric neighborhood sum
    expr = (2.0 * f
            + shift_fields(f, (h, 0, 0))
            - 0.5 * shift_fields(f, (0, -h, 0))
            + 0.25 * shift_fields(f, (0, 0, h)))
    st = ps.Stencil({g: expr}, halo_shape=h, rank_shape=grid_shape)

    pad = tuple(n + 2 * h for n in grid_shape)
    rng = np.random.default_rng(7)
    fx = torch.as_tensor(rng.random(pad))
    gx = torch.zeros(grid_shape, dtype=torch.float64)
    st(f=fx, g=gx)

    a = fx.numpy()
    c = slice(h, -h)
    want = (2.0 * a[c, c, c]
            + a[2 * h:, c, c][:grid_shape[0]]
            - 0.5 * a[c, 0:-2 * h, c]
            + 0.25 * a[c, c, 2 * h:][:, :, :grid_shape[2]])
    assert np.allclose(gx.numpy(), want)


def test_streaming_stencil_same_result(grid_shape=(12, 12, 12), h=1):
    f = Field("f", offset="h")
    g = Field("g", offset=0)
    lap = sum(
        shift_fields(f, tuple(s * int(m == d) for m in range(3)))
        + shift_fields(f, tuple(-s * int(m == d) for m in range(3)))
        - 2 * f
        for d in range(3) for s in [1])
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(2)
    fx = torch.rand(pad, dtype=torch.float64)
    out1 = torch.zeros(grid_shape, dtype=torch.float64)
    out2 = torch.zeros(grid_shape, dtype=torch.float64)
    ps.Stencil({g: lap}, halo_shape=h, rank_shape=grid_shape)(
        f=fx, g=out1)
    ps.StreamingStencil({g: lap}, halo_shape=h,
                        rank_shape=grid_shape)(f=fx, g=out2)
    assert torch.equal(out1, out2)
