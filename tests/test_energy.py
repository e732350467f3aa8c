"""Energy reducers vs a direct numpy oracle (analogue of reference
test/test_energy.py; reducer forms: pystella/sectors.py:133-144)."""

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.sectors import get_rho_and_p


def test_scalar_sector_energy_vs_numpy(grid_shape=(16, 16, 16), h=2):
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.21, 0.22, 0.23)
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return 0.5 * f[0]**2 + 0.1 * f[0]**2 * f[1]**2

    sector = ps.ScalarSector(2, potential=pot)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    red = ps.Reduction(decomp, sector, halo_shape=h,
                       callback=get_rho_and_p, rank_shape=grid_shape,
                       grid_size=gsize)

    rng = np.random.default_rng(9)
    pad = tuple(n + 2 * h for n in grid_shape)
    f = torch.as_tensor(rng.random((2,) + pad))
    dfdt = torch.as_tensor(rng.random((2,) + pad))
    lap = torch.zeros((2,) + grid_shape, dtype=torch.float64)
    derivs(fx=f, lap=lap)          # shares halos + FD Laplacian
    a = np.array([1.3])

    out = red(f=f, dfdt=dfdt, lap_f=lap, a=a)

    # numpy oracle
    cut = (slice(None),) + (slice(h, -h),) * 3
    fn = f.numpy()[cut]
    dn = dfdt.numpy()[cut]
    ln = lap.numpy()
    kin = [np.mean(dn[i]**2 / 2) / a[0]**2 for i in range(2)]
    grad = [np.mean(-fn[i] * ln[i] / 2) / a[0]**2 for i in range(2)]
    potv = np.mean(0.5 * fn[0]**2 + 0.1 * fn[0]**2 * fn[1]**2)

    assert np.allclose(out["kinetic"], kin, rtol=1e-12)
    assert np.allclose(out["gradient"], grad, rtol=1e-12)
    assert np.allclose(out["potential"], [potv], rtol=1e-12)
    total = sum(kin) + sum(grad) + potv
    assert np.allclose(out["total"], total, rtol=1e-12)
    pressure = sum(kin) - sum(grad) / 3 - potv
    assert np.allclose(out["pressure"], pressure, rtol=1e-12)
