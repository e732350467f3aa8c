"""PowerSpectra tests: bin-count sanity and spectrum of a known mode
(style of reference test/test_spectra.py)."""

import numpy as np
import torch

import pystella_amd as ps


def setup(grid_shape=(16, 16, 16), L=10.0):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.float64)
    dk = tuple(2 * np.pi / L for _ in range(3))
    volume = L**3
    spectra = ps.PowerSpectra(decomp, fft, dk, volume)
    return decomp, fft, spectra


def test_bin_counts_sum(grid_shape=(16, 16, 16)):
    _, _, spectra = setup(grid_shape)
    # r2c double-counting must make the bins sum to N^3
    # (reference test/test_spectra.py:67)
    assert abs(spectra.bin_counts.sum() - np.prod(grid_shape)) < 1e-10


def test_single_mode_spectrum(grid_shape=(16, 16, 16), L=10.0):
    decomp, fft, spectra = setup(grid_shape, L)
    # f = A cos(k1 x): power concentrated in the |k| = k1 bin
    dx = L / grid_shape[0]
    x = np.arange(grid_shape[0]) * dx
    A = 3.0
    f3 = A * np.cos(2 * np.pi * x / L)[:, None, None] \
        * np.ones(grid_shape)
    spec = spectra(torch.as_tensor(f3), k_power=3)
    peak = np.argmax(spec)
    assert peak == 1, spec[:5]
    # everything else is negligible
    rest = np.delete(spec, peak)
    assert rest.max() < 1e-12 * spec[peak]


def test_parseval(grid_shape=(16, 16, 16), L=10.0):
    """Σ Δ²(k)/k³·bin-volume consistency: total variance matches
    ⟨f²⟩ for a random field (loose Parseval-style check)."""
    decomp, fft, spectra = setup(grid_shape, L)
    rng = np.random.default_rng(0)
    f3 = rng.standard_normal(grid_shape)
    # k_power=0: Δ² with |k|^0 weighting
    spec = spectra(torch.as_tensor(f3), k_power=0)
    # sum over bins of spec * bin_counts recovers norm * Σ|fk|²·count
    total = (spec * spectra.bin_counts).sum()
    fk = np.fft.rfftn(f3)
    counts = 2. * np.ones_like(fk.real)
    counts[..., 0] = 1.
    counts[..., -1] = 1. if grid_shape[2] % 2 == 0 else 2.
    expect = spectra.norm * (counts * np.abs(fk)**2).sum()
    assert abs(total - expect) / expect < 1e-10
