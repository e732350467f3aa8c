"""Projector identity tests: transversality, pol↔vec round trips,
transverse-traceless-ness (style of reference test/test_projectors.py)."""

import numpy as np
import pytest
import torch

import pystella_amd as ps
from pystella_amd.sectors import tensor_index as tid


def setup(grid_shape=(12, 12, 12), L=10.0, h=0):
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.float64)
    dk = tuple(2 * np.pi / L for _ in range(3))
    dx = tuple(L / n for n in grid_shape)
    proj = ps.Projector(fft, h, dk, dx)
    return fft, proj


def random_vector_k(fft, seed=0):
    rng = np.random.default_rng(seed)
    shape = (3,) + tuple(fft.shape(True))
    v = rng.standard_normal(shape) + 1j * rng.standard_normal(shape)
    return torch.as_tensor(v)


def test_transversality():
    fft, proj = setup()
    vec = random_vector_k(fft)
    proj.transversify(vector=vec)
    kdotv = sum(proj.kvec[mu] * vec[mu] for mu in range(3))
    assert kdotv.abs().max().item() < 1e-10


def test_pol_roundtrip():
    fft, proj = setup()
    vec = random_vector_k(fft, 1)
    proj.transversify(vector=vec)
    kshape = tuple(fft.shape(True))
    plus = torch.zeros(kshape, dtype=vec.dtype)
    minus = torch.zeros(kshape, dtype=vec.dtype)
    proj.vec_to_pol(plus=plus, minus=minus, vector=vec)
    back = torch.zeros_like(vec)
    proj.pol_to_vec(plus=plus, minus=minus, vector=back)
    # round trip reproduces the transverse field except where k≅0
    mask = ~proj.kvec_zero
    for mu in range(3):
        diff = (back[mu] - vec[mu])[mask].abs().max().item()
        assert diff < 1e-10, mu


def test_transverse_traceless():
    fft, proj = setup()
    rng = np.random.default_rng(2)
    kshape = tuple(fft.shape(True))
    hij = torch.as_tensor(
        rng.standard_normal((6,) + kshape)
        + 1j * rng.standard_normal((6,) + kshape))
    proj.transverse_traceless(hij=hij)
    # traceless: h_11 + h_22 + h_33 = 0
    trace = hij[tid(1, 1)] + hij[tid(2, 2)] + hij[tid(3, 3)]
    assert trace.abs().max().item() < 1e-10
    # transverse: k_i h_ij = 0 for each j
    for j in range(1, 4):
        div = sum(proj.kvec[i - 1] * hij[tid(i, j)] for i in range(1, 4))
        assert div.abs().max().item() < 1e-8, j


def test_tt_idempotent():
    fft, proj = setup()
    rng = np.random.default_rng(3)
    kshape = tuple(fft.shape(True))
    hij = torch.as_tensor(
        rng.standard_normal((6,) + kshape)
        + 1j * rng.standard_normal((6,) + kshape))
    proj.transverse_traceless(hij=hij)
    once = hij.clone()
    proj.transverse_traceless(hij=hij)
    assert (hij - once).abs().max().item() < 1e-10


def test_tensor_pol_roundtrip():
    fft, proj = setup()
    kshape = tuple(fft.shape(True))
    rng = np.random.default_rng(4)
    plus = torch.as_tensor(rng.standard_normal(kshape)
                           + 1j * rng.standard_normal(kshape))
    minus = torch.as_tensor(rng.standard_normal(kshape)
                            + 1j * rng.standard_normal(kshape))
    hij = torch.zeros((6,) + kshape, dtype=plus.dtype)
    proj.pol_to_tensor(plus=plus, minus=minus, hij=hij)
    p2 = torch.zeros_like(plus)
    m2 = torch.zeros_like(minus)
    proj.tensor_to_pol(plus=p2, minus=m2, hij=hij)
    mask = ~proj.kvec_zero
    assert (p2 - plus)[mask].abs().max().item() < 1e-10
    assert (m2 - minus)[mask].abs().max().item() < 1e-10


@pytest.mark.parametrize("h", [0, 2])
def test_effective_momenta(h):
    # with h != 0, eff momenta are the FD stencil eigenvalues
    fft, proj = setup(h=h)
    kx = proj.eff_mom["eff_mom_x"].numpy()
    assert kx[0] == 0.
    if h:
        from pystella_amd.derivs import FirstCenteredDifference
        import numpy as np
        stencil = FirstCenteredDifference(h)
        n = fft.grid_shape[0]
        L = 10.0
        dk = 2 * np.pi / L
        dx = L / n
        kk = ps.fourier.dft.fftfreq(n)
        expect = stencil.get_eigenvalues(dk * kk, dx)
        expect[np.abs(kk) == n // 2] = 0.
        expect[kk == 0] = 0.
        assert np.allclose(kx, expect)


def test_decompose_vector_aliased_outputs():
    """decompose_vector with plus/minus/lng as views of the INPUT
    vector (how PowerSpectra.vector_decomposition calls it, matching
    reference spectra.py:300-314) must equal the separate-buffer
    result — the one-kernel reference semantics read every input
    before writing (regression: the torch chain used to compute the
    longitudinal mode from already-overwritten components)."""
    fft, proj = setup((12, 10, 8))
    kshape = tuple(fft.shape(True))
    rng = np.random.default_rng(17)
    v0 = torch.as_tensor(rng.standard_normal((3,) + kshape)
                         + 1j * rng.standard_normal((3,) + kshape))
    for tak in (False, True):
        plus = torch.empty(kshape, dtype=torch.complex128)
        minus = torch.empty_like(plus)
        lng = torch.empty_like(plus)
        proj.decompose_vector(vector=v0.clone(), plus=plus, minus=minus,
                              lng=lng, times_abs_k=tak)
        va = v0.clone()
        proj.decompose_vector(vector=va, plus=va[0], minus=va[1],
                              lng=va[2], times_abs_k=tak)
        assert torch.allclose(va[0], plus, atol=1e-13)
        assert torch.allclose(va[1], minus, atol=1e-13)
        assert torch.allclose(va[2], lng, atol=1e-13), \
            (va[2] - lng).abs().max()
