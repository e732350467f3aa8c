"""End-to-end example runs on CPU (style of reference
test/test_examples.py): wave equation and a short scalar-preheating run
whose Friedmann constraint must stay small."""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "examples"))


def test_wave_equation(tmp_path):
    import wave_equation
    energy = wave_equation.main(
        ["--grid-shape", "16", "16", "16", "--end-time", "0.2"])
    assert np.isfinite(energy) and energy > 0


def test_scalar_preheating(tmp_path):
    import scalar_preheating
    os.chdir(tmp_path)
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "16", "16", "16", "--end-time", "0.5",
         "--device", "cpu", "--no-output"])
    constraint = expand.constraint(energy["total"])
    assert np.isfinite(constraint)
    # Friedmann constraint stays small over the evolution (the reference
    # asserts ~5.6e-8 at 32^3 with end-time 1: test/test_examples.py:33)
    assert constraint < 1e-5, constraint


def test_scalar_preheating_output(tmp_path):
    import scalar_preheating
    os.chdir(tmp_path)
    scalar_preheating.main(
        ["--grid-shape", "16", "16", "16", "--end-time", "0.3",
         "--device", "cpu", "--outfile", "testout"])
    import pystella_amd as ps
    out = ps.OutputFile.__new__(ps.OutputFile)
    from pystella_amd.output import _DirStore
    store = _DirStore("testout")
    energy = store.read("energy", "total")
    assert energy.shape[0] >= 1
    assert np.isfinite(energy).all()
