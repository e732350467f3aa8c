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
    # default backend is the self-contained HDF5 writer
    from pystella_amd.hdf5 import read_file
    tree = read_file("testout.h5")
    energy = np.asarray(
        tree["children"]["energy"]["children"]["total"]["data"])
    assert energy.shape[0] >= 1
    assert np.isfinite(energy).all()
    # provenance attrs captured at the root (reference output.py:98-155)
    assert "argv" in tree["attrs"] and "hostname" in tree["attrs"]


def test_scalar_preheating_gws(tmp_path):
    """Gravitational-wave sector end to end (CPU, small grid)."""
    import scalar_preheating
    os.chdir(tmp_path)
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "12", "12", "12", "--end-time", "0.3",
         "--device", "cpu", "--no-output", "--gravitational-waves"])
    assert np.isfinite(expand.constraint(energy["total"]))


def _preheating_dist_worker(rank, world_size):
    import scalar_preheating
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "16", "16", "16", "--proc-shape", "2", "1", "1",
         "--end-time", "0.2", "--device", "cpu", "--no-output"])
    # all ranks agree on the (allreduced) energy and expansion state
    assert np.isfinite(energy["total"])
    import pystella_amd as ps
    import torch.distributed as dist
    import torch
    t = torch.tensor([energy["total"], float(expand.a[0])])
    t0 = t.clone()
    dist.broadcast(t0, src=0)
    assert torch.allclose(t, t0), (rank, t, t0)


def test_scalar_preheating_distributed():
    from tests.conftest import run_distributed
    run_distributed(_preheating_dist_worker, 2)


def test_scalar_preheating_golden(tmp_path):
    """Physics regression with golden values (analogue of the
    reference's golden Friedmann-constraint check,
    test/test_examples.py:31-67, which asserts 5.5725530301309334e-08
    at 32^3: our value differs because the RNG stream differs, but it
    is fixed by the deterministic seed here)."""
    import scalar_preheating
    os.chdir(tmp_path)
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "16", "16", "16", "--end-time", "1.0",
         "--device", "cpu", "--no-output"])
    constraint = float(expand.constraint(energy["total"]))
    assert abs(constraint - 2.6929495300365147e-08) < 1e-3 * \
        2.6929495300365147e-08 + 1e-12, constraint
    assert abs(float(expand.a[0]) - 1.5573428265664833) < 1e-6


def test_bench_distributed_cpu(tmp_path):
    """The driver's exact multi-rank launch pattern against bench.py
    (torchrun, 2 ranks, gloo on CPU, tiny grid)."""
    import json
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--standalone", "--local-addr", "127.0.0.1",
           "--nnodes=1", "--nproc-per-node", "2",
           "--redirects", "3", "--log-dir", str(tmp_path / "trlogs"),
           os.path.join(repo, "bench.py"),
           "--gpus", "2", "--steps", "2", "--warmup", "1",
           "--grid", "16", "--device", "cpu"]
    from tests.conftest import run_torchrun
    out, logs = run_torchrun(cmd, repo, tmp_path / "trlogs")
    assert out.returncode == 0, (out.stderr[-1500:], logs[-1500:])
    line = [ln for ln in logs.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "decomp3d[2, 1, 1]"
    assert d["value"] > 0


def test_bench_distributed_gws_cpu(tmp_path):
    """2-rank gloo bench with the GW tensor sector (multi-family
    fused path under the driver's launch pattern)."""
    import glob
    import json
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--standalone", "--local-addr", "127.0.0.1",
           "--nnodes=1", "--nproc-per-node", "2",
           "--redirects", "3", "--log-dir", str(tmp_path / "trlogs"),
           os.path.join(repo, "bench.py"),
           "--gpus", "2", "--steps", "1", "--warmup", "0",
           "--grid", "16", "--device", "cpu", "--gws"]
    from tests.conftest import run_torchrun
    out, logs = run_torchrun(cmd, repo, tmp_path / "trlogs")
    assert out.returncode == 0, (out.stderr[-1500:], logs[-1500:])
    line = [ln for ln in logs.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["model"] == "scalar_preheating+gw"


def test_bench_8rank_cpu(tmp_path):
    """8-rank (2,2,2) bench on gloo — the exact topology of the
    driver's N=8 strong-scaling run."""
    import glob
    import json
    import subprocess
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           "--nnodes=1", "--nproc-per-node", "8",
           "--redirects", "3", "--log-dir", str(tmp_path / "trlogs"),
           os.path.join(repo, "bench.py"),
           "--gpus", "8", "--steps", "2", "--warmup", "1",
           "--grid", "16", "--device", "cpu"]
    from tests.conftest import run_torchrun
    out, logs = run_torchrun(cmd, repo, tmp_path / "trlogs")
    assert out.returncode == 0, (out.stderr[-1500:], logs[-1500:])
    line = [ln for ln in logs.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["parallelism"] == "decomp3d[2, 2, 2]"
    assert d["value"] > 0


def test_scalar_preheating_multi_chi(tmp_path):
    """Per-scalar coupling lists (--gsq with two values -> three
    scalars): CLI surface matches reference scalar_preheating.py:52-58
    nargs="*" and actually evolves the extra scalar."""
    import scalar_preheating
    os.chdir(tmp_path)
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "12", "12", "12", "--end-time", "0.2",
         "--device", "cpu", "--no-output",
         "--gsq", "2.5e-7", "1e-7", "--mchi", "0", "1e-7"])
    assert np.isfinite(energy["total"])


def test_scalar_preheating_fp32(tmp_path):
    """--dtype float32 runs the whole pipeline (fields, stencils, FFT,
    histogram) in fp32 (reference dtype-generality parity)."""
    import scalar_preheating
    os.chdir(tmp_path)
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "12", "12", "12", "--end-time", "0.2",
         "--device", "cpu", "--no-output", "--dtype", "float32"])
    assert np.isfinite(energy["total"])


def _preheating_gws_8rank_worker(rank, world_size):
    """GW example incl. spectra output at the (2,2,2) N=8 topology:
    the full observables path (pencil FFT pz>1, TT projection, binned
    spectra, HDF5 output on rank 0) end to end."""
    import tempfile
    import scalar_preheating
    os.chdir(tempfile.mkdtemp())
    expand, energy = scalar_preheating.main(
        ["--grid-shape", "16", "16", "16",
         "--proc-shape", "2", "2", "2",
         "--end-time", "0.05", "--end-scale-factor", "1.0001",
         "--device", "cpu", "--gravitational-waves",
         "--outfile", f"gwout"])
    assert np.isfinite(energy["total"])


def test_scalar_preheating_gws_8rank():
    from tests.conftest import run_distributed
    run_distributed(_preheating_gws_8rank_worker, 8)


def test_bench_4rank_gws_cpu(tmp_path):
    """4-rank (2,2,1) gloo bench with the GW tensor sector: multi-
    family ring path + overlapped exchange at a 2-D topology."""
    import glob
    import json
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           "--nnodes=1", "--nproc-per-node", "4",
           "--redirects", "3", "--log-dir", str(tmp_path / "trlogs"),
           os.path.join(repo, "bench.py"),
           "--gpus", "4", "--steps", "2", "--warmup", "1",
           "--grid", "16", "--device", "cpu", "--gws"]
    from tests.conftest import run_torchrun
    out, logs = run_torchrun(cmd, repo, tmp_path / "trlogs")
    assert out.returncode == 0, (out.stderr[-1500:], logs[-1500:])
    line = [ln for ln in logs.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["parallelism"] == "decomp3d[2, 2, 1]"
    assert d["config"]["model"] == "scalar_preheating+gw"


def test_tutorial(tmp_path):
    """The runnable tutorial (examples/tutorial.py, the executable
    form of docs/TUTORIAL.md) covers every layer end to end."""
    import tutorial
    os.chdir(tmp_path)
    energy = tutorial.main(["--device", "cpu", "--n", "12"])
    assert np.isfinite(energy["total"])
    assert os.path.exists("tutorial_out.h5")
    assert os.path.exists("tutorial_ckpt.h5")
