import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X (run with -m gpu on a GPU box)")


def pytest_addoption(parser):
    parser.addoption("--grid_shape", default="32,32,32")
    parser.addoption("--proc_shape", default="1,1,1")


@pytest.fixture
def grid_shape(request):
    return tuple(int(x) for x in
                 request.config.getoption("--grid_shape").split(","))


@pytest.fixture
def proc_shape(request):
    return tuple(int(x) for x in
                 request.config.getoption("--proc_shape").split(","))


def run_distributed(fn, world_size=2, args=(), attempts=2):
    """Spawn `world_size` processes running fn(rank, world_size, *args)
    under a gloo process group (CPU).  Retries once on failure —
    multi-process rendezvous/spawn is occasionally flaky under CI
    load (a persistently failing test still fails)."""
    import torch.multiprocessing as mp
    import tempfile
    last = None
    for _ in range(attempts):
        init_file = tempfile.NamedTemporaryFile(delete=False).name
        try:
            mp.spawn(_dist_worker,
                     args=(world_size, init_file, fn, args),
                     nprocs=world_size, join=True)
            return
        except Exception as e:          # noqa: BLE001
            last = e
    raise last


def _dist_worker(rank, world_size, init_file, fn, args):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{init_file}", rank=rank,
        world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_torchrun(cmd, cwd, log_dir, attempts=3, timeout=240):
    """Run a torchrun command, retrying on failure (multi-process
    rendezvous is occasionally flaky under CI load); returns
    (CompletedProcess, combined log text).

    Each RETRY rebinds --master-port to a freshly-probed free port:
    the caller picks its port by bind-and-close, so the original can
    be stolen in the close->rendezvous window (or still be held by an
    orphan of a failed attempt) — retrying on the same port would then
    fail all attempts identically."""
    import glob
    import socket
    import subprocess
    import time
    env = {k: v for k, v in __import__("os").environ.items()
           if k not in ("RANK", "LOCAL_RANK", "WORLD_SIZE",
                        "MASTER_ADDR", "MASTER_PORT", "LOCAL_WORLD_SIZE",
                        "GROUP_RANK", "TORCHELASTIC_RUN_ID")}
    out = None
    logs = ""
    for attempt in range(attempts):
        c = list(cmd)
        if attempt and "--master-port" in c:
            with socket.socket() as s:
                s.bind(("127.0.0.1", 0))
                c[c.index("--master-port") + 1] = \
                    str(s.getsockname()[1])
        try:
            out = subprocess.run(c, capture_output=True, text=True,
                                 timeout=timeout, cwd=cwd, env=env)
        except subprocess.TimeoutExpired as e:
            out = subprocess.CompletedProcess(
                c, returncode=-1,
                stdout=(e.stdout or b"").decode(errors="replace")
                if isinstance(e.stdout, bytes) else (e.stdout or ""),
                stderr="torchrun attempt timed out")
            time.sleep(2)
            continue
        logs = "\n".join(open(f).read() for f in glob.glob(
            str(log_dir) + "/**/*.log", recursive=True))
        if out.returncode == 0:
            return out, logs
        time.sleep(2)
    return out, logs
