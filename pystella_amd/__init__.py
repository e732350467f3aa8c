"""pystella_amd: an MI355X-native framework for distributed stencil-PDE /
lattice field-theory simulation.

A from-scratch re-design of the capabilities of ``zachjweiner/pystella``
for AMD Instinct MI355X (gfx950):

* symbolic fields/sectors front-end (own expression core, no pymbolic),
* hand-written CDNA4 HIP kernels for the hot ops (stencils, fused RK
  stages, reductions, histograms, k-space maps), with user physics
  expressions spliced into the kernel templates via hiprtc,
* one process per GPU; halo exchange and collectives via
  torch.distributed — RCCL over xGMI on ROCm ("nccl" backend), gloo on
  CPU for tests,
* 3-D FFTs via torch.fft (rocFFT) on the same pencil layout.

The flat public API mirrors the reference package
(reference: pystella/__init__.py:117-155).
"""

from pystella_amd.field import (  # noqa: F401
    Field, DynamicField, index_fields, shift_fields, substitute, diff,
    collect_field_indices, indices_to_domain, infer_field_domains,
    get_field_args, var,
)
from pystella_amd.decomp import DomainDecomposition, init_distributed  # noqa: F401
from pystella_amd.elementwise import ElementWiseMap  # noqa: F401
from pystella_amd.stencil import Stencil, StreamingStencil  # noqa: F401
from pystella_amd.reduction import Reduction, FieldStatistics  # noqa: F401
from pystella_amd.histogram import Histogrammer, FieldHistogrammer  # noqa: F401
from pystella_amd.derivs import (  # noqa: F401
    FiniteDifferencer, FirstCenteredDifference, SecondCenteredDifference,
)
from pystella_amd.step import (  # noqa: F401
    Stepper, RungeKuttaStepper, LowStorageRKStepper,
    RungeKutta4, RungeKutta3SSP, RungeKutta3Heun, RungeKutta3Nystrom,
    RungeKutta3Ralston, RungeKutta2Midpoint, RungeKutta2Heun,
    RungeKutta2Ralston, LowStorageRK54, LowStorageRK144, LowStorageRK134,
    LowStorageRK124, LowStorageRK3Williamson, LowStorageRK3Inhomogeneous,
    LowStorageRK3SSP, all_steppers,
)
from pystella_amd.sectors import (  # noqa: F401
    Sector, ScalarSector, TensorPerturbationSector, get_rho_and_p,
)
from pystella_amd.expansion import Expansion  # noqa: F401
from pystella_amd.fourier import (  # noqa: F401
    DFT, RayleighGenerator, Projector, PowerSpectra, SpectralCollocator,
    SpectralPoissonSolver,
)
from pystella_amd.multigrid import (  # noqa: F401
    MultiGridSolver, FullApproximationScheme,
)
from pystella_amd.output import OutputFile  # noqa: F401

import logging
logger = logging.getLogger(__name__)


def choose_device(local_rank=None):
    """Bind this process to one GPU by node-local rank.

    Analogue of reference ``choose_device_and_make_context``
    (pystella/__init__.py:46-102): one process per GPU; the local rank
    comes from torchrun's ``LOCAL_RANK`` unless given.

    :returns: the selected :class:`torch.device`.
    """
    import os
    import torch
    if not torch.cuda.is_available():
        return torch.device("cpu")
    if local_rank is None:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dev = torch.device("cuda", local_rank % torch.cuda.device_count())
    torch.cuda.set_device(dev)
    return dev


# compatibility alias mirroring the reference name
choose_device_and_make_context = choose_device


class DisableLogging:
    """Context manager silencing a logger
    (reference pystella/__init__.py:105-114)."""

    def __init__(self, logger_to_disable=None):
        self.logger = logger_to_disable or logging.getLogger()

    def __enter__(self):
        self.original_level = self.logger.level
        self.logger.setLevel(logging.CRITICAL + 1)

    def __exit__(self, *exc):
        self.logger.setLevel(self.original_level)

__version__ = "2.0.0"
