# Copyright (c) Flashy-AMD authors.
"""flashy_amd — an MI355X-native minimal solver framework for deep learning.

A from-scratch framework with the capabilities of facebookresearch/flashy
(metric logging to multiple backends, automatic stateful checkpointing, and
DDP-alternative distributed utilities), re-designed for AMD Instinct MI355X:
one process per GPU over RCCL/xGMI, bucketed overlapped gradient sync on a
side HIP stream, HIP-graph step capture, hand-written CDNA4 (gfx950) HIP
kernels for the hot ops, and pinned-host streamed checkpoints.

Public surface (parity: /root/reference/flashy/__init__.py:11-15):
``distrib``, ``adversarial`` modules; ``Formatter``, ``ResultLogger``,
``LogProgressBar``, ``bold``, ``setup_logging``, ``BaseSolver``, ``averager``.
Extras beyond the reference: ``xp`` (experiment runtime), ``graph``
(HIP-graph capture), ``models``, ``ops`` (CDNA4 kernels).
"""

__version__ = "0.2.0a1"

import os as _os

# MIOpen's implicit-GEMM conv solver class mis-executes under hipGraph
# REPLAY on this stack (ROCm 7.x, bf16): a torch-module model captured with
# autocast replays NaN after 2-4 replays on ~50-70% of processes.  Solver
# bisect: base 8/12 NaN processes, CONV_IMPLICIT_GEMM=0 -> 0/12,
# CONV_GEMM=0 (more shapes onto implicit-GEMM) -> 12/12
# (scripts/graph_nan_hunt.py).  MIOpen reads this at FIRST conv and caches
# find results per shape, so it must be set before any torch conv runs —
# i.e. at package import.  The native NHWC kernels never touch MIOpen;
# only fallback torch-module models are affected (they pick the next
# solver class).  Export MIOPEN_DEBUG_CONV_IMPLICIT_GEMM=1 to override
# when graph capture is not used.
_os.environ.setdefault("MIOPEN_DEBUG_CONV_IMPLICIT_GEMM", "0")

from . import adversarial  # noqa: F401,E402
from . import distrib  # noqa: F401
from . import graph  # noqa: F401
from . import xp  # noqa: F401
from .formatter import Formatter  # noqa: F401
from .logging import LogProgressBar, ResultLogger, bold, setup_logging  # noqa: F401
from .solver import BaseSolver  # noqa: F401
from .utils import averager  # noqa: F401
