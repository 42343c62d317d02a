# Copyright (c) Flashy-AMD authors.
"""Experiment-logger backends (LocalFS always; TensorBoard/WandB soft deps)."""
from .base import ExperimentLogger  # noqa: F401
from .localfs import LocalFSLogger  # noqa: F401
