"""Parallel execution plane: gradient synchronization and the
captured-step data-parallel fast path (see also
``hydragnn_amd/train/captured.py`` and
``hydragnn_amd/utils/distributed`` for process bootstrap / wrappers).
"""

from .grad_sync import FlatGradSync

__all__ = ["FlatGradSync"]
