"""Warning types for kfac_amd (parity: reference kfac/warnings.py:6)."""

from __future__ import annotations


class ExperimentalFeatureWarning(Warning):
    """Warning for experimental features."""
