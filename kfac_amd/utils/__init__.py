"""Utility subpackage: covariance helpers, tracing, schedules.

Convenience re-exports of the parity-named modules.
"""

from kfac_amd.hyperparams import exp_decay_factor_averaging
from kfac_amd.layers.utils import append_bias_ones
from kfac_amd.layers.utils import get_cov
from kfac_amd.layers.utils import reshape_data
from kfac_amd.tracing import clear_trace
from kfac_amd.tracing import get_trace
from kfac_amd.tracing import log_trace
from kfac_amd.tracing import trace

__all__ = [
    'append_bias_ones',
    'get_cov',
    'reshape_data',
    'trace',
    'get_trace',
    'log_trace',
    'clear_trace',
    'exp_decay_factor_averaging',
]
