"""kfac_amd: MI355X-native distributed K-FAC/KAISA gradient preconditioner.

A from-scratch CDNA4 (gfx950) implementation of the capabilities of
gpauloski/kfac-pytorch: per-layer Kronecker-factored curvature (A = a^T a,
G = g^T g), periodic eigen/inverse of the damped factors, Kronecker
gradient preconditioning, and the KAISA gradient-worker-fraction placement
strategy — with the hot ops as hand-written HIP/MFMA kernels and the
collectives as RCCL over xGMI.

Public API mirrors the reference package: ``KFACPreconditioner(model,
...).step()`` between loss.backward() and optimizer.step().
"""

from kfac_amd import assignment
from kfac_amd import base_preconditioner
from kfac_amd import distributed
from kfac_amd import enums
from kfac_amd import gpt_neox
from kfac_amd import hyperparams
from kfac_amd import layers
from kfac_amd import ops
from kfac_amd import preconditioner
from kfac_amd import scheduler
from kfac_amd import tracing
from kfac_amd import warnings
from kfac_amd.preconditioner import KFACPreconditioner

__version__ = '0.1.0'

__all__ = [
    'KFACPreconditioner',
    'assignment',
    'base_preconditioner',
    'distributed',
    'enums',
    'gpt_neox',
    'hyperparams',
    'layers',
    'ops',
    'preconditioner',
    'scheduler',
    'tracing',
    'warnings',
    '__version__',
]
