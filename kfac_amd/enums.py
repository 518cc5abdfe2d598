"""Enums controlling K-FAC behavior.

Parity with reference kfac/enums.py:8-55 (gpauloski/kfac-pytorch), re-designed
for the MI355X build: the allreduce default is bucketed because small
per-layer factor allreduces are latency-bound on xGMI (7 point-to-point
links per GPU) and bucketing amortizes per-collective launch cost.
"""

from __future__ import annotations

import enum


class AllreduceMethod(enum.Enum):
    """Allreduce method for factor communication.

    ALLREDUCE: one async allreduce per factor tensor.
    ALLREDUCE_BUCKETED: factors are packed into flat buckets (default cap
        25 MB) and allreduced together — preferred on xGMI where a ring
        allreduce of a small (n,n) factor is launch/latency bound.
    """

    ALLREDUCE = 1
    ALLREDUCE_BUCKETED = 2


class AssignmentStrategy(enum.Enum):
    """Load-balancing cost model for assigning factor work to ranks.

    COMPUTE: balance by eigendecomposition cost, proportional to n^3.
    MEMORY: balance by factor storage, proportional to n^2.
    """

    COMPUTE = 1
    MEMORY = 2


class ComputeMethod(enum.Enum):
    """Second-order compute method.

    EIGEN: eigendecomposition of damped factors (default; allows fused
        eigenvalue-outer-product preconditioning).
    INVERSE: explicit damped matrix inverse.
    """

    EIGEN = 1
    INVERSE = 2


class DistributedStrategy(enum.Enum):
    """Predefined gradient-worker-fraction strategies (KAISA).

    COMM_OPT: grad_worker_fraction = 1.0. Every rank preconditions its own
        gradient; eigendecompositions are broadcast to all ranks. Minimizes
        per-step communication, maximizes memory.
    MEM_OPT: grad_worker_fraction = 1/world_size. Only the inverse worker
        preconditions; the preconditioned gradient is broadcast every step.
        Minimizes memory.
    HYBRID_OPT: grad_worker_fraction = 0.5.
    """

    COMM_OPT = 1
    MEM_OPT = 2
    HYBRID_OPT = 3
