"""Explicit-inverse K-FAC layer.

Parity with reference kfac/layers/inverse.py:22-234: damped Cholesky
inverse in fp32 cast to inv_dtype, symmetric-triu-aware broadcast,
precondition grad = G^-1 @ grad @ A^-1 (fused via kfac_amd.ops on GPU).
"""

from __future__ import annotations

from typing import Any

import torch
import torch.distributed as dist

from kfac_amd import ops
from kfac_amd.distributed import get_rank
from kfac_amd.layers.base import _wait
from kfac_amd.layers.base import KFACBaseLayer


class KFACInverseLayer(KFACBaseLayer):
    """K-FAC layer preconditioning via explicit damped inverses."""

    SECOND_ORDER_KEYS = ('a_inv', 'g_inv')

    def __init__(self, *args: Any, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        self._a_inv: Any = None
        self._g_inv: Any = None

    @property
    def a_inv(self) -> torch.Tensor | None:
        """Inverse of the damped A factor."""
        self._a_inv = _wait(self._a_inv)
        return self._a_inv

    @a_inv.setter
    def a_inv(self, value: Any) -> None:
        self._a_inv = value

    @property
    def g_inv(self) -> torch.Tensor | None:
        """Inverse of the damped G factor."""
        self._g_inv = _wait(self._g_inv)
        return self._g_inv

    @g_inv.setter
    def g_inv(self, value: Any) -> None:
        self._g_inv = value

    def memory_usage(self) -> dict[str, int]:
        """Add inverse state to the byte accounting."""
        sizes = super().memory_usage()

        def nbytes(t: torch.Tensor | None) -> int:
            return 0 if t is None else t.nelement() * t.element_size()

        sizes['a_inverses'] = nbytes(self.a_inv)
        sizes['g_inverses'] = nbytes(self.g_inv)
        return sizes

    def broadcast_a_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast A^-1 from the inverse worker (triu-packed if enabled)."""
        if self.a_inv is None:
            if get_rank() == src:
                raise RuntimeError(
                    f'Attempt to broadcast A inv from src={src} but this '
                    'rank has not computed A inv yet.',
                )
            a = self.a_factor
            assert isinstance(a, torch.Tensor)
            self.a_inv = torch.empty(a.shape, device=a.device, dtype=self.inv_dtype)
        self.a_inv = self.tdc.broadcast(
            self.a_inv,
            src=src,
            group=group,
            symmetric=self.symmetric_factors and self.symmetry_aware,
        )

    def broadcast_g_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast G^-1 from the inverse worker (triu-packed if enabled)."""
        if self.g_inv is None:
            if get_rank() == src:
                raise RuntimeError(
                    f'Attempt to broadcast G inv from src={src} but this '
                    'rank has not computed G inv yet.',
                )
            g = self.g_factor
            assert isinstance(g, torch.Tensor)
            self.g_inv = torch.empty(g.shape, device=g.device, dtype=self.inv_dtype)
        self.g_inv = self.tdc.broadcast(
            self.g_inv,
            src=src,
            group=group,
            symmetric=self.symmetric_factors and self.symmetry_aware,
        )

    def _damped_inverse(
        self,
        factor: torch.Tensor,
        prev: torch.Tensor | None,
        damping: float,
    ) -> torch.Tensor:
        """(factor + damping I)^-1, warm-started from the previous
        phase's inverse when available.

        The INVERSE-method analog of the eigen path's warm solver:
        factors are slowly-drifting EMAs, so two Newton-Schulz
        iterations from the previous inverse replace the fresh
        factorize+invert (ops.refine_inverse docstring has the cost
        model). Residual-certified — a failed refinement falls back to
        the exact inverse, so numerics never degrade. Disable with
        KFAC_AMD_WARM_INV=0.
        """
        import os

        if (
            isinstance(prev, torch.Tensor)
            and prev.shape == factor.shape
            and os.environ.get('KFAC_AMD_WARM_INV', '1') == '1'
        ):
            f32 = factor.to(torch.float32)
            m = f32 + damping * torch.eye(
                f32.size(0), dtype=f32.dtype, device=f32.device,
            )
            x, ok = ops.refine_inverse(m, prev.to(torch.float32))
            if ok:
                return x.to(self.inv_dtype)
        return ops.inv_damped(factor, damping).to(self.inv_dtype)

    def compute_a_inv(self, damping: float = 0.001) -> None:
        """A^-1 = (A + damping I)^-1 in fp32 (reference inverse.py:186-199)."""
        a = self.a_factor
        if not isinstance(a, torch.Tensor):
            raise RuntimeError('Cannot invert A before A has been computed')
        prev = self._a_inv if isinstance(self._a_inv, torch.Tensor) else None
        self.a_inv = self._damped_inverse(a, prev, damping)

    def compute_g_inv(self, damping: float = 0.001) -> None:
        """G^-1 = (G + damping I)^-1 in fp32 (reference inverse.py:201-213)."""
        g = self.g_factor
        if not isinstance(g, torch.Tensor):
            raise RuntimeError('Cannot invert G before G has been computed')
        prev = self._g_inv if isinstance(self._g_inv, torch.Tensor) else None
        self.g_inv = self._damped_inverse(g, prev, damping)

    def preconditioned_grad(self, damping: float = 0.001) -> None:
        """grad <- G^-1 @ grad @ A^-1 (reference inverse.py:215-234)."""
        a_inv = self.a_inv
        g_inv = self.g_inv
        if a_inv is None or g_inv is None:
            raise RuntimeError(
                'Inverses for both A and G have not been computed',
            )
        grad = self.module.get_grad()
        self.grad = ops.precond_inverse(grad, a_inv, g_inv)
