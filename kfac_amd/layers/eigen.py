"""Eigendecomposition K-FAC layer.

Parity with reference kfac/layers/eigen.py:20-385: state {QA, QG, dA, dG,
dGdA}, prediv_eigenvalues fusion, eigenvalue clamp >= 0, empty-alloc
broadcast protocol. Compute goes through kfac_amd.ops: on MI355X the
Kronecker precondition chain (QG^T @ grad @ QA -> elementwise -> QG @ v @
QA^T) is a fused kernel sequence and the eigendecomposition runs in fp32
via rocSOLVER (batched CDNA4 Jacobi path for small factors planned).
"""

from __future__ import annotations

from typing import Any

import torch
import torch.distributed as dist

from kfac_amd import ops
from kfac_amd.distributed import get_rank
from kfac_amd.layers.base import _wait
from kfac_amd.layers.base import KFACBaseLayer


class KFACEigenLayer(KFACBaseLayer):
    """K-FAC layer preconditioning via eigendecomposition of A and G."""

    SECOND_ORDER_KEYS = ('qa', 'qg', 'da', 'dg', 'dgda')

    def __init__(self, *args: Any, prediv_eigenvalues: bool = False, **kwargs: Any) -> None:
        """Init KFACEigenLayer.

        Args:
            prediv_eigenvalues: precompute dGdA = 1/(outer(dG,dA)+damping)
                on the G inverse worker (requires colocated factors);
                trades memory for a cheaper precondition stage.
            *args/**kwargs: see KFACBaseLayer.
        """
        super().__init__(*args, **kwargs)
        self.prediv_eigenvalues = prediv_eigenvalues
        self._qa: Any = None
        self._qg: Any = None
        self._da: Any = None
        self._dg: Any = None
        self._dgda: Any = None

    # Future-wrapped eigen state ------------------------------------------

    @property
    def qa(self) -> torch.Tensor | None:
        """Eigenvectors of A."""
        self._qa = _wait(self._qa)
        return self._qa

    @qa.setter
    def qa(self, value: Any) -> None:
        self._qa = value

    @property
    def qg(self) -> torch.Tensor | None:
        """Eigenvectors of G."""
        self._qg = _wait(self._qg)
        return self._qg

    @qg.setter
    def qg(self, value: Any) -> None:
        self._qg = value

    @property
    def da(self) -> torch.Tensor | None:
        """Eigenvalues of A."""
        self._da = _wait(self._da)
        return self._da

    @da.setter
    def da(self, value: Any) -> None:
        self._da = value

    @property
    def dg(self) -> torch.Tensor | None:
        """Eigenvalues of G."""
        self._dg = _wait(self._dg)
        return self._dg

    @dg.setter
    def dg(self, value: Any) -> None:
        self._dg = value

    @property
    def dgda(self) -> torch.Tensor | None:
        """Precomputed 1/(outer(dG,dA)+damping)."""
        self._dgda = _wait(self._dgda)
        return self._dgda

    @dgda.setter
    def dgda(self, value: Any) -> None:
        self._dgda = value

    def memory_usage(self) -> dict[str, int]:
        """Add eigendecomposition state to the byte accounting."""
        sizes = super().memory_usage()

        def nbytes(t: torch.Tensor | None) -> int:
            return 0 if t is None else t.nelement() * t.element_size()

        sizes['a_inverses'] = nbytes(self.qa) + nbytes(self.da)
        sizes['g_inverses'] = (
            nbytes(self.qg) + nbytes(self.dg) + nbytes(self.dgda)
        )
        return sizes

    # Communication --------------------------------------------------------

    def broadcast_a_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast QA (+dA unless prediv) from the inverse worker."""
        if self.qa is None or (not self.prediv_eigenvalues and self.da is None):
            if get_rank() == src:
                raise RuntimeError(
                    f'Attempt to broadcast A inv from src={src} but this '
                    'rank has not computed A inv yet.',
                )
            a = self.a_factor
            assert isinstance(a, torch.Tensor)
            self.qa = torch.empty(a.shape, device=a.device, dtype=self.inv_dtype)
            self.da = torch.empty(a.shape[0], device=a.device, dtype=self.inv_dtype)
        self.qa = self.tdc.broadcast(self.qa, src=src, group=group)
        if not self.prediv_eigenvalues:
            assert self.da is not None
            self.da = self.tdc.broadcast(self.da, src=src, group=group)

    def broadcast_g_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast QG (+dG, or dGdA when prediv) from the inverse worker."""
        if (
            self.qg is None
            or (not self.prediv_eigenvalues and self.dg is None)
            or (self.prediv_eigenvalues and self.dgda is None)
        ):
            if get_rank() == src:
                raise RuntimeError(
                    f'Attempt to broadcast G inv from src={src} but this '
                    'rank has not computed G inv yet.',
                )
            g = self.g_factor
            assert isinstance(g, torch.Tensor)
            self.qg = torch.empty(g.shape, device=g.device, dtype=self.inv_dtype)
            if not self.prediv_eigenvalues:
                self.dg = torch.empty(
                    g.shape[0], device=g.device, dtype=self.inv_dtype,
                )
            else:
                a = self.a_factor
                assert isinstance(a, torch.Tensor)
                self.dgda = torch.empty(
                    (g.shape[0], a.shape[0]),
                    device=g.device,
                    dtype=self.inv_dtype,
                )
        self.qg = self.tdc.broadcast(self.qg, src=src, group=group)
        if not self.prediv_eigenvalues:
            assert self.dg is not None
            self.dg = self.tdc.broadcast(self.dg, src=src, group=group)
        else:
            assert self.dgda is not None
            self.dgda = self.tdc.broadcast(self.dgda, src=src, group=group)

    # Compute ---------------------------------------------------------------

    def compute_a_inv(self, damping: float = 0.001) -> None:
        """Eigendecompose A in fp32 (reference eigen.py:295-321)."""
        a = self.a_factor
        if not isinstance(a, torch.Tensor):
            raise RuntimeError('Cannot eigendecompose A before A has been computed')
        if self.symmetric_factors:
            da, qa = ops.eigh(a, clamp=False)
        else:
            dac, qac = torch.linalg.eig(a)
            da, qa = dac.real, qac.real
        self.qa = qa.to(self.inv_dtype)
        self.da = torch.clamp(da.to(self.inv_dtype), min=0.0)

    def compute_g_inv_no_prediv(self) -> None:
        """Eigendecompose G without the dGdA fusion (batched-inverse path:
        the preconditioner applies prediv after both factors are done)."""
        g = self.g_factor
        if not isinstance(g, torch.Tensor):
            raise RuntimeError('Cannot eigendecompose G before G has been computed')
        if self.symmetric_factors:
            dg, qg = ops.eigh(g, clamp=False)
        else:
            dgc, qgc = torch.linalg.eig(g)
            dg, qg = dgc.real, qgc.real
        self.qg = qg.to(self.inv_dtype)
        self.dg = torch.clamp(dg.to(self.inv_dtype), min=0.0)

    def compute_g_inv(self, damping: float = 0.001) -> None:
        """Eigendecompose G; optionally fuse dGdA (reference eigen.py:323-348)."""
        self.compute_g_inv_no_prediv()
        if self.prediv_eigenvalues:
            da = self.da
            assert da is not None
            self.dgda = 1 / (torch.outer(self.dg, da) + damping)
            self.dg = None
            self.da = None

    def preconditioned_grad(self, damping: float = 0.001) -> None:
        """Fused Kronecker-eigenbasis precondition (reference eigen.py:350-385)."""
        qa = self.qa
        qg = self.qg
        if (
            qa is None
            or qg is None
            or (not self.prediv_eigenvalues and (self.da is None or self.dg is None))
            or (self.prediv_eigenvalues and self.dgda is None)
        ):
            raise RuntimeError(
                'Eigendecompositions for both A and G have not been computed',
            )
        grad = self.module.get_grad()
        if self.prediv_eigenvalues:
            self.grad = ops.precond_eigen(grad, qa, qg, dgda=self.dgda)
        else:
            self.grad = ops.precond_eigen(
                grad, qa, qg, da=self.da, dg=self.dg, damping=damping,
            )
