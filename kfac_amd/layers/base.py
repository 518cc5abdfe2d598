"""K-FAC layer state machine.

Parity surface with reference kfac/layers/base.py:19-423 (same method
names and call protocol: save_layer_input/grad_output -> update_* ->
reduce_* -> compute_*_inv -> broadcast_*_inv -> preconditioned_grad ->
broadcast_grad -> update_grad), with an MI355X-first storage design:

- Factors and batch accumulators are persistent fp32 buffers reused
  across steps (no per-step allocation churn; HBM3E is the bound, so the
  covariance contribution is accumulated in ONE fused HIP kernel pass via
  kfac_amd.ops instead of materializing patch matrices / intermediate
  covariance tensors like the reference's get_a_factor chain).
- Inputs may be bf16 (AMP): the covariance kernel takes bf16 operands into
  MFMA with fp32 accumulation — strictly better numerics than the
  reference's store-in-training-dtype default.
- Every communicated attribute is Tensor | Future with wait-on-first-read
  property getters (reference base.py:94-128) — the async-overlap
  mechanism that lets RCCL collectives ride backward compute.
- update_grad accepts a 0-dim device tensor scale so the kl-clip factor
  never forces a host sync (reference takes a float computed via .item()).
"""

from __future__ import annotations

from typing import Any
from typing import Callable

import torch
import torch.distributed as dist

from kfac_amd.distributed import Future
from kfac_amd.distributed import get_rank
from kfac_amd.distributed import TorchDistributedCommunicator
from kfac_amd.enums import AllreduceMethod
from kfac_amd.layers.modules import ModuleHelper


def _wait(value: Any) -> Any:
    """Resolve a Future-like (anything with .wait()) to its tensor."""
    if hasattr(value, 'wait'):
        return value.wait()
    return value


class KFACBaseLayer:
    """Per-module K-FAC state and compute/communication methods."""

    def __init__(
        self,
        module: ModuleHelper,
        *,
        tdc: TorchDistributedCommunicator,
        allreduce_method: AllreduceMethod = AllreduceMethod.ALLREDUCE_BUCKETED,
        factor_dtype: torch.dtype | None = None,
        grad_scaler: Callable[[], float] | None = None,
        inv_dtype: torch.dtype = torch.float32,
        symmetry_aware: bool = False,
    ) -> None:
        """Init KFACBaseLayer.

        Args:
            module: ModuleHelper wrapping the torch module.
            tdc: shared communicator.
            allreduce_method: per-factor or bucketed allreduce.
            factor_dtype: dtype for storing factors. None -> fp32
                (accumulation dtype of the MFMA covariance kernels).
            grad_scaler: callable returning the AMP grad scale; G
                contributions are divided by scale^2 inside the fused
                covariance epilogue (reference base.py:364-366 divides the
                activations, costing an extra full pass over HBM).
            inv_dtype: dtype for storing inverses/eigendecompositions.
            symmetry_aware: triu-pack symmetric factors on the wire.
        """
        if hasattr(grad_scaler, 'get_scale'):  # GradScaler instance
            grad_scaler = grad_scaler.get_scale  # type: ignore[union-attr]
        self.module = module
        self.tdc = tdc
        self.allreduce_method = allreduce_method
        self.factor_dtype = factor_dtype if factor_dtype is not None else torch.float32
        self.grad_scaler: Callable[[], float] | None = grad_scaler
        self.inv_dtype = inv_dtype
        self.symmetry_aware = symmetry_aware

        self.eps = 1e-10
        self.symmetric_factors = self.module.has_symmetric_factors()
        # Tensors whose side-stream covariance kernels may still be in
        # flight; held so autograd cannot free them early. Cleared by the
        # preconditioner after joining the side stream.
        self._pending_inputs: list[torch.Tensor] = []

        # Persistent accumulators for the current batch (fp32).
        self._a_batch: torch.Tensor | None = None
        self._g_batch: torch.Tensor | None = None
        self._a_count: int = 0
        self._g_count: int = 0
        # Running-average factors (Tensor | Future).
        self._a_factor: Any = None
        self._g_factor: Any = None
        # Preconditioned gradient (Tensor | Future).
        self._grad: Any = None

    def __repr__(self) -> str:
        return f'{self.__class__.__name__}({repr(self.module)})'

    # -- Future-wrapped state ---------------------------------------------

    @property
    def a_factor(self) -> torch.Tensor | None:
        """A factor (waits if a reduction is in flight)."""
        self._a_factor = _wait(self._a_factor)
        return self._a_factor

    @a_factor.setter
    def a_factor(self, value: Any) -> None:
        self._a_factor = value

    @property
    def g_factor(self) -> torch.Tensor | None:
        """G factor (waits if a reduction is in flight)."""
        self._g_factor = _wait(self._g_factor)
        return self._g_factor

    @g_factor.setter
    def g_factor(self, value: Any) -> None:
        self._g_factor = value

    @property
    def grad(self) -> torch.Tensor | None:
        """Preconditioned gradient (waits if a broadcast is in flight)."""
        self._grad = _wait(self._grad)
        return self._grad

    @grad.setter
    def grad(self, value: Any) -> None:
        self._grad = value

    # -- serialization ----------------------------------------------------

    # second-order attributes included in a state_dict when the caller
    # opts in (include_second_order=True); subclasses list theirs
    SECOND_ORDER_KEYS: tuple[str, ...] = ()

    def state_dict(
        self, include_second_order: bool = False,
    ) -> dict[str, torch.Tensor | None]:
        """Factors only by default — the reference-compatible format
        (inverses recomputable, reference base.py:130-142).  With
        ``include_second_order`` the eigendecompositions/inverses are
        embedded so a resume can skip the recomputation phase entirely
        (and keeps the warm-solver basis continuity across restarts).
        """
        sd: dict[str, torch.Tensor | None] = {
            'A': self.a_factor, 'G': self.g_factor,
        }
        if include_second_order:
            for key in self.SECOND_ORDER_KEYS:
                value = getattr(self, key)
                if isinstance(value, torch.Tensor):
                    sd[f'so_{key}'] = value
        return sd

    def load_state_dict(self, state_dict: dict[str, torch.Tensor | None]) -> None:
        """Load A/G factors (and any embedded second-order state),
        moving them to the module's device."""
        if 'A' not in state_dict or 'G' not in state_dict:
            raise KeyError(
                "KFACBaseLayer state_dict must contain keys 'A' and 'G'",
            )
        device = self.module.device
        if state_dict['A'] is not None:
            self.a_factor = state_dict['A'].to(device)
        if state_dict['G'] is not None:
            self.g_factor = state_dict['G'].to(device)
        for key in self.SECOND_ORDER_KEYS:
            value = state_dict.get(f'so_{key}')
            if isinstance(value, torch.Tensor):
                setattr(self, key, value.to(device))

    def has_second_order_state(self) -> bool:
        """True if this layer's preconditioning state is ready (either
        freshly computed or restored from a checkpoint)."""
        return any(
            isinstance(getattr(self, key), torch.Tensor)
            for key in self.SECOND_ORDER_KEYS
        )

    def memory_usage(self) -> dict[str, int]:
        """Bytes used by per-layer state (reference base.py:167-184)."""

        def nbytes(t: torch.Tensor | None) -> int:
            return 0 if t is None else t.nelement() * t.element_size()

        return {
            'a_factors': nbytes(self.a_factor if self._a_factor is not None else None),
            'g_factors': nbytes(self.g_factor if self._g_factor is not None else None),
            'a_batch': nbytes(self._a_batch) if self._a_count > 0 else 0,
            'g_batch': nbytes(self._g_batch) if self._g_count > 0 else 0,
        }

    # -- factor accumulation (hot path: fwd/bwd hooks) --------------------

    def save_layer_input(self, input_: list[torch.Tensor]) -> None:
        """Accumulate this minibatch's A-factor contribution.

        One fused covariance kernel per call: im2col/bias-ones/SYRK all in
        one pass (reference chains get_a_factor -> get_cov tensors,
        base.py:345-357).
        """
        a = input_[0]
        if self._a_batch is None:
            shape = self.module.a_factor_shape
            bdt = (
                torch.float64
                if self.factor_dtype == torch.float64
                else torch.float32
            )
            self._a_batch = torch.zeros(shape, dtype=bdt, device=a.device)
        beta = 0.0 if self._a_count == 0 else 1.0
        if a.is_cuda:
            from kfac_amd.streams import cov_stream

            s = cov_stream(a.device)
            s.wait_stream(torch.cuda.current_stream(a.device))
            with torch.cuda.stream(s):
                self.module.accumulate_a_factor(a, self._a_batch, beta, 1.0)
            self._pending_inputs.append(a)
        else:
            self.module.accumulate_a_factor(a, self._a_batch, beta, 1.0)
        self._a_count += 1

    def save_layer_grad_output(self, grad_output: tuple[torch.Tensor, ...]) -> None:
        """Accumulate this minibatch's G-factor contribution.

        AMP unscale is folded into the covariance coefficient
        (1/scale^2) instead of a separate elementwise divide.
        """
        g = grad_output[0]
        if self._g_batch is None:
            shape = self.module.g_factor_shape
            bdt = (
                torch.float64
                if self.factor_dtype == torch.float64
                else torch.float32
            )
            self._g_batch = torch.zeros(shape, dtype=bdt, device=g.device)
        coeff = 1.0
        if self.grad_scaler is not None:
            sc = float(self.grad_scaler())
            coeff = 1.0 / (sc * sc)
        beta = 0.0 if self._g_count == 0 else 1.0
        if g.is_cuda:
            from kfac_amd.streams import cov_stream

            s = cov_stream(g.device)
            s.wait_stream(torch.cuda.current_stream(g.device))
            with torch.cuda.stream(s):
                self.module.accumulate_g_factor(g, self._g_batch, beta, coeff)
            self._pending_inputs.append(g)
        else:
            self.module.accumulate_g_factor(g, self._g_batch, beta, coeff)
        self._g_count += 1

    def _factor_ctx(self) -> Any:
        """Stream context for factor EMA/reduce ops.

        While side-stream covariance kernels are in flight (hook path),
        the EMA and the allreduce launch must be ordered after them on
        the same side stream; once the preconditioner has joined the
        stream (pending list cleared), plain current-stream execution is
        correct.
        """
        if self._pending_inputs and self._pending_inputs[0].is_cuda:
            from kfac_amd.streams import cov_stream

            return torch.cuda.stream(
                cov_stream(self._pending_inputs[0].device),
            )
        import contextlib

        return contextlib.nullcontext()

    def clear_pending(self) -> None:
        """Release input refs after the cov stream has been joined."""
        self._pending_inputs.clear()

    def reset_batch(self) -> None:
        """Drop accumulated batch contributions (buffers are kept)."""
        self._a_count = 0
        self._g_count = 0

    def update_a_factor(self, alpha: float = 0.95) -> None:
        """EMA-merge the accumulated batch into the A factor.

        factor = alpha*factor + (1-alpha)*mean(batch); identity init on
        first update (reference base.py:375-390).
        """
        if self._a_count == 0 or self._a_batch is None:
            return
        w = (1.0 - alpha) / self._a_count
        with self._factor_ctx():
            if self._a_factor is None:
                f = self._a_batch.clone().mul_(w)
                f.diagonal().add_(alpha)
                self.a_factor = f.to(self.factor_dtype)
            else:
                a = self.a_factor
                assert a is not None
                a.mul_(alpha).add_(self._a_batch.to(a.dtype), alpha=w)
        self._a_count = 0

    def update_g_factor(self, alpha: float = 0.95) -> None:
        """EMA-merge the accumulated batch into the G factor."""
        if self._g_count == 0 or self._g_batch is None:
            return
        w = (1.0 - alpha) / self._g_count
        with self._factor_ctx():
            if self._g_factor is None:
                f = self._g_batch.clone().mul_(w)
                f.diagonal().add_(alpha)
                self.g_factor = f.to(self.factor_dtype)
            else:
                g = self.g_factor
                assert g is not None
                g.mul_(alpha).add_(self._g_batch.to(g.dtype), alpha=w)
        self._g_count = 0

    # -- communication -----------------------------------------------------

    def _allreduce_fn(self) -> Callable[..., Any]:
        if self.allreduce_method == AllreduceMethod.ALLREDUCE:
            return self.tdc.allreduce
        if self.allreduce_method == AllreduceMethod.ALLREDUCE_BUCKETED:
            return self.tdc.allreduce_bucketed
        raise AssertionError(f'Unknown allreduce_method={self.allreduce_method}')

    def reduce_a_factor(self, group: dist.ProcessGroup | None = None) -> None:
        """Launch async allreduce-average of A over ``group``."""
        if self.a_factor is None:
            raise RuntimeError('a_factor is None, cannot reduce')
        with self._factor_ctx():
            self.a_factor = self._allreduce_fn()(
                self.a_factor,
                average=True,
                symmetric=self.symmetric_factors and self.symmetry_aware,
                group=group,
            )

    def reduce_g_factor(self, group: dist.ProcessGroup | None = None) -> None:
        """Launch async allreduce-average of G over ``group``."""
        if self.g_factor is None:
            raise RuntimeError('g_factor is None, cannot reduce')
        with self._factor_ctx():
            self.g_factor = self._allreduce_fn()(
                self.g_factor,
                average=True,
                symmetric=self.symmetric_factors and self.symmetry_aware,
                group=group,
            )

    def broadcast_grad(
        self,
        src: int,
        group: dist.ProcessGroup | None = None,
    ) -> None:
        """Broadcast the preconditioned gradient from ``src``.

        Non-src ranks allocate an empty receive buffer
        (reference base.py:224-252).
        """
        if self.grad is None:
            if get_rank() == src:
                raise RuntimeError(
                    f'Attempt to broadcast gradient from src={src} but this '
                    'rank has not computed the preconditioned gradient yet.',
                )
            self.grad = torch.empty_like(self.module.get_grad())
        self.grad = self.tdc.broadcast(self.grad, src=src, group=group)

    # -- abstract second-order methods ------------------------------------

    def broadcast_a_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast A's second-order state from its inverse worker."""
        raise NotImplementedError

    def broadcast_g_inv(self, src: int, group: dist.ProcessGroup | None = None) -> None:
        """Broadcast G's second-order state from its inverse worker."""
        raise NotImplementedError

    def compute_a_inv(self, damping: float = 0.001) -> None:
        """Compute A's second-order state on the assigned rank."""
        raise NotImplementedError

    def compute_g_inv(self, damping: float = 0.001) -> None:
        """Compute G's second-order state on the assigned rank."""
        raise NotImplementedError

    def preconditioned_grad(self, damping: float = 0.001) -> None:
        """Compute the preconditioned gradient for this layer."""
        raise NotImplementedError

    # -- gradient update ---------------------------------------------------

    def update_grad(self, scale: float | torch.Tensor | None = None) -> None:
        """Write the (optionally scaled) preconditioned grad into the module.

        ``scale`` may be a 0-dim device tensor (fused kl-clip path — no
        host sync) or a float (reference-compatible).
        """
        grad = self.grad
        if grad is None:
            raise RuntimeError(
                'preconditioned gradient is None. update_grad() called '
                'before preconditioned_grad()?',
            )
        if scale is not None:
            grad = grad * scale
        self.module.set_grad(grad)
        self.grad = None
