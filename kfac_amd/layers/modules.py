"""Module helpers: adapters from torch.nn modules to K-FAC factor math.

Parity with reference kfac/layers/modules.py:13-237, but factor computation
is routed through kfac_amd.ops so that on MI355X the A/G covariance is ONE
fused HIP kernel (im2col + bias-ones + SYRK + EMA epilogue) instead of the
reference's pad/unfold/cat/GEMM chain — the conv patch matrix is never
materialized in HBM.

The helper API accumulates *into* a caller-owned fp32 factor tensor:
``accumulate_a_factor(x, out, beta, coeff_scale)`` computes
``out = beta*out + coeff_scale * A_contribution(x)``.
"""

from __future__ import annotations

from typing import cast

import torch

from kfac_amd import ops


class ModuleHelper:
    """Base adapter wrapping one torch.nn module."""

    def __init__(self, module: torch.nn.Module):
        """Wrap ``module``."""
        self.module = module

    def __repr__(self) -> str:
        return f'{self.__class__.__name__}({repr(self.module)})'

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        """Shape of the A (input covariance) factor."""
        raise NotImplementedError

    @property
    def g_factor_shape(self) -> tuple[int, int]:
        """Shape of the G (output-gradient covariance) factor."""
        raise NotImplementedError

    @property
    def device(self) -> torch.device:
        """Device holding the module parameters."""
        return next(self.module.parameters()).device

    def accumulate_a_factor(
        self,
        a: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        """out = beta*out + coeff_scale * A(a)."""
        raise NotImplementedError

    def accumulate_g_factor(
        self,
        g: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        """out = beta*out + coeff_scale * G(g)."""
        raise NotImplementedError

    def get_a_factor(self, a: torch.Tensor) -> torch.Tensor:
        """Return this minibatch's A-factor contribution (allocating form
        of accumulate_a_factor; reference modules.py:47-50 interface)."""
        out = torch.zeros(
            self.a_factor_shape,
            dtype=torch.float32,
            device=a.device,
        )
        self.accumulate_a_factor(a, out, 0.0, 1.0)
        return out

    def get_g_factor(self, g: torch.Tensor) -> torch.Tensor:
        """Return this minibatch's G-factor contribution (allocating form
        of accumulate_g_factor)."""
        out = torch.zeros(
            self.g_factor_shape,
            dtype=torch.float32,
            device=g.device,
        )
        self.accumulate_g_factor(g, out, 0.0, 1.0)
        return out

    def get_grad(self) -> torch.Tensor:
        """Combined (out, in[+1]) gradient matrix, bias as last column.

        Reference modules.py:56-69.
        """
        g = cast(torch.Tensor, self.module.weight.grad)
        g = g.view(g.size(0), -1)
        if self.has_bias():
            g = torch.cat(
                [g, self.module.bias.grad.view(-1, 1)],  # type: ignore[union-attr]
                1,
            )
        return g

    def get_bias_grad(self) -> torch.Tensor:
        """Gradient of the bias parameter."""
        return cast(torch.Tensor, self.module.bias.grad)

    def get_weight_grad(self) -> torch.Tensor:
        """Gradient of the weight parameter."""
        return cast(torch.Tensor, self.module.weight.grad)

    def has_bias(self) -> bool:
        """True if the module has a bias parameter."""
        return getattr(self.module, 'bias', None) is not None

    def has_symmetric_factors(self) -> bool:
        """True if A and G are symmetric (all built-in helpers are)."""
        return True

    def set_grad(self, grad: torch.Tensor) -> None:
        """Write a combined gradient matrix back into weight/bias grads.

        Reference modules.py:87-97.
        """
        if self.has_bias():
            weight_grad = grad[:, :-1].reshape(self.get_weight_grad().size())
            bias_grad = grad[:, -1:].reshape(self.get_bias_grad().size())
            self.module.bias.grad = bias_grad.contiguous()  # type: ignore[union-attr]
        else:
            weight_grad = grad.reshape(self.get_weight_grad().size())
        self.module.weight.grad = weight_grad.contiguous()  # type: ignore[union-attr]


class LinearModuleHelper(ModuleHelper):
    """Adapter for torch.nn.Linear (reference modules.py:100-141)."""

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        n = self.module.weight.size(1) + int(self.has_bias())  # type: ignore[operator]
        return (n, n)

    @property
    def g_factor_shape(self) -> tuple[int, int]:
        n = self.module.weight.size(0)  # type: ignore[operator]
        return (n, n)

    def accumulate_a_factor(
        self,
        a: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        a = a.reshape(-1, a.shape[-1])
        coeff = coeff_scale / a.size(0)
        ops.cov_linear(a, bias=self.has_bias(), out=out, beta=beta, coeff=coeff)

    def accumulate_g_factor(
        self,
        g: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        g = g.reshape(-1, g.shape[-1])
        coeff = coeff_scale / g.size(0)
        ops.cov_linear(g, bias=False, out=out, beta=beta, coeff=coeff)


class Conv2dModuleHelper(ModuleHelper):
    """Adapter for torch.nn.Conv2d (reference modules.py:144-237).

    A is the covariance of im2col patches (divided by spatial size, ones
    column appended before division — semantics of modules.py:170-178);
    G is the covariance of NCHW output-grads flattened over batch*spatial
    (modules.py:180-192). On GPU both are single fused HIP kernels.
    """

    def __init__(self, module: torch.nn.Conv2d):
        self.module = module

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        kh, kw = self.module.kernel_size  # type: ignore[misc]
        n = self.module.in_channels * kh * kw + int(self.has_bias())
        return (n, n)

    @property
    def g_factor_shape(self) -> tuple[int, int]:
        n = self.module.out_channels
        return (n, n)

    def accumulate_a_factor(
        self,
        a: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        ops.cov_conv_a(
            a,
            kernel_size=cast(tuple, self.module.kernel_size),
            stride=cast(tuple, self.module.stride),
            padding=cast(tuple, self.module.padding),
            bias=self.has_bias(),
            out=out,
            beta=beta,
            coeff_scale=coeff_scale,
        )

    def accumulate_g_factor(
        self,
        g: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        ops.cov_conv_g(g, out=out, beta=beta, coeff_scale=coeff_scale)


class Conv1dModuleHelper(Conv2dModuleHelper):
    """Adapter for torch.nn.Conv1d — treated as a (length x 1) Conv2d.

    Capability extension beyond the reference (which supports Linear and
    Conv2d only, modules.py:36-43): the (N, C, L) input is viewed as
    (N, C, L, 1) so the existing fused conv covariance kernels apply
    unchanged with kw = sw = 1, pw = 0.
    """

    def __init__(self, module: torch.nn.Conv1d):
        self.module = module

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        (k,) = cast(tuple, self.module.kernel_size)
        n = self.module.in_channels * k + int(self.has_bias())
        return (n, n)

    def accumulate_a_factor(
        self,
        a: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        (k,) = cast(tuple, self.module.kernel_size)
        (s,) = cast(tuple, self.module.stride)
        (p,) = cast(tuple, self.module.padding)
        ops.cov_conv_a(
            a.unsqueeze(-1).contiguous(),
            kernel_size=(k, 1),
            stride=(s, 1),
            padding=(p, 0),
            bias=self.has_bias(),
            out=out,
            beta=beta,
            coeff_scale=coeff_scale,
        )

    def accumulate_g_factor(
        self,
        g: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        ops.cov_conv_g(
            g.unsqueeze(-1).contiguous(),
            out=out,
            beta=beta,
            coeff_scale=coeff_scale,
        )


class Conv3dModuleHelper(ModuleHelper):
    """Adapter for torch.nn.Conv3d (extension beyond the reference).

    Patch extraction (3D im2col) runs as torch unfold views; the heavy
    covariance SYRK still routes through ops.cov_linear (the fused HIP
    kernel on GPU) with the conv coefficient 1/(M*s^2) — same semantics
    as Conv2d: [patches, ones] / spatial_size, cov scaled by rows.
    """

    def __init__(self, module: torch.nn.Conv3d):
        self.module = module

    @property
    def a_factor_shape(self) -> tuple[int, int]:
        kd, kh, kw = cast(tuple, self.module.kernel_size)
        n = self.module.in_channels * kd * kh * kw + int(self.has_bias())
        return (n, n)

    @property
    def g_factor_shape(self) -> tuple[int, int]:
        n = self.module.out_channels
        return (n, n)

    def _patches(self, x: torch.Tensor) -> tuple[torch.Tensor, int]:
        kd, kh, kw = cast(tuple, self.module.kernel_size)
        sd, sh, sw = cast(tuple, self.module.stride)
        pd, ph, pw = cast(tuple, self.module.padding)
        if pd or ph or pw:
            x = torch.nn.functional.pad(x, (pw, pw, ph, ph, pd, pd))
        p = x.unfold(2, kd, sd).unfold(3, kh, sh).unfold(4, kw, sw)
        # (N, C, OD, OH, OW, kd, kh, kw) -> (N, OD, OH, OW, C, kd, kh, kw)
        p = p.permute(0, 2, 3, 4, 1, 5, 6, 7)
        spatial = p.size(1) * p.size(2) * p.size(3)
        flat = p.reshape(-1, p.size(4) * p.size(5) * p.size(6) * p.size(7))
        return flat.contiguous(), spatial

    def accumulate_a_factor(
        self,
        a: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        flat, s = self._patches(a)
        m = flat.size(0)
        ops.cov_linear(
            flat,
            bias=self.has_bias(),
            out=out,
            beta=beta,
            coeff=coeff_scale / (m * s * s),
        )

    def accumulate_g_factor(
        self,
        g: torch.Tensor,
        out: torch.Tensor,
        beta: float,
        coeff_scale: float,
    ) -> None:
        s = g.size(2) * g.size(3) * g.size(4)
        rows = g.permute(0, 2, 3, 4, 1).reshape(-1, g.size(1)).contiguous()
        m = rows.size(0)
        ops.cov_linear(
            rows,
            bias=False,
            out=out,
            beta=beta,
            coeff=coeff_scale / (m * s * s),
        )
