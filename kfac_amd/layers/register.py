"""Walk a model and build K-FAC layers for supported modules.

Parity with reference kfac/layers/register.py:20-95: leaf-module walk,
regex skip-list matched against both the module name and class name,
requires_grad exclusion, Linear/Conv2d -> helper mapping.
"""

from __future__ import annotations

import re
from typing import Any

import torch

from kfac_amd.layers.base import KFACBaseLayer
from kfac_amd.layers.modules import Conv2dModuleHelper
from kfac_amd.layers.modules import LinearModuleHelper
from kfac_amd.layers.modules import ModuleHelper

KNOWN_MODULES = {'linear', 'conv2d'}


def get_flattened_modules(
    root: torch.nn.Module,
) -> list[tuple[str, torch.nn.Module]]:
    """All leaf modules of ``root`` as (name, module) pairs."""
    return [
        (name, module)
        for name, module in root.named_modules()
        if len(list(module.children())) == 0
    ]


def requires_grad(module: torch.nn.Module) -> bool:
    """True if every parameter of ``module`` requires grad.

    Mixed requires_grad within a module is unsupported
    (reference register.py:31-33).
    """
    return all(p.requires_grad for p in module.parameters())


def get_module_helper(module: torch.nn.Module) -> ModuleHelper | None:
    """Map a module to its ModuleHelper, or None if unsupported."""
    if isinstance(module, torch.nn.Linear):
        return LinearModuleHelper(module)
    if isinstance(module, torch.nn.Conv2d):
        if module.groups != 1:
            # Grouped/depthwise convs have block-diagonal Kronecker
            # factors the K-FAC formulation here (and the reference,
            # which crashes on them mid-precondition) does not model:
            # skip cleanly so the layer trains unpreconditioned.
            import warnings

            warnings.warn(
                f'K-FAC does not support grouped Conv2d '
                f'(groups={module.groups}); layer will not be '
                'preconditioned.',
                stacklevel=2,
            )
            return None
        if module.dilation != (1, 1) or not isinstance(
            module.padding, (tuple, list),
        ):
            # String padding ('same'/'valid') has no (ph, pw) for the
            # fused im2col, and dilated kernels would need dilated patch
            # extraction — neither is modeled, so skip with a warning
            # (same guard as the Conv1d/Conv3d helpers) rather than
            # computing a silently wrong A factor.
            import warnings

            warnings.warn(
                f'K-FAC Conv2d supports dilation=(1, 1) and numeric '
                f'padding only (got dilation={module.dilation}, '
                f'padding={module.padding!r}); layer will not be '
                'preconditioned.',
                stacklevel=2,
            )
            return None
        return Conv2dModuleHelper(module)
    if isinstance(module, torch.nn.Conv1d):
        if module.groups != 1 or not isinstance(
            module.padding, (tuple, list),
        ):
            return None
        from kfac_amd.layers.modules import Conv1dModuleHelper

        return Conv1dModuleHelper(module)
    if isinstance(module, torch.nn.Conv3d):
        if module.groups != 1 or not isinstance(
            module.padding, (tuple, list),
        ):
            return None
        from kfac_amd.layers.modules import Conv3dModuleHelper

        return Conv3dModuleHelper(module)
    return None


def any_match(query: str, patterns: list[str]) -> bool:
    """True if any regex in ``patterns`` fully matches ``query``."""
    return any(re.fullmatch(p, query) is not None for p in patterns)


def register_modules(
    model: torch.nn.Module,
    kfac_layer_type: type[KFACBaseLayer],
    skip_layers: list[str],
    **layer_kwargs: Any,
) -> dict[torch.nn.Module, tuple[str, KFACBaseLayer]]:
    """Build a KFAC layer for every supported leaf module.

    Args:
        model: the model to register.
        kfac_layer_type: KFACEigenLayer or KFACInverseLayer.
        skip_layers: regex patterns; a module whose *name* or *class name*
            matches any pattern is skipped (reference register.py:46-54).
        **layer_kwargs: forwarded to the KFAC layer constructor.

    Returns:
        dict mapping module -> (name, KFAC layer).
    """
    modules = get_flattened_modules(model)
    kfac_layers: dict[torch.nn.Module, tuple[str, KFACBaseLayer]] = {}
    for name, module in modules:
        if (
            not any_match(name, skip_layers)
            and not any_match(module.__class__.__name__, skip_layers)
            and requires_grad(module)
        ):
            helper = get_module_helper(module)
            if helper is None:
                continue
            layer = kfac_layer_type(helper, **layer_kwargs)
            # Use non-ambiguous name: strip DDP's 'module.' prefix so
            # state_dict keys match between DDP and bare models.
            clean = name.replace('module.', '', 1) if name.startswith('module.') else name
            kfac_layers[module] = (clean, layer)
    return kfac_layers
