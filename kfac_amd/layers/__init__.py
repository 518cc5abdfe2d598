"""K-FAC layer implementations."""

from kfac_amd.layers.base import KFACBaseLayer
from kfac_amd.layers.eigen import KFACEigenLayer
from kfac_amd.layers.inverse import KFACInverseLayer
from kfac_amd.layers.modules import Conv2dModuleHelper
from kfac_amd.layers.modules import LinearModuleHelper
from kfac_amd.layers.modules import ModuleHelper
from kfac_amd.layers.register import register_modules

__all__ = [
    'KFACBaseLayer',
    'KFACEigenLayer',
    'KFACInverseLayer',
    'ModuleHelper',
    'LinearModuleHelper',
    'Conv2dModuleHelper',
    'register_modules',
]
