"""Tensor/pipeline-parallel (GPT-NeoX-style) K-FAC support.

Feature parity with reference kfac/gpt_neox/  without the
DeepSpeed dependency: a self-contained 3D topology description replaces
``PipeModelDataParallelTopology`` and registration matches Megatron-style
``ColumnParallelLinear`` / ``RowParallelLinear`` modules by class name.
"""

from kfac_amd.gpt_neox.assignment import GPTNeoXAssignment
from kfac_amd.gpt_neox.layer import GPTNeoXKFACEigenLayer
from kfac_amd.gpt_neox.modules import GPTNeoXLinearModuleHelper
from kfac_amd.gpt_neox.preconditioner import GPTNeoXKFACPreconditioner
from kfac_amd.gpt_neox.topology import PipeModelDataTopology

__all__ = [
    'GPTNeoXAssignment',
    'GPTNeoXKFACEigenLayer',
    'GPTNeoXLinearModuleHelper',
    'GPTNeoXKFACPreconditioner',
    'PipeModelDataTopology',
]
