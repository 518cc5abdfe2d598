"""Parallelism subpackage: placement, communication, topology.

Convenience re-exports — the implementations live in their parity-named
modules (kfac_amd.assignment / kfac_amd.distributed / kfac_amd.gpt_neox)
so users of the reference package find them under the same names.
"""

from kfac_amd.assignment import KAISAAssignment
from kfac_amd.assignment import WorkAssignment
from kfac_amd.distributed import TorchDistributedCommunicator
from kfac_amd.gpt_neox.assignment import GPTNeoXAssignment
from kfac_amd.gpt_neox.topology import PipeModelDataTopology

__all__ = [
    'KAISAAssignment',
    'WorkAssignment',
    'TorchDistributedCommunicator',
    'GPTNeoXAssignment',
    'PipeModelDataTopology',
]
