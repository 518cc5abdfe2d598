"""Asynchronous collective communication for K-FAC.

Parity surface with reference kfac/distributed.py:124-465
(TorchDistributedCommunicator, bucketed allreduce, triu wire format,
rank/world helpers), re-designed for RCCL over xGMI:

- ``torch.distributed`` with backend "nccl" IS RCCL on ROCm; async ops run
  on RCCL's side HIP stream, so factor allreduces launched from backward
  hooks overlap the remaining backward compute.
- Averages are implemented by pre-scaling the send buffer by 1/world
  instead of a future callback multiply (reference distributed.py:190-246):
  a callback runs on a host callback thread and costs a host round trip per
  tensor; the pre-scale fuses into the same pass that packs the bucket.
- Buckets are keyed by the *rank set* of the group, fixing the latent
  size-keying bug noted in the reference (distributed.py:376-378) where two
  distinct groups of equal size would share a bucket.
- Symmetric factors can be sent triu-packed (half the bytes): on the
  latency-bound small-factor allreduces over 7 xGMI point-to-point links,
  wire bytes are the per-link bound.
"""

from __future__ import annotations

from typing import Callable

import torch
import torch.distributed as dist

from kfac_amd import ops


class NonSquareTensorError(Exception):
    """Raised when a symmetric op receives a non-square tensor."""


def get_triu(tensor: torch.Tensor) -> torch.Tensor:
    """Upper-triangle wire format (reference distributed.py:422-446)."""
    if tensor.dim() != 2 or tensor.size(0) != tensor.size(1):
        raise NonSquareTensorError(
            f'get_triu requires a square tensor, got {tuple(tensor.shape)}',
        )
    return ops.triu_pack(tensor)


def fill_triu(
    shape: tuple[int, ...],
    triu_tensor: torch.Tensor,
) -> torch.Tensor:
    """Rebuild the symmetric matrix from its packed upper triangle
    (reference distributed.py:448-465)."""
    if len(shape) != 2 or shape[0] != shape[1]:
        raise NonSquareTensorError(f'fill_triu requires a square shape, got {shape}')
    return ops.triu_unpack(triu_tensor, shape[0])


def get_rank(group: dist.ProcessGroup | None = None) -> int:
    """Rank of this process (0 if torch.distributed is uninitialized)."""
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(group)
    return 0


def get_world_size(group: dist.ProcessGroup | None = None) -> int:
    """World size (1 if torch.distributed is uninitialized)."""
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(group)
    return 1


class Future:
    """Handle for an in-flight collective; ``wait()`` returns the tensor.

    Mirrors the reference's Tensor|Future attribute pattern
    (kfac/layers/base.py:94-128): layer state can hold either a concrete
    tensor or one of these; the property getter waits on first read.
    """

    def __init__(
        self,
        work: dist.Work | None,
        result: torch.Tensor,
        post: Callable[[torch.Tensor], torch.Tensor] | None = None,
    ) -> None:
        self._work = work
        self._result = result
        self._post = post
        self._done = False

    def wait(self) -> torch.Tensor:
        """Block until the collective completes; return the tensor."""
        if not self._done:
            if self._work is not None:
                self._work.wait()
            if self._post is not None:
                self._result = self._post(self._result)
            self._done = True
        return self._result


class AllreduceTensorBucket:
    """One flat allreduce bucket for a single (group, dtype, device).

    Tensors are packed into one flat buffer, allreduced together, and
    unpacked into their original storage on first wait. Packing via a
    single ``torch.cat`` launch; entries keep views into the flat buffer.
    """

    def __init__(self, cap_bytes: int) -> None:
        self._cap = cap_bytes
        self._tensors: list[torch.Tensor] = []
        self._bytes = 0
        self._flat: torch.Tensor | None = None
        self._work: dist.Work | None = None
        self._unpacked = False

    @property
    def size(self) -> int:
        """Bytes currently in the bucket (reference distributed.py:60)."""
        return self._bytes

    def communicated(self) -> bool:
        """True once the bucket's allreduce has been launched."""
        return self._flat is not None

    def add_tensor(self, tensor: torch.Tensor) -> int:
        """Reference-compatible alias of append()."""
        return self.append(tensor)

    def fits(self, tensor: torch.Tensor) -> bool:
        return self._bytes + tensor.numel() * tensor.element_size() <= self._cap

    def append(self, tensor: torch.Tensor) -> int:
        if self._flat is not None:
            raise RuntimeError('bucket already communicated')
        self._tensors.append(tensor)
        self._bytes += tensor.numel() * tensor.element_size()
        return len(self._tensors) - 1

    def communicate(self, group: dist.ProcessGroup | None, scale: float) -> None:
        if self._flat is not None:
            raise RuntimeError('bucket communicated twice')
        flats = [t.reshape(-1) for t in self._tensors]
        self._flat = torch.cat(flats)
        if scale != 1.0:
            self._flat.mul_(scale)
        if dist.is_available() and dist.is_initialized():
            self._work = dist.all_reduce(
                self._flat, group=group, async_op=True,
            )
        else:
            # world of one (same fallback as get_world_size): the scaled
            # flat buffer IS the result
            self._work = None

    def wait_and_unpack(self) -> None:
        if self._unpacked:
            return
        if self._flat is None:
            raise RuntimeError('bucket waited before communicate()')
        if self._work is not None:
            self._work.wait()
        if (
            self._flat.is_cuda
            and self._flat.dtype == torch.float32
            and all(t.is_contiguous() for t in self._tensors)
        ):
            # fused scatter (K13): one kernel instead of one copy
            # launch per member tensor
            ext = ops._load_ext()
            if ext is not None:
                ext.bucket_unpack(self._flat, self._tensors)
                self._unpacked = True
                return
        offset = 0
        for t in self._tensors:
            n = t.numel()
            t.copy_(self._flat[offset : offset + n].view_as(t))
            offset += n
        self._unpacked = True


class _BucketFuture:
    """Future for a single tensor inside a bucket."""

    def __init__(
        self,
        tensor: torch.Tensor,
        comm: TorchDistributedCommunicator,
        key: tuple,
        post: Callable[[torch.Tensor], torch.Tensor] | None = None,
    ) -> None:
        self._tensor = tensor
        self._comm = comm
        self._key = key
        self._bucket: AllreduceTensorBucket | None = None
        self._post = post
        self._done = False

    def _attach(self, bucket: AllreduceTensorBucket) -> None:
        self._bucket = bucket

    def wait(self) -> torch.Tensor:
        if not self._done:
            assert self._bucket is not None
            if self._bucket._flat is None:
                # Bucket not launched yet: flush this key so the data is
                # in flight, then wait.
                self._comm._flush_key(self._key)
            self._bucket.wait_and_unpack()
            if self._post is not None:
                self._tensor = self._post(self._tensor)
            self._done = True
        return self._tensor


class TorchDistributedCommunicator:
    """Async allreduce/broadcast with bucketing and triu packing."""

    def __init__(self, bucket_cap_mb: float = 25.0) -> None:
        """Init communicator.

        Args:
            bucket_cap_mb: max flat-bucket size in MiB. On one MI355X node
                factors are small (sum over ResNet-50 ≈ 120 MB fp32), so a
                25 MiB bucket gives a handful of large collectives per
                factor step — large enough to be bandwidth- not
                latency-bound on xGMI rings.
        """
        self._cap_bytes = int(bucket_cap_mb * 1024 * 1024)
        # key -> currently-open (not yet launched) bucket
        self._open: dict[tuple, AllreduceTensorBucket] = {}
        # launched buckets kept alive until their tensors are unpacked
        self._inflight: list[AllreduceTensorBucket] = []
        # group handle + averaging flag per bucket key
        self._groups: dict[tuple, tuple] = {}

    @property
    def bucket_cap_bytes(self) -> int:
        """Flat-bucket size cap in bytes."""
        return self._cap_bytes

    def group_ranks(self, group: dist.ProcessGroup | None) -> frozenset[int]:
        """Membership of a process group.

        Keyed by actual rank membership — the reference's version keyed
        by group SIZE (distributed.py:376-378), silently sharing buckets
        between distinct equal-sized groups.
        """
        if group is None or not (dist.is_available() and dist.is_initialized()):
            return frozenset()
        return frozenset(dist.get_process_group_ranks(group))

    # -- plain collectives -------------------------------------------------

    def allreduce(
        self,
        tensor: torch.Tensor,
        *,
        average: bool = True,
        group: dist.ProcessGroup | None = None,
        symmetric: bool = False,
    ) -> torch.Tensor | Future:
        """Async allreduce (optionally averaged / triu-packed).

        Returns the tensor directly if world size is 1.
        """
        world = get_world_size(group)
        if world <= 1:
            return tensor
        if symmetric:
            if tensor.dim() != 2 or tensor.size(0) != tensor.size(1):
                raise NonSquareTensorError(
                    f'symmetric allreduce of non-square tensor '
                    f'{tuple(tensor.shape)}',
                )
            n = tensor.size(0)
            packed = ops.triu_pack(tensor)
            if average:
                packed.div_(world)
            work = dist.all_reduce(packed, group=group, async_op=True)
            return Future(work, packed, post=lambda p: ops.triu_unpack(p, n))
        send = tensor.div_(world) if average else tensor
        work = dist.all_reduce(send, group=group, async_op=True)
        return Future(work, send)

    def broadcast(
        self,
        tensor: torch.Tensor,
        *,
        src: int,
        group: dist.ProcessGroup | None = None,
        symmetric: bool = False,
    ) -> torch.Tensor | Future:
        """Async broadcast (optionally triu-packed)."""
        if get_world_size(group) <= 1:
            return tensor
        if symmetric:
            if tensor.dim() != 2 or tensor.size(0) != tensor.size(1):
                raise NonSquareTensorError(
                    f'symmetric broadcast of non-square tensor '
                    f'{tuple(tensor.shape)}',
                )
            n = tensor.size(0)
            packed = ops.triu_pack(tensor)
            work = dist.broadcast(packed, src=src, group=group, async_op=True)
            return Future(work, packed, post=lambda p: ops.triu_unpack(p, n))
        work = dist.broadcast(tensor, src=src, group=group, async_op=True)
        return Future(work, tensor)

    # -- bucketed allreduce ------------------------------------------------

    def _group_key(self, group: dist.ProcessGroup | None) -> tuple:
        if group is None or not (dist.is_available() and dist.is_initialized()):
            return (None,)
        ranks = tuple(sorted(dist.get_process_group_ranks(group)))
        return (ranks,)

    def allreduce_bucketed(
        self,
        tensor: torch.Tensor,
        *,
        average: bool = True,
        group: dist.ProcessGroup | None = None,
        symmetric: bool = False,
    ) -> torch.Tensor | Future:
        """Append tensor to the bucket for ``group``; future resolves after
        the bucket's fused allreduce completes.

        The bucket is launched when it reaches the size cap or at
        ``flush_allreduce_buckets()`` (called by the preconditioner after
        the last factor of the step is produced).
        """
        world = get_world_size(group)
        if world <= 1:
            return tensor
        # average is part of the key: the scale is applied per BUCKET at
        # launch, so tensors with different averaging must not share one
        # (otherwise the flag recorded by the bucket's first tensor would
        # silently mis-scale later appends).
        key = self._group_key(group) + (tensor.dtype, tensor.device, average)
        if symmetric:
            if tensor.dim() != 2 or tensor.size(0) != tensor.size(1):
                raise NonSquareTensorError(
                    f'symmetric allreduce of non-square tensor '
                    f'{tuple(tensor.shape)}',
                )
            n = tensor.size(0)
            payload = ops.triu_pack(tensor)
            fut = _BucketFuture(
                payload, self, key, post=lambda p: ops.triu_unpack(p, n),
            )
        else:
            payload = tensor
            fut = _BucketFuture(payload, self, key)

        bucket = self._open.get(key)
        if bucket is not None and not bucket.fits(payload):
            self._launch(key, group, average)
            bucket = None
        if bucket is None:
            bucket = AllreduceTensorBucket(self._cap_bytes)
            self._open[key] = bucket
            self._groups[key] = (group, average)
        bucket.append(payload)
        fut._attach(bucket)
        return fut

    def _launch(self, key: tuple, group: dist.ProcessGroup | None, average: bool) -> None:
        bucket = self._open.pop(key, None)
        if bucket is None:
            return
        scale = 1.0 / get_world_size(group) if average else 1.0
        bucket.communicate(group, scale)
        self._inflight.append(bucket)

    def _flush_key(self, key: tuple) -> None:
        if key in self._open and key in self._groups:
            group, average = self._groups[key]
            self._launch(key, group, average)

    def flush_allreduce_buckets(self) -> None:
        """Launch every open bucket (trailing partial buckets) and drop
        references to buckets whose tensors have been unpacked."""
        for key in list(self._open.keys()):
            group, average = self._groups[key]
            self._launch(key, group, average)
        self._inflight = [b for b in self._inflight if not b._unpacked]
