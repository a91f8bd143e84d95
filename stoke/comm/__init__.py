# -*- coding: utf-8 -*-
"""Single process-group abstraction over RCCL (and gloo for CPU CI).

This is the one distributed backend of the framework: where the reference
dispatched to torch DDP / Horovod / DeepSpeed process groups
(``stoke/distributed.py:491-538, 759-784, 1308-1316``) everything here runs on
one ``torch.distributed`` group — backend "nccl" IS RCCL on ROCm, riding the
7-link xGMI mesh inside a node.  gloo is kept for CPU-only tests; collectives
missing on gloo (reduce_scatter_tensor / all_gather_into_tensor) get
functional fallbacks so the shard engines are testable without a GPU.
"""

from contextlib import contextmanager
from typing import List, Optional

import torch
import torch.distributed as dist

from stoke.configs import BackendOptions


def _backend_value(backend) -> str:
    if isinstance(backend, BackendOptions):
        return backend.value
    return str(backend).strip()


class StokeProcessGroup:
    """Thin owner of the global process group + collective helpers.

    Rendezvous keeps the reference's env:// + LOCAL_RANK contract
    (``status.py:511-539``) so ``torchrun`` works unchanged; optional mpi4py
    discovery covers ``mpirun`` launches.
    """

    def __init__(
        self,
        backend: str = "nccl",
        init_method: str = "env://",
        local_rank: Optional[int] = None,
        auto_mpi_discovery: bool = False,
    ):
        self.backend = _backend_value(backend)
        self.init_method = init_method
        if auto_mpi_discovery and not dist.is_initialized():
            self._mpi_discovery()
        if not dist.is_initialized():
            dist.init_process_group(backend=self.backend, init_method=init_method)
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.local_rank = local_rank if local_rank is not None else self.rank
        if self.backend == "nccl":
            torch.cuda.set_device(self.local_rank)
            self.device = torch.device("cuda", self.local_rank)
        else:
            self.device = torch.device("cpu")

    @staticmethod
    def _mpi_discovery():
        """Populate torch.distributed env vars from an MPI launch (mpirun parity)."""
        import os

        try:
            from mpi4py import MPI
        except ImportError:
            return
        comm = MPI.COMM_WORLD
        os.environ.setdefault("RANK", str(comm.Get_rank()))
        os.environ.setdefault("WORLD_SIZE", str(comm.Get_size()))
        # Rank 0's hostname is broadcast as the master address
        master = comm.bcast(os.uname()[1] if comm.Get_rank() == 0 else None, root=0)
        os.environ.setdefault("MASTER_ADDR", master)
        os.environ.setdefault("MASTER_PORT", "29500")

    # ----------------------------------------------------------- collectives
    def barrier(self):
        if self.backend == "nccl":
            dist.barrier(device_ids=[self.local_rank])
        else:
            dist.barrier()

    def all_reduce(self, tensor: torch.Tensor, async_op: bool = False):
        return dist.all_reduce(tensor, op=dist.ReduceOp.SUM, async_op=async_op)

    def broadcast(self, tensor: torch.Tensor, src: int = 0, async_op: bool = False):
        return dist.broadcast(tensor, src=src, async_op=async_op)

    def reduce(self, tensor: torch.Tensor, dst: int, async_op: bool = False):
        return dist.reduce(tensor, dst=dst, op=dist.ReduceOp.SUM, async_op=async_op)

    def reduce_scatter_flat(
        self, output: torch.Tensor, input_flat: torch.Tensor, async_op: bool = False
    ):
        """SUM-reduce ``input_flat`` (world_size * shard) and scatter shards.

        The shard engines are built reduce-scatter-first: on the full xGMI
        mesh each reduce-scatter phase spreads traffic over all 7 links where
        a ring all-reduce is single-link-bound.
        """
        if self.backend == "nccl":
            return dist.reduce_scatter_tensor(output, input_flat, async_op=async_op)
        # gloo fallback: all-reduce then local slice (CPU CI only)
        work = dist.all_reduce(input_flat, async_op=async_op)
        shard = input_flat.view(self.world_size, -1)[self.rank]
        if async_op:
            class _Wrap:
                def __init__(self, w, out, sh):
                    self._w, self._out, self._sh = w, out, sh

                def wait(self):
                    self._w.wait()
                    self._out.copy_(self._sh.view_as(self._out))

            return _Wrap(work, output, shard)
        output.copy_(shard.view_as(output))
        return None

    def all_gather_flat(
        self, output_flat: torch.Tensor, input_shard: torch.Tensor, async_op: bool = False
    ):
        """Gather equal shards from every rank into one flat tensor."""
        if self.backend == "nccl":
            return dist.all_gather_into_tensor(
                output_flat, input_shard.contiguous(), async_op=async_op
            )
        chunks = list(output_flat.view(self.world_size, -1).unbind(0))
        work = dist.all_gather(chunks, input_shard.contiguous().view(-1), async_op=async_op)
        return work

    def gather_object(self, obj, dst: int = 0):
        out = [None] * self.world_size if self.rank == dst else None
        dist.gather_object(obj, out, dst=dst)
        return out

    def all_gather_object(self, obj) -> List:
        out = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def broadcast_object(self, obj, src: int = 0):
        box = [obj]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def broadcast_module_states(self, module: torch.nn.Module, src: int = 0,
                                bucket_bytes: int = 256 * 1024 * 1024):
        """Broadcast all parameters and buffers from ``src`` in flat buckets.

        One flat copy + one RCCL broadcast per ~256 MB keeps the launch count
        low on the 7-link mesh instead of per-tensor broadcasts.
        """
        if self.world_size == 1:
            return
        tensors = [p.data for p in module.parameters()] + [
            b.data for b in module.buffers()
        ]
        bucket: List[torch.Tensor] = []
        size = 0
        for t in tensors:
            bucket.append(t)
            size += t.numel() * t.element_size()
            if size >= bucket_bytes:
                self._broadcast_bucket(bucket, src)
                bucket, size = [], 0
        if bucket:
            self._broadcast_bucket(bucket, src)

    def _broadcast_bucket(self, bucket: List[torch.Tensor], src: int):
        # Mixed dtypes cannot be concatenated; group by dtype
        by_dtype = {}
        for t in bucket:
            by_dtype.setdefault(t.dtype, []).append(t)
        for dt, ts in by_dtype.items():
            flat = torch.cat([t.reshape(-1) for t in ts])
            dist.broadcast(flat, src=src)
            offset = 0
            for t in ts:
                n = t.numel()
                t.copy_(flat[offset : offset + n].view_as(t))
                offset += n

    def sync_loss(self, loss: torch.Tensor, device=None):
        """Mean-reduce a scalar loss across ranks, ASYNCHRONOUSLY.

        Returns a float-like :class:`stoke.utils.LazyLoss`: the all-reduce
        is launched async (no barrier, unlike the reference at
        ``distributed.py:619-646``) and the one D2H read happens only when
        a real float is needed — per-microbatch loss tracking no longer
        stalls the host.
        """
        from stoke.utils import LazyLoss

        t = loss.detach().clone().float()
        if self.world_size > 1:
            work = dist.all_reduce(t, async_op=True)
            return LazyLoss(t, work=work, div=self.world_size)
        return LazyLoss(t)

    def clean(self):
        if dist.is_initialized():
            dist.destroy_process_group()
