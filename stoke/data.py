# -*- coding: utf-8 -*-
"""Data loading: device-placement DataLoader shim + bucketed distributed sampler.

API mirrors the reference (``stoke/data.py:24-516``); the sampler is a fresh
implementation with the same guarantees: per-replica slices are disjoint,
every emitted batch comes from one bucket (similar sample lengths -> minimal
padding waste), deterministic shuffling via (seed + epoch), and equal sample
counts per replica.
"""

import itertools
from math import ceil
from typing import Dict, Iterator, List, Optional, Tuple, Union

import numpy as np
import torch
import torch.distributed as dist
from torch.utils.data import DataLoader as DL
from torch.utils.data import Dataset
from torch.utils.data.distributed import Sampler

from stoke.utils import T_co, _collate_fn_t, _worker_init_fn_t, place_data_on_gpu


class StokeDataLoader(DL):
    """torch DataLoader that places yielded batches on the training device.

    Device/precision flags are injected by ``Stoke.DataLoader`` so user code
    never handles placement (reference ``stoke/data.py:24-108``).
    """

    def __init__(
        self, dataset: Dataset[T_co], gpu: bool, fp16: Optional[str], **kwargs
    ):
        super().__init__(dataset, **kwargs)
        self._gpu = gpu
        self._fp16 = fp16

    def __iter__(self):
        for val in super().__iter__():
            yield val if not self._gpu else place_data_on_gpu(val, self._fp16)


class BucketedDistributedSampler(Sampler[T_co]):
    """Distributed sampler that batches similar-length samples together.

    Given ``sorted_idx`` (dataset indices sorted by the bucketing key, e.g.
    sequence length), the index space is split into ``buckets`` contiguous
    ranges; each batch is drawn from within one bucket so padding waste stays
    low, while bucket and slice order are shuffled every epoch.

    Constructor/attribute surface matches the reference
    (``stoke/data.py:111-266``).
    """

    def __init__(
        self,
        dataset: Dataset,
        buckets: int,
        batch_size: int,
        sorted_idx: List,
        backend=None,
        allow_bucket_overlap: bool = False,
        num_replicas: Optional[int] = None,
        rank: Optional[int] = None,
        shuffle: bool = True,
        seed: int = 0,
        drop_last: bool = False,
        info_rank: int = 0,
    ) -> None:
        num_replicas, rank = self._resolve_world(num_replicas, rank)
        self.num_replicas = num_replicas
        self.rank = rank
        self.epoch = 0
        self.drop_last = drop_last
        self.shuffle = shuffle
        self.seed = seed
        self.buckets = buckets
        self.sorted_n_samples = sorted_idx
        self.batch_size = batch_size
        self.allow_bucket_overlap = allow_bucket_overlap
        # One "slice" feeds every replica one batch
        self.slice_size = self.batch_size * self.num_replicas
        self.num_samples_per_bucket = self._get_size(
            len(dataset), self.buckets, self.drop_last
        )
        self.num_slices_per_bucket = self._get_size(
            self.num_samples_per_bucket, self.slice_size, self.drop_last
        )
        if self.num_samples_per_bucket < self.slice_size:
            raise ValueError(
                f"Stoke -- Resulting number of slices (batch * replicas) per bucket "
                f"({self.num_samples_per_bucket}) is less than the batch size "
                f"({self.batch_size})"
            )
        if self.num_slices_per_bucket < 2:
            raise ValueError(
                f"Stoke -- Number of slices per bucket {self.num_slices_per_bucket} is less than 2 "
                f"which is not recommended"
            )
        if self.num_samples_per_bucket < 100:
            raise ValueError(
                f"Stoke -- Number of samples per bucket {self.num_samples_per_bucket} is less than 100 "
                f"which is not recommended as this might lead to dropping of excessive data"
            )
        self.bucket_idx = [
            list(val) for val in np.array_split(self.sorted_n_samples, self.buckets)
        ]
        self.rounded_num_samples_per_bucket = (
            self.slice_size * self.num_slices_per_bucket
        )
        self.rounded_num_samples_per_replica = (
            self.num_slices_per_bucket * self.batch_size * self.buckets
        )
        # Residual (cross-bucket) batches only exist when drop_last trims the
        # buckets; guarding on drop_last avoids the reference's negative-count
        # edge case when allow_bucket_overlap is set without drop_last.
        if self.allow_bucket_overlap and self.drop_last:
            self.rounded_num_samples_per_replica += (
                (len(dataset) - (self.rounded_num_samples_per_bucket * self.buckets))
                // self.slice_size
            ) * self.batch_size
        if self.rank == info_rank:
            print(
                f"Stoke -- BucketedDistributedSampler -- # Samples Per Bucket: "
                f"{self.rounded_num_samples_per_bucket}, # of Samples Per Replica: "
                f"{self.rounded_num_samples_per_replica}"
            )

    @staticmethod
    def _resolve_world(num_replicas: Optional[int], rank: Optional[int]):
        if num_replicas is None or rank is None:
            if not (dist.is_available() and dist.is_initialized()):
                raise RuntimeError(
                    "Requires distributed package (torch.dist) to be available"
                )
            if num_replicas is None:
                num_replicas = dist.get_world_size()
            if rank is None:
                rank = dist.get_rank()
        return num_replicas, rank

    @staticmethod
    def _get_size(data_len: int, split_var: int, drop_last: bool = False) -> int:
        return data_len // split_var if drop_last else ceil(data_len / split_var)

    def __iter__(self) -> Iterator[T_co]:
        g = torch.Generator()
        g.manual_seed(self.seed + self.epoch)
        # Per-bucket shuffle
        if self.shuffle:
            indices = [
                [val[i] for i in torch.randperm(len(val), generator=g).tolist()]
                for val in self.bucket_idx
            ]
        else:
            indices = [list(val) for val in self.bucket_idx]
        # Pad each bucket up to a whole number of slices by re-using samples
        # from within the same bucket (keeps batches length-homogeneous)
        slices: List[List] = []
        residual: List = []
        need = self.num_slices_per_bucket * self.slice_size
        for vals in indices:
            if len(vals) < need:
                pad = [vals[i % len(vals)] for i in range(need - len(vals))]
                vals = vals + pad
            elif len(vals) > need:
                residual.extend(vals[need:])
                vals = vals[:need]
            # Cut into slices, take this replica's strided share of each
            for sidx in range(self.num_slices_per_bucket):
                sl = vals[sidx * self.slice_size : (sidx + 1) * self.slice_size]
                slices.append(sl[self.rank : self.slice_size : self.num_replicas])
        # Batch up residual (cross-bucket) samples if allowed
        if self.drop_last and self.allow_bucket_overlap and len(residual) >= self.slice_size:
            for sidx in range(len(residual) // self.slice_size):
                sl = residual[sidx * self.slice_size : (sidx + 1) * self.slice_size]
                slices.append(sl[self.rank : self.slice_size : self.num_replicas])
        # Shuffle slice order (same permutation on every replica)
        if self.shuffle:
            order = torch.randperm(len(slices), generator=g).tolist()
            slices = [slices[i] for i in order]
        final = list(itertools.chain(*slices))
        assert len(final) == self.rounded_num_samples_per_replica, (
            f"sampler produced {len(final)} != {self.rounded_num_samples_per_replica}"
        )
        return iter(final)

    def __len__(self) -> int:
        return self.rounded_num_samples_per_replica

    def set_epoch(self, epoch: int) -> None:
        """Change the shuffling seed each epoch (same contract as torch's
        DistributedSampler)."""
        self.epoch = epoch
