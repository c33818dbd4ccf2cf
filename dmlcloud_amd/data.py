"""Deterministic dataset sharding and batch utilities.

Capability parity with reference dmlcloud/util/data.py:11-341:
rank-strided sharding, chunked sharding with overlap (temporal windows),
xarray dataset sharding (optional dependency, duck-typed), DataLoader
worker-id folding (worker rank = rank * num_workers + worker_id),
prefetch/batch wrappers, and batch interleaving.

MI355X-native difference: on device tensors, interleave_batches /
interleave_dict_batches execute as ONE descriptor-table gather kernel
(ops/csrc/copy.hip) instead of the reference's N^2 Python slice-copy loop
(reference data.py:295-301), and pinned staging enables async H2D on a
side stream.
"""

from concurrent.futures import ThreadPoolExecutor
from typing import Iterable, List, Optional, Sequence

import numpy as np
import torch
import torch.distributed as dist
from torch.utils.data import IterableDataset, get_worker_info

from . import ops

try:
    import xarray as xr

    _HAS_XARRAY = True
except ImportError:
    xr = None
    _HAS_XARRAY = False


def shard_indices(
    num_elements: int,
    rank: int,
    world_size: int,
    shuffle: bool = False,
    even_shards: bool = True,
    seed: int = 0,
) -> List[int]:
    """Strided [rank::world_size] sharding.

    even_shards: every worker receives the same number of elements; the
    trailing remainder is dropped.
    """
    indices = np.arange(num_elements)
    if shuffle:
        np.random.Generator(np.random.MT19937(seed)).shuffle(indices)
    if even_shards:
        indices = indices[: num_elements - num_elements % world_size]
    return indices[rank::world_size].tolist()


def chunk_and_shard_indices(
    num_elements: int,
    chunk_size: int,
    rank: int,
    world_size: int,
    chunk_overlap: int = 0,
    even_shards: bool = True,
    equal_chunks: bool = True,
    shuffle: bool = False,
    seed: int = 0,
):
    """Shard half-open (start, start+chunk_size+overlap) windows across
    ranks — e.g. temporal windows of weather data."""
    if equal_chunks:
        num_chunks = num_elements // chunk_size
    else:
        num_chunks = (num_elements + chunk_size - 1) // chunk_size

    chunk_indices = shard_indices(num_chunks, rank, world_size, shuffle=shuffle, even_shards=even_shards, seed=seed)
    chunks = []
    for chunk_idx in chunk_indices:
        start = chunk_idx * chunk_size
        end = start + chunk_size + chunk_overlap
        chunks.append((start, end))
    return chunks


def shard_sequence(
    sequence: Sequence,
    rank: int,
    world_size: int,
    shuffle: bool = False,
    even_shards: bool = True,
    seed: int = 0,
):
    indices = shard_indices(len(sequence), rank, world_size, shuffle=shuffle, even_shards=even_shards, seed=seed)
    return [sequence[i] for i in indices]


def sharded_xr_dataset(
    ds,
    dim: str,
    chunk_size: int,
    chunk_overlap: int = 0,
    even_shards: bool = True,
    equal_chunks: bool = True,
    shuffle: bool = False,
    seed: int = 0,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    process_group=None,
    load: bool = False,
    load_kwargs: Optional[dict] = None,
) -> Iterable:
    """Yield rank-sharded chunks of an xarray Dataset/DataArray along `dim`.

    Duck-typed: any object with ``len(ds[dim])`` and ``.isel({dim: slice})``
    works, so xarray itself is an optional dependency.
    """
    if rank is None:
        rank = dist.get_rank(process_group)
    if world_size is None:
        world_size = dist.get_world_size(process_group)

    num_elements = len(ds[dim])
    chunks = chunk_and_shard_indices(
        num_elements,
        chunk_size,
        rank,
        world_size,
        chunk_overlap=chunk_overlap,
        even_shards=even_shards,
        equal_chunks=equal_chunks,
        shuffle=shuffle,
        seed=seed,
    )
    for start, end in chunks:
        chunk = ds.isel({dim: slice(start, end)})
        if load:
            chunk.load(**(load_kwargs or {}))
        yield chunk


def _effective_rank(rank: int, world_size: int):
    """Fold the DataLoader worker id into the rank so loader workers shard
    disjointly: worker rank = rank * num_workers + worker_id."""
    worker_info = get_worker_info()
    if worker_info is None:
        return rank, world_size
    return rank * worker_info.num_workers + worker_info.id, world_size * worker_info.num_workers


class ShardedSequenceDataset(IterableDataset):
    def __init__(
        self,
        sequence: Sequence,
        shuffle: bool = False,
        even_shards: bool = True,
        seed: int = 0,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
    ):
        self.sequence = sequence
        self.shuffle = shuffle
        self.even_shards = even_shards
        self.seed = seed
        self.rank = rank if rank is not None else dist.get_rank()
        self.world_size = world_size if world_size is not None else dist.get_world_size()
        self.epoch = 0

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        n = len(self.sequence)
        if self.even_shards:
            return (n - n % self.world_size) // self.world_size
        return (n - self.rank + self.world_size - 1) // self.world_size

    def __iter__(self):
        rank, world_size = _effective_rank(self.rank, self.world_size)
        shards = shard_sequence(
            self.sequence,
            rank,
            world_size,
            shuffle=self.shuffle,
            even_shards=self.even_shards,
            seed=self.seed + self.epoch,
        )
        return iter(shards)


class ShardedXrDataset(IterableDataset):
    def __init__(
        self,
        ds,
        dim: str,
        chunk_size: int,
        chunk_overlap: int = 0,
        even_shards: bool = True,
        equal_chunks: bool = True,
        shuffle: bool = False,
        seed: int = 0,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
        process_group=None,
        load: bool = False,
        load_kwargs: Optional[dict] = None,
    ):
        self.ds = ds
        self.dim = dim
        self.chunk_size = chunk_size
        self.chunk_overlap = chunk_overlap
        self.even_shards = even_shards
        self.equal_chunks = equal_chunks
        self.shuffle = shuffle
        self.seed = seed
        self.load = load
        self.load_kwargs = load_kwargs

        self.rank = rank if rank is not None else dist.get_rank(process_group)
        self.world_size = world_size if world_size is not None else dist.get_world_size(process_group)
        self._num_iters = 0

    def set_epoch(self, epoch: int):
        self._num_iters = epoch

    def __iter__(self):
        rank, world_size = _effective_rank(self.rank, self.world_size)
        return sharded_xr_dataset(
            self.ds,
            self.dim,
            self.chunk_size,
            chunk_overlap=self.chunk_overlap,
            even_shards=self.even_shards,
            equal_chunks=self.equal_chunks,
            shuffle=self.shuffle,
            seed=self.seed + self._num_iters,
            rank=rank,
            world_size=world_size,
            load=self.load,
            load_kwargs=self.load_kwargs,
        )


class DownstreamDataset(IterableDataset):
    def __init__(self, source_ds: Iterable):
        self.source_ds = source_ds

    def set_epoch(self, epoch: int):
        if hasattr(self.source_ds, 'set_epoch'):
            self.source_ds.set_epoch(epoch)

    def __len__(self):
        return len(self.source_ds)


class PrefetchDataset(DownstreamDataset):
    """Background-thread lookahead of `num_elements` items."""

    def __init__(self, source_ds: Iterable, num_elements: int):
        super().__init__(source_ds)
        self.num_elements = num_elements

    def __iter__(self):
        pool = ThreadPoolExecutor(max_workers=1)
        iter_ = iter(self.source_ds)
        with pool:
            futures = [pool.submit(next, iter_) for _ in range(self.num_elements)]
            while True:
                future = futures.pop(0)
                try:
                    element = future.result()
                except StopIteration:
                    return
                futures += [pool.submit(next, iter_)]
                yield element


class BatchDataset(DownstreamDataset):
    def __init__(self, source_ds: Iterable, batch_size: int, drop_remainder: bool = False):
        super().__init__(source_ds)
        self.batch_size = batch_size
        self.drop_remainder = drop_remainder

    def __len__(self):
        if self.drop_remainder:
            return len(self.source_ds) // self.batch_size
        return (len(self.source_ds) + self.batch_size - 1) // self.batch_size

    def __iter__(self):
        batch = []
        for element in self.source_ds:
            batch.append(element)
            if len(batch) == self.batch_size:
                yield batch
                batch = []
        if batch and not self.drop_remainder:
            yield batch


def _interleave_group(batches: List[torch.Tensor], memory: torch.Tensor, slice_size: int):
    """memory[i, j*s:(j+1)*s] = batches[j][i*s:(i+1)*s] for all i, j.

    Device tensors: one descriptor-table gather kernel. CPU: torch copies.
    """
    num_batches = len(batches)
    if memory.is_cuda:
        srcs, dsts = [], []
        for i in range(num_batches):
            for j in range(num_batches):
                srcs.append(batches[j][i * slice_size : (i + 1) * slice_size].contiguous())
                dsts.append(memory[i, j * slice_size : (j + 1) * slice_size])
        ops.chunked_copy(srcs, dsts)
    else:
        for i in range(num_batches):
            for j in range(num_batches):
                memory[i, j * slice_size : (j + 1) * slice_size] = batches[j][i * slice_size : (i + 1) * slice_size]


def interleave_batches(
    iterable: Iterable[torch.Tensor], num_batches: int, pin_memory: bool = False
) -> Iterable[torch.Tensor]:
    """Re-pack N consecutive batches into N interleaved batches.

    Returned batches must be used immediately or copied (they view a
    shared staging buffer).
    """
    if num_batches < 1:
        raise ValueError('num_batches must be greater than 0')
    if num_batches == 1:
        yield from iterable
        return

    batches = []
    memory = None
    slice_size = None
    for batch in iterable:
        if memory is None:
            batch_size = batch.shape[0]
            slice_size = batch_size // num_batches
            if batch_size % num_batches != 0:
                raise ValueError(f'Batch dimension ({batch_size}) must be divisible by num_batches={num_batches}')
            memory = torch.empty(
                (num_batches, *batch.shape), dtype=batch.dtype, device=batch.device, pin_memory=pin_memory
            )
        batches.append(batch)
        if len(batches) == num_batches:
            _interleave_group(batches, memory, slice_size)
            batches = []
            for i in range(num_batches):
                yield memory[i]


def interleave_dict_batches(
    iterable: Iterable[dict], num_batches: int, pin_memory: bool = False
) -> Iterable[dict]:
    """interleave_batches for dict-of-tensor batches."""
    if num_batches < 1:
        raise ValueError('num_batches must be greater than 0')
    if num_batches == 1:
        yield from iterable
        return

    batches = []
    memory = {}
    slice_size = {}
    for batch in iterable:
        if not memory:
            for k, tensor in batch.items():
                batch_size = tensor.shape[0]
                if batch_size % num_batches != 0:
                    raise ValueError(f'Batch dimension ({batch_size}) must be divisible by num_batches={num_batches}')
                slice_size[k] = batch_size // num_batches
                memory[k] = torch.empty(
                    (num_batches, *tensor.shape), dtype=tensor.dtype, device=tensor.device, pin_memory=pin_memory
                )
        batches.append(batch)
        if len(batches) == num_batches:
            for k in memory:
                _interleave_group([b[k] for b in batches], memory[k], slice_size[k])
            batches = []
            for i in range(num_batches):
                yield {k: memory[k][i] for k in memory}
