"""Sharding determinism and batch utilities (mirrors the reference's
coverage in test/test_data.py including the DataLoader-worker
interleaving contract)."""

import sys

import pytest
import torch
from torch.utils.data import DataLoader

from dmlcloud_amd.data import (
    BatchDataset,
    PrefetchDataset,
    ShardedSequenceDataset,
    chunk_and_shard_indices,
    interleave_batches,
    interleave_dict_batches,
    shard_indices,
    shard_sequence,
    sharded_xr_dataset,
)


class TestShardIndices:
    def test_even(self):
        assert shard_indices(10, 0, 2) == [0, 2, 4, 6, 8]
        assert shard_indices(10, 1, 2) == [1, 3, 5, 7, 9]

    def test_uneven_drops(self):
        # even_shards drops the remainder
        assert shard_indices(11, 0, 2) == [0, 2, 4, 6, 8]
        assert shard_indices(11, 1, 2) == [1, 3, 5, 7, 9]

    def test_uneven_keeps(self):
        assert shard_indices(11, 0, 2, even_shards=False) == [0, 2, 4, 6, 8, 10]
        assert shard_indices(11, 1, 2, even_shards=False) == [1, 3, 5, 7, 9]

    def test_shuffle_deterministic(self):
        a = shard_indices(10, 0, 2, shuffle=True, seed=42)
        b = shard_indices(10, 0, 2, shuffle=True, seed=42)
        c = shard_indices(10, 0, 2, shuffle=True, seed=43)
        assert a == b
        assert a != c

    def test_shuffle_partitions(self):
        parts = [shard_indices(12, r, 3, shuffle=True, seed=7) for r in range(3)]
        combined = sorted(i for p in parts for i in p)
        assert combined == list(range(12))

    def test_world1(self):
        assert shard_indices(4, 0, 1) == [0, 1, 2, 3]


class TestChunkAndShard:
    def test_basic(self):
        chunks = chunk_and_shard_indices(20, 5, 0, 2)
        assert chunks == [(0, 5), (10, 15)]
        chunks = chunk_and_shard_indices(20, 5, 1, 2)
        assert chunks == [(5, 10), (15, 20)]

    def test_overlap(self):
        chunks = chunk_and_shard_indices(20, 5, 0, 2, chunk_overlap=2)
        assert chunks == [(0, 7), (10, 17)]

    def test_unequal_chunks(self):
        chunks = chunk_and_shard_indices(11, 5, 0, 1, equal_chunks=False)
        assert chunks == [(0, 5), (5, 10), (10, 15)]

    def test_equal_chunks_drops(self):
        chunks = chunk_and_shard_indices(11, 5, 0, 1, equal_chunks=True)
        assert chunks == [(0, 5), (5, 10)]


class TestShardSequence:
    def test_basic(self):
        seq = list('abcdefgh')
        assert shard_sequence(seq, 0, 2) == ['a', 'c', 'e', 'g']
        assert shard_sequence(seq, 1, 2) == ['b', 'd', 'f', 'h']


class _FakeXr:
    """Duck-typed stand-in for an xarray DataArray over one dimension."""

    def __init__(self, n, dim='time'):
        self.n = n
        self.dim = dim
        self.data = list(range(n))

    def __getitem__(self, dim):
        assert dim == self.dim
        return self.data

    def isel(self, indexers):
        sl = indexers[self.dim]
        sub = _FakeXr(0, self.dim)
        sub.data = self.data[sl]
        sub.n = len(sub.data)
        return sub

    def load(self):
        return self


def _rank_chunks(n, chunk_size, world_size, **kwargs):
    """[rank][chunk] element lists for every rank of a _FakeXr(n)."""
    ds = _FakeXr(n)
    return [
        [c.data for c in sharded_xr_dataset(ds, 'time', chunk_size, rank=r, world_size=world_size, **kwargs)]
        for r in range(world_size)
    ]


class TestShardedXr:
    """Chunked sharding case matrix (reference test/test_data.py:57-170,
    365-441 — same cases, duck-typed stand-in instead of xarray)."""

    def test_basic(self):
        ds = _FakeXr(20)
        chunks = list(sharded_xr_dataset(ds, 'time', 5, rank=0, world_size=2))
        assert [c.data for c in chunks] == [[0, 1, 2, 3, 4], [10, 11, 12, 13, 14]]

    def test_overlap(self):
        ds = _FakeXr(20)
        chunks = list(sharded_xr_dataset(ds, 'time', 5, chunk_overlap=2, rank=1, world_size=2))
        assert [c.data for c in chunks] == [[5, 6, 7, 8, 9, 10, 11], [15, 16, 17, 18, 19]]

    def test_all_ranks_cover(self):
        ds = _FakeXr(30)
        seen = []
        for r in range(3):
            for c in sharded_xr_dataset(ds, 'time', 5, rank=r, world_size=3):
                seen.extend(c.data)
        assert sorted(seen) == list(range(30))

    def test_w3_exact(self):
        """100 elements / chunk 15 / 3 ranks: 6 full chunks round-robin."""
        per_rank = _rank_chunks(100, 15, 3)
        assert [len(cs) for cs in per_rank] == [2, 2, 2]
        assert per_rank[0][0] == list(range(0, 15))
        assert per_rank[1][0] == list(range(15, 30))
        assert per_rank[2][0] == list(range(30, 45))
        assert per_rank[0][1] == list(range(45, 60))
        assert per_rank[1][1] == list(range(60, 75))
        assert per_rank[2][1] == list(range(75, 90))

    def test_uneven_shards(self):
        """even_shards=False keeps the 5th chunk: rank 2 gets one chunk."""
        per_rank = _rank_chunks(100, 20, 3, even_shards=False)
        assert [len(cs) for cs in per_rank] == [2, 2, 1]
        assert per_rank[0] == [list(range(0, 20)), list(range(60, 80))]
        assert per_rank[1] == [list(range(20, 40)), list(range(80, 100))]
        assert per_rank[2] == [list(range(40, 60))]

    def test_unequal_chunks(self):
        """equal_chunks=False emits a short trailing chunk."""
        per_rank = _rank_chunks(110, 20, 3, equal_chunks=False)
        assert [len(cs) for cs in per_rank] == [2, 2, 2]
        assert per_rank[2][1] == list(range(100, 110))  # size 10
        assert per_rank[0][1] == list(range(60, 80))

    def test_shuffled(self):
        """Shuffled chunks: every chunk stays contiguous, the union covers
        the 6 full chunks, the order differs from unshuffled."""
        per_rank = _rank_chunks(100, 15, 3, shuffle=True, seed=0)
        flattened = [x for cs in per_rank for c in cs for x in c]
        assert sorted(flattened) == list(range(90))
        assert flattened != list(range(90))
        for cs in per_rank:
            for chunk in cs:
                assert chunk == list(range(chunk[0], chunk[-1] + 1))

    def test_overlap_w3_exact(self):
        """overlap=5 extends every window right by 5 elements."""
        per_rank = _rank_chunks(100, 15, 3, chunk_overlap=5)
        assert [len(cs) for cs in per_rank] == [2, 2, 2]
        assert per_rank[0][0] == list(range(0, 20))
        assert per_rank[1][0] == list(range(15, 35))
        assert per_rank[2][0] == list(range(30, 50))
        assert per_rank[0][1] == list(range(45, 65))
        assert per_rank[1][1] == list(range(60, 80))
        assert per_rank[2][1] == list(range(75, 95))

    def test_overlap_unequal_uneven(self):
        """All three flags at once: 7 chunks, rank 0 also gets the clipped
        tail window (90, 110) -> [90:100]."""
        per_rank = _rank_chunks(100, 15, 3, chunk_overlap=5, even_shards=False, equal_chunks=False)
        assert [len(cs) for cs in per_rank] == [3, 2, 2]
        assert per_rank[0][0] == list(range(0, 20))
        assert per_rank[1][0] == list(range(15, 35))
        assert per_rank[2][0] == list(range(30, 50))
        assert per_rank[0][1] == list(range(45, 65))
        assert per_rank[1][1] == list(range(60, 80))
        assert per_rank[2][1] == list(range(75, 95))
        assert per_rank[0][2] == list(range(90, 100))


class _FlattenChunks(torch.utils.data.IterableDataset):
    """Yield the individual elements of each chunk a wrapped chunked
    dataset produces (the reference's _Unzip, for exact-order asserts)."""

    def __init__(self, chunked):
        self.chunked = chunked

    def __iter__(self):
        for chunk in self.chunked:
            yield from chunk.data


def _loader_order(n, chunk_size, rank, world_size, num_workers=2):
    from dmlcloud_amd.data import ShardedXrDataset

    chunked = ShardedXrDataset(_FakeXr(n), 'time', chunk_size, rank=rank, world_size=world_size)
    loader = DataLoader(_FlattenChunks(chunked), batch_size=1, num_workers=num_workers, prefetch_factor=1)
    return [int(b) for b in loader]


class TestShardedXrWorkerInterleaving:
    """Exact element order through DataLoader workers — the behavioral
    contract of worker-id folding (reference test/test_data.py:171-363)."""

    def test_two_workers_world1(self):
        # effective world 2: worker0 owns chunks 0,2,4; worker1 owns 1,3,5.
        # batch_size=1 round-robins the workers element by element.
        out = _loader_order(100, 15, rank=0, world_size=1)
        expected = []
        for lo_a, lo_b in ((0, 15), (30, 45), (60, 75)):
            for i in range(15):
                expected += [lo_a + i, lo_b + i]
        assert out == expected

    def test_two_workers_world2_rank0(self):
        # effective world 4, 6 chunks, even_shards drops 2: rank0 gets
        # chunks 0 (worker0) and 1 (worker1), interleaved per element
        out = _loader_order(100, 15, rank=0, world_size=2)
        expected = [x for i in range(15) for x in (i, 15 + i)]
        assert out == expected

    def test_two_workers_world2_rank1(self):
        out = _loader_order(100, 15, rank=1, world_size=2)
        expected = [x for i in range(15) for x in (30 + i, 45 + i)]
        assert out == expected


class TestShardedSequenceDataset:
    def test_no_workers(self):
        ds = ShardedSequenceDataset(list(range(8)), rank=0, world_size=2)
        assert list(ds) == [0, 2, 4, 6]

    def test_worker_folding(self):
        """The exact interleaved order with num_workers=2 is the behavioral
        contract: worker rank = rank * num_workers + worker_id."""
        ds = ShardedSequenceDataset(list(range(16)), rank=0, world_size=2)
        loader = DataLoader(ds, batch_size=None, num_workers=2)
        out = [int(x) for x in loader]
        # rank0.worker0 -> effective rank 0 of 4: [0,4,8,12]
        # rank0.worker1 -> effective rank 1 of 4: [1,5,9,13]
        # DataLoader round-robins workers per item
        assert out == [0, 1, 4, 5, 8, 9, 12, 13]

    def test_worker_folding_rank1(self):
        ds = ShardedSequenceDataset(list(range(16)), rank=1, world_size=2)
        loader = DataLoader(ds, batch_size=None, num_workers=2)
        out = [int(x) for x in loader]
        assert out == [2, 3, 6, 7, 10, 11, 14, 15]

    def test_set_epoch_changes_shuffle(self):
        ds = ShardedSequenceDataset(list(range(32)), rank=0, world_size=1, shuffle=True, seed=0)
        ds.set_epoch(0)
        a = list(ds)
        ds.set_epoch(1)
        b = list(ds)
        assert a != b
        assert sorted(a) == sorted(b)


class TestPrefetchBatch:
    def test_prefetch_order(self):
        src = ShardedSequenceDataset(list(range(10)), rank=0, world_size=1)
        ds = PrefetchDataset(src, 3)
        assert list(ds) == list(range(10))

    def test_batch(self):
        src = ShardedSequenceDataset(list(range(10)), rank=0, world_size=1)
        ds = BatchDataset(src, 4)
        batches = list(ds)
        assert batches == [[0, 1, 2, 3], [4, 5, 6, 7], [8, 9]]
        assert len(ds) == 3

    def test_batch_drop_remainder(self):
        src = ShardedSequenceDataset(list(range(10)), rank=0, world_size=1)
        ds = BatchDataset(src, 4, drop_remainder=True)
        assert list(ds) == [[0, 1, 2, 3], [4, 5, 6, 7]]
        assert len(ds) == 2

    def test_prefetch_propagates_producer_error(self):
        def exploding():
            yield 1
            yield 2
            raise RuntimeError('boom in producer')

        ds = PrefetchDataset(exploding(), 2)
        it = iter(ds)
        assert next(it) == 1
        assert next(it) == 2
        with pytest.raises(RuntimeError, match='boom in producer'):
            next(it)

    def test_prefetch_stays_ahead(self):
        produced = []

        def tracking():
            for i in range(6):
                produced.append(i)
                yield i

        ds = PrefetchDataset(tracking(), 3)
        it = iter(ds)
        first = next(it)
        assert first == 0
        # the producer thread ran ahead of the consumer
        assert len(produced) >= 2


class TestInterleave:
    def test_content(self):
        batches = [torch.arange(8) + 8 * i for i in range(2)]
        out = list(interleave_batches(iter(batches), 2))
        assert len(out) == 2
        # out[i][j*s:(j+1)*s] = batches[j][i*s:(i+1)*s], s=4
        assert out[0].tolist() == [0, 1, 2, 3, 8, 9, 10, 11]
        assert out[1].tolist() == [4, 5, 6, 7, 12, 13, 14, 15]

    def test_passthrough_single(self):
        batches = [torch.arange(4)]
        out = list(interleave_batches(iter(batches), 1))
        assert out[0].tolist() == [0, 1, 2, 3]

    def test_indivisible_raises(self):
        with pytest.raises(ValueError):
            list(interleave_batches(iter([torch.arange(5)]), 2))

    def test_dict_batches(self):
        batches = [{'x': torch.arange(8) + 8 * i} for i in range(2)]
        out = list(interleave_dict_batches(iter(batches), 2))
        assert out[0]['x'].tolist() == [0, 1, 2, 3, 8, 9, 10, 11]
        assert out[1]['x'].tolist() == [4, 5, 6, 7, 12, 13, 14, 15]

    def test_2d(self):
        batches = [torch.arange(8).reshape(4, 2) + 8 * i for i in range(2)]
        out = list(interleave_batches(iter(batches), 2))
        assert out[0].tolist() == [[0, 1], [2, 3], [8, 9], [10, 11]]


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
