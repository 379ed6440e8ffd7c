"""BatchSamplerShard / IterableDatasetShard semantics (oracle: the reference's
exhaustive small-N enumeration strategy, tests/test_data_loader.py)."""

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st
import torch
from torch.utils.data import BatchSampler, DataLoader, IterableDataset, SequentialSampler, TensorDataset

from accelerate_amd.data_loader import (
    BatchSamplerShard,
    DataLoaderShard,
    IterableDatasetShard,
    SeedableRandomSampler,
    SkipBatchSampler,
    SkipDataLoader,
    prepare_data_loader,
    skip_first_batches,
)


def make_batch_sampler(n_items, batch_size, drop_last=False):
    return BatchSampler(SequentialSampler(range(n_items)), batch_size=batch_size, drop_last=drop_last)


def shards(n_items, batch_size, num_processes, drop_last=False, split_batches=False, even_batches=True):
    bs = make_batch_sampler(n_items, batch_size, drop_last)
    return [
        list(
            BatchSamplerShard(
                bs, num_processes=num_processes, process_index=i, split_batches=split_batches, even_batches=even_batches
            )
        )
        for i in range(num_processes)
    ]


class TestBatchSamplerShardDeal:
    def test_even_division(self):
        out = shards(24, 3, 2)
        assert out[0] == [[0, 1, 2], [6, 7, 8], [12, 13, 14], [18, 19, 20]]
        assert out[1] == [[3, 4, 5], [9, 10, 11], [15, 16, 17], [21, 22, 23]]

    def test_tail_wraparound_even_batches(self):
        # 22 items, bs 3, n 2: last batch [21] is short -> rank1 pads from the start
        out = shards(22, 3, 2)
        assert out[0] == [[0, 1, 2], [6, 7, 8], [12, 13, 14], [18, 19, 20]]
        assert out[1] == [[3, 4, 5], [9, 10, 11], [15, 16, 17], [21, 0, 1]]

    def test_tail_not_even_batches(self):
        out = shards(22, 3, 2, even_batches=False)
        assert out[0] == [[0, 1, 2], [6, 7, 8], [12, 13, 14], [18, 19, 20]]
        assert out[1] == [[3, 4, 5], [9, 10, 11], [15, 16, 17], [21]]

    def test_drop_last(self):
        # 22 items, bs 3, drop_last: 7 batches -> last incomplete cycle dropped
        out = shards(22, 3, 2, drop_last=True)
        assert out[0] == [[0, 1, 2], [6, 7, 8], [12, 13, 14]]
        assert out[1] == [[3, 4, 5], [9, 10, 11], [15, 16, 17]]

    def test_degenerate_tiny_dataset(self):
        # fewer samples than one global batch: wrap repeatedly
        out = shards(2, 3, 2)
        assert out[0] == [[0, 1, 0]]
        assert out[1] == [[1, 0, 1]]

    def test_incomplete_cycle_full_batches(self):
        # 9 items bs 3 n 2: 3 batches; rank0's batch idx2 completes an odd cycle
        out = shards(9, 3, 2)
        # idx0->r0, idx1->r1 (cycle yields), idx2->r0 full but cycle incomplete
        assert out[0][0] == [0, 1, 2]
        assert out[1][0] == [3, 4, 5]
        # tail: rank0 yields [6,7,8]; rank1 wraps for a rectangular cycle
        assert out[0][1] == [6, 7, 8]
        assert out[1][1] == [0, 1, 2]
        assert len(out[0]) == len(out[1]) == 2

    def test_lengths_match_iteration(self):
        for n_items in (2, 9, 22, 23, 24, 30):
            for bs in (1, 3, 4):
                for n in (1, 2, 4):
                    for even in (True, False):
                        for drop in (True, False):
                            if drop and n_items < bs * n:
                                continue
                            sh = [
                                BatchSamplerShard(
                                    make_batch_sampler(n_items, bs, drop),
                                    num_processes=n,
                                    process_index=i,
                                    even_batches=even,
                                )
                                for i in range(n)
                            ]
                            for s in sh:
                                produced = list(s)
                                if even or drop:
                                    assert len(produced) == len(s), (n_items, bs, n, even, drop, s.process_index)

    def test_even_batches_rectangular(self):
        # with even_batches, every rank yields the same number of full batches
        for n_items in (5, 11, 17, 23):
            out = shards(n_items, 4, 2)
            assert len(out[0]) == len(out[1])
            for r in out:
                for b in r:
                    assert len(b) == 4


class TestBatchSamplerShardSplit:
    def test_split_even(self):
        out = shards(16, 4, 2, split_batches=True)
        assert out[0] == [[0, 1], [4, 5], [8, 9], [12, 13]]
        assert out[1] == [[2, 3], [6, 7], [10, 11], [14, 15]]

    def test_split_tail_pad(self):
        out = shards(14, 4, 2, split_batches=True)
        # last global batch [12,13] -> padded to [12,13,0,1]
        assert out[0][-1] == [12, 13]
        assert out[1][-1] == [0, 1]

    def test_split_requires_divisible(self):
        with pytest.raises(ValueError):
            BatchSamplerShard(make_batch_sampler(16, 3, False), num_processes=2, process_index=0, split_batches=True)


class RandomIterable(IterableDataset):
    def __init__(self, n):
        self.n = n

    def __iter__(self):
        yield from range(self.n)


class TestIterableDatasetShard:
    def test_basic(self):
        ds = IterableDatasetShard(RandomIterable(16), batch_size=2, num_processes=2, process_index=0, drop_last=False)
        assert list(ds) == [0, 1, 4, 5, 8, 9, 12, 13]
        ds1 = IterableDatasetShard(RandomIterable(16), batch_size=2, num_processes=2, process_index=1, drop_last=False)
        assert list(ds1) == [2, 3, 6, 7, 10, 11, 14, 15]

    def test_tail_pad(self):
        # 10 items, real batch 4: buffers [0-3],[4-7], tail [8,9] padded from first buffer
        ds0 = IterableDatasetShard(RandomIterable(10), batch_size=2, num_processes=2, process_index=0, drop_last=False)
        ds1 = IterableDatasetShard(RandomIterable(10), batch_size=2, num_processes=2, process_index=1, drop_last=False)
        assert list(ds0) == [0, 1, 4, 5, 8, 9]
        assert list(ds1) == [2, 3, 6, 7, 0, 1]

    def test_drop_last(self):
        ds0 = IterableDatasetShard(RandomIterable(10), batch_size=2, num_processes=2, process_index=0, drop_last=True)
        assert list(ds0) == [0, 1, 4, 5]


class TestDataLoaderShard:
    def test_end_of_dataloader_flag(self):
        ds = TensorDataset(torch.arange(8).float())
        dl = DataLoaderShard(ds, batch_size=2)
        batches = []
        for i, b in enumerate(dl):
            batches.append(b)
            expected = i == 3
            assert dl.end_of_dataloader == expected
        assert len(batches) == 4

    def test_rng_sync_same_permutation(self):
        # single process: seedable sampler reproducibility across epochs with set_epoch
        ds = TensorDataset(torch.arange(10).float())
        sampler = SeedableRandomSampler(data_source=ds, generator=torch.Generator().manual_seed(0))
        order1 = list(sampler)
        sampler.set_epoch(0)
        order2 = list(sampler)
        assert order1 == order2

    def test_prepare_single_process_passthrough(self):
        from accelerate_amd.state import PartialState

        PartialState()
        ds = TensorDataset(torch.arange(10).float())
        dl = DataLoader(ds, batch_size=2)
        prepared = prepare_data_loader(dl, put_on_device=False)
        seen = torch.cat([b[0] for b in prepared])
        assert torch.equal(seen, torch.arange(10).float())


class TestSkip:
    def test_skip_batch_sampler(self):
        bs = make_batch_sampler(16, 4)
        skipped = SkipBatchSampler(bs, skip_batches=2)
        assert list(skipped) == [[8, 9, 10, 11], [12, 13, 14, 15]]
        assert len(skipped) == 2

    def test_skip_data_loader(self):
        ds = TensorDataset(torch.arange(16).float())
        dl = SkipDataLoader(ds, batch_size=4, skip_batches=2)
        seen = torch.cat([b[0] for b in dl])
        assert torch.equal(seen, torch.arange(8, 16).float())

    def test_skip_first_batches(self):
        from accelerate_amd.state import PartialState

        PartialState()
        ds = TensorDataset(torch.arange(16).float())
        dl = DataLoader(ds, batch_size=4)
        new_dl = skip_first_batches(dl, num_batches=2)
        seen = torch.cat([b[0] for b in new_dl])
        assert torch.equal(seen, torch.arange(8, 16).float())


class TestBatchSamplerShardProperties:
    """Property-based invariant sweep (hypothesis): across the whole
    parameter grid, per-rank shards must partition the sample stream with
    the documented padding/drop semantics (reference tests enumerate these
    by hand; the sweep covers the space)."""

    @given(
        n_samples=st.integers(1, 64),
        batch_size=st.integers(1, 8),
        n_ranks=st.integers(1, 4),
        split_batches=st.booleans(),
        even_batches=st.booleans(),
        drop_last=st.booleans(),
    )
    @settings(max_examples=200, deadline=None, derandomize=True)
    def test_partition_invariants(self, n_samples, batch_size, n_ranks, split_batches, even_batches, drop_last):
        from torch.utils.data import BatchSampler, SequentialSampler

        from accelerate_amd.data_loader import BatchSamplerShard

        if split_batches and batch_size % n_ranks != 0:
            return  # documented constraint (raises)
        base = BatchSampler(SequentialSampler(range(n_samples)), batch_size, drop_last)
        shards = [
            list(BatchSamplerShard(base, num_processes=n_ranks, process_index=r,
                                   split_batches=split_batches, even_batches=even_batches))
            for r in range(n_ranks)
        ]
        # 1. with even_batches every rank yields the same number of batches
        #    (lockstep collectives); even_batches=False may be ragged — that
        #    is exactly the join_uneven_inputs use case
        if even_batches:
            lens = {len(s) for s in shards}
            assert len(lens) == 1, f"ragged shard lengths {lens}"
        # 2. batches on one rank all have equal size when even_batches
        if even_batches and not drop_last:
            sizes = {len(b) for s in shards for b in s}
            assert len(sizes) <= 1, sizes
        # 3. union of yielded indices ⊆ dataset, and ⊇ dataset when nothing
        #    is dropped (drop_last may drop a tail; padding duplicates allowed)
        seen = {i for s in shards for b in s for i in b}
        assert seen <= set(range(n_samples))
        if not drop_last and sum(len(s) for s in shards) > 0:
            assert seen == set(range(n_samples)), "lost samples without drop_last"
        # 4. with drop_last, no duplicates at all
        if drop_last:
            flat = [i for s in shards for b in s for i in b]
            assert len(flat) == len(set(flat))


class CountingDataset(TensorDataset):
    """TensorDataset counting __getitem__ calls — proxy for items fetched."""

    def __init__(self, n):
        super().__init__(torch.arange(n).float())
        self.fetches = 0

    def __getitem__(self, idx):
        self.fetches += 1
        return super().__getitem__(idx)


class TestAdviceRegressions:
    """Regression tests for the round-1 advisor findings."""

    def test_user_generator_not_clobbered(self):
        # prepare_data_loader must not replace a user-supplied shuffle
        # generator with a fixed seed-42 one (non-XLA framework).
        ds = TensorDataset(torch.arange(20).float())
        gen = torch.Generator().manual_seed(7)
        dl = DataLoader(ds, batch_size=1, shuffle=True, generator=gen)
        prepared = prepare_data_loader(dl)
        order = [int(b[0].item()) for b in prepared]

        gen2 = torch.Generator().manual_seed(7)
        expected = [int(b[0].item()) for b in DataLoader(ds, batch_size=1, shuffle=True, generator=gen2)]
        assert order == expected, "user generator was clobbered (shuffle order != seed-7 order)"

    def test_shard_state_snapshot_before_yield(self):
        # With a stateful base loader, the snapshot returned by state_dict()
        # must reflect batches the CALLER consumed, not the lookahead fetch
        # and not only end-of-epoch (ADVICE item on _update_state_dict timing).
        ds = CountingDataset(5)
        dl = prepare_data_loader(DataLoader(ds, batch_size=1))
        assert isinstance(dl, DataLoaderShard)
        dl.base_dataloader.state_dict = lambda: {"fetched": ds.fetches}

        it = iter(dl)
        next(it)  # consumed 1 (lookahead has fetched 2)
        assert dl.state_dict()["fetched"] == 1
        next(it)  # consumed 2
        assert dl.state_dict()["fetched"] == 2


class SimpleIterable(IterableDataset):
    def __init__(self, n=100):
        self.n = n

    def __iter__(self):
        for _ in range(self.n):
            yield torch.rand(1)

    def __len__(self):
        return self.n

    def set_epoch(self, epoch):
        self.epoch = epoch


class EpochedBatchSampler(BatchSampler):
    """Batch sampler with its own epoch-seeded generator (reference fixture
    SimpleBatchSampler semantics)."""

    def __init__(self, sampler, batch_size, drop_last, generator, seed):
        super().__init__(sampler, batch_size, drop_last)
        self.generator = generator
        self.seed = seed
        self.epoch = 0

    def __iter__(self):
        self.generator.manual_seed(self.seed + self.epoch)
        return super().__iter__()

    def set_epoch(self, epoch):
        self.epoch = epoch


class TestDynamicBatchSize:
    """Varying-batch-size sharding (reference: varying-batch-size cases in
    tests/test_data_loader.py — whole-batch deal + batch-granular padding)."""

    BS5 = [[0, 1, 2], [3, 4], [5, 6, 7, 8], [9, 10, 11], [12, 13]]

    def shards(self, bs, n, **kw):
        return [list(BatchSamplerShard(bs, n, i, **kw)) for i in range(n)]

    def test_not_even(self):
        s = self.shards(self.BS5, 2, even_batches=False)
        assert s[0] == [[0, 1, 2], [5, 6, 7, 8], [12, 13]]
        assert s[1] == [[3, 4], [9, 10, 11]]
        lens = [len(BatchSamplerShard(self.BS5, 2, i, even_batches=False)) for i in range(2)]
        assert lens == [3, 2]

    def test_even_no_padding_needed(self):
        s = self.shards(self.BS5[:4], 2, even_batches=True)
        assert s[0] == [[0, 1, 2], [5, 6, 7, 8]]
        assert s[1] == [[3, 4], [9, 10, 11]]

    def test_even_pads_whole_batches(self):
        s = self.shards(self.BS5, 2, even_batches=True)
        assert s[0] == [[0, 1, 2], [5, 6, 7, 8], [12, 13]]
        assert s[1] == [[3, 4], [9, 10, 11], [0, 1, 2]]

    def test_single_batch_degenerate(self):
        s = self.shards([[0, 1, 2]], 2, even_batches=True)
        assert s == [[[0, 1, 2]], [[0, 1, 2]]]
        s = self.shards([[0, 1, 2]], 3, even_batches=True)
        assert s == [[[0, 1, 2]], [[0, 1, 2]], [[0, 1, 2]]]

    def test_drop_last_drops_incomplete_round(self):
        class DropLastList:
            drop_last = True

            def __init__(self, data):
                self.data = data

            def __iter__(self):
                return iter(self.data)

            def __len__(self):
                return len(self.data)

        s = self.shards(DropLastList(self.BS5), 2, even_batches=True)
        assert s[0] == [[0, 1, 2], [5, 6, 7, 8]]
        assert s[1] == [[3, 4], [9, 10, 11]]

    def test_three_processes(self):
        bs = [[0], [1, 2], [3, 4, 5], [6], [7, 8], [9, 10, 11], [12, 13]]
        s = self.shards(bs, 3, even_batches=True)
        assert s[0] == [[0], [6], [12, 13]]
        assert s[1] == [[1, 2], [7, 8], [0]]
        assert s[2] == [[3, 4, 5], [9, 10, 11], [1, 2]]

    def test_split_batches_rejects_dynamic(self):
        with pytest.raises(ValueError, match="split_batches"):
            BatchSamplerShard([[0, 1], [2, 3]], 2, 0, split_batches=True)

    def test_split_batches_rejects_non_divisible(self):
        base = BatchSampler(SequentialSampler(range(20)), batch_size=3, drop_last=False)
        with pytest.raises(ValueError, match="divisible"):
            BatchSamplerShard(base, 2, 0, split_batches=True)


class TestReferenceEdgeCases:
    def test_iterable_none_batch_size(self):
        dl = prepare_data_loader(DataLoader(SimpleIterable(20), batch_size=None))
        for d in dl:
            assert isinstance(d, torch.Tensor)

    def test_iterable_non_tensor_samples(self):
        def collate(features):
            return {"tensor": torch.stack(features), "non_tensor": "constant"}

        dl = prepare_data_loader(DataLoader(SimpleIterable(10), batch_size=4, collate_fn=collate))
        for d in dl:
            assert isinstance(d["tensor"], torch.Tensor)
            assert d["non_tensor"] == "constant"

    def test_end_of_dataloader_two_epochs(self):
        from accelerate_amd.data_loader import DataLoaderShard as DLS

        dl = DLS(list(range(16)), batch_size=4)
        for _ in range(2):  # flag must reset per epoch
            for idx, _ in enumerate(dl):
                assert dl.end_of_dataloader == (idx == 3)

    def test_set_epoch_reaches_custom_batch_sampler(self):
        ds = list(range(16))
        bs = EpochedBatchSampler(SequentialSampler(ds), 4, False, torch.Generator(), seed=12)
        dl = prepare_data_loader(DataLoader(ds, batch_sampler=bs))
        assert bs.epoch == 0
        dl.set_epoch(1)
        assert bs.epoch == 1

    def test_skip_first_batches_preserves_iteration(self):
        from accelerate_amd.data_loader import DataLoaderShard as DLS

        ds = list(range(16))
        bs = EpochedBatchSampler(SequentialSampler(ds), 4, False, torch.Generator(), seed=42)
        dl = DLS(ds, batch_sampler=bs)
        dl.set_epoch(1)
        assert dl.iteration == 1
        resumed = skip_first_batches(dl, num_batches=2)
        assert resumed.iteration == 1

    def test_skip_first_batches_does_not_reset_sampler_epoch(self):
        from accelerate_amd.data_loader import DataLoaderShard as DLS

        ds = list(range(16))
        sampler = SeedableRandomSampler(data_source=ds, seed=3)
        bs = EpochedBatchSampler(sampler, 4, False, torch.Generator(), seed=42)
        dl = DLS(ds, batch_sampler=bs)
        dl.set_epoch(1)
        resumed = skip_first_batches(dl, num_batches=2)
        next(iter(resumed))
        assert sampler.epoch == 1

    def test_dataloader_cleanup_no_leak(self):
        import gc
        import weakref

        from accelerate_amd.data_loader import DataLoaderShard as DLS

        dl = DLS(list(range(16)), batch_size=4)
        it = iter(dl)
        assert next(it).tolist() == [0, 1, 2, 3]
        ref = weakref.ref(dl)
        del dl, it
        gc.collect()
        assert ref() is None, "DataLoaderShard leaked after deletion mid-iteration"

    def test_reproducibility_seedable_sampler(self):
        from accelerate_amd import set_seed

        orders = []
        for proc in range(2):
            set_seed(21)  # every rank enters prepare with the same RNG state
            ds = TensorDataset(torch.arange(32).float())
            dl = prepare_data_loader(
                DataLoader(ds, batch_size=4, shuffle=True),
                num_processes=2,
                process_index=proc,
                use_seedable_sampler=True,
            )
            epoch_orders = []
            for _ in range(2):
                epoch_orders.append([int(x) for (b,) in dl for x in b])
            orders.append(epoch_orders)
        # the two ranks' shards must partition the same global permutation:
        # no overlap within an epoch, and epoch orders differ across epochs
        for e in range(2):
            assert not (set(orders[0][e]) & set(orders[1][e]))
        assert orders[0][0] != orders[0][1]
