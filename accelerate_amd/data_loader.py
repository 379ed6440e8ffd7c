"""Data loading: per-rank sharding, prefetch, device placement
(reference: data_loader.py — behavior-parity documented per class; the
implementation is our own).

Sharding modes:
- default ("deal" mode): batch ``k`` from the inner sampler goes to rank
  ``k % n``. A one-batch lookahead ensures every rank only yields once the
  whole cycle has full batches; ``even_batches`` wraps indices around so the
  tail cycle stays rectangular (reference: data_loader.py:213-272).
- ``split_batches``: every rank iterates the same global batch and keeps its
  contiguous slice ``[bs/n*rank : bs/n*(rank+1)]`` (reference: :191-211).

Device placement: batches move H2D with ``non_blocking=True``; when the
source loader pins memory the copy overlaps the next host-side fetch on the
HIP copy engine.
"""

import math
from contextlib import suppress
from typing import Callable, List, Optional, Union

import torch
from torch.utils.data import BatchSampler, DataLoader, IterableDataset, RandomSampler

from .logging import get_logger
from .state import GradientState, PartialState
from .utils.dataclasses import DistributedType, RNGType
from .utils.operations import (
    broadcast,
    broadcast_object_list,
    concatenate,
    find_batch_size,
    get_data_structure,
    initialize_tensors,
    send_to_device,
    slice_tensors,
)
from .utils.random_utils import synchronize_rng_states

logger = get_logger(__name__)

# kwargs of torch DataLoader that we re-plumb when rebuilding a loader
_PYTORCH_DATALOADER_KWARGS = {
    "batch_size": 1,
    "shuffle": False,
    "sampler": None,
    "batch_sampler": None,
    "num_workers": 0,
    "collate_fn": None,
    "pin_memory": False,
    "drop_last": False,
    "timeout": 0,
    "worker_init_fn": None,
    "multiprocessing_context": None,
    "generator": None,
    "prefetch_factor": None,
    "persistent_workers": False,
    "pin_memory_device": "",
}


class SeedableRandomSampler(RandomSampler):
    """RandomSampler reseeded with ``seed + epoch`` each iteration so every
    process draws identical permutations (reference: data_loader.py:73-108)."""

    def __init__(self, *args, **kwargs):
        self.epoch = kwargs.pop("epoch", 0)
        self.initial_seed = kwargs.pop("seed", None)
        super().__init__(*args, **kwargs)
        if self.generator is None:
            self.generator = torch.Generator()
            if self.initial_seed is None:
                # DETERMINISTIC: the seed the process was last seeded with
                # (set_seed), identical across ranks — never a fresh entropy
                # draw (reference: data_loader.py:88 torch.random.initial_seed)
                self.initial_seed = torch.random.initial_seed() % 2**31
        elif self.initial_seed is None:
            self.initial_seed = self.generator.initial_seed()

    def __iter__(self):
        self.generator.manual_seed(self.initial_seed + self.epoch)
        yield from super().__iter__()
        self.set_epoch(self.epoch + 1)

    def set_epoch(self, epoch: int):
        self.epoch = epoch


class BatchSamplerShard(BatchSampler):
    """Shard an inner ``BatchSampler`` across ``num_processes`` ranks
    (reference: data_loader.py:110-272; semantics described in the module
    docstring)."""

    def __init__(
        self,
        batch_sampler: BatchSampler,
        num_processes: int = 1,
        process_index: int = 0,
        split_batches: bool = False,
        even_batches: bool = True,
    ):
        self.batch_sampler = batch_sampler
        self.num_processes = num_processes
        self.process_index = process_index
        self.split_batches = split_batches
        self.even_batches = even_batches
        self.batch_size = getattr(batch_sampler, "batch_size", None)
        self.drop_last = getattr(batch_sampler, "drop_last", False)
        if split_batches and self.batch_size is None:
            raise ValueError(
                "split_batches needs a fixed batch size to slice; this batch sampler has none "
                "(dynamic batch sizes work in the default deal mode)."
            )
        if split_batches and self.batch_size % num_processes != 0:
            raise ValueError(
                f"split_batches slices each global batch {num_processes} ways, so the batch size "
                f"must be divisible by the process count (got batch_size={self.batch_size}, "
                f"num_processes={num_processes})."
            )

    @property
    def total_length(self):
        return len(self.batch_sampler)

    def __len__(self):
        if self.split_batches:
            # every rank sees every (sliced) batch
            return len(self.batch_sampler)
        if len(self.batch_sampler) % self.num_processes == 0:
            return len(self.batch_sampler) // self.num_processes
        length = len(self.batch_sampler) // self.num_processes
        if self.drop_last:
            return length
        elif self.even_batches:
            return length + 1
        else:
            # ranks below the remainder get one extra batch
            return length + 1 if self.process_index < len(self.batch_sampler) % self.num_processes else length

    def __iter__(self):
        return self._iter_split() if self.split_batches else self._iter_deal()

    def _iter_split(self):
        wrap_pool = None
        per_rank = self.batch_size // self.num_processes
        lo, hi = per_rank * self.process_index, per_rank * (self.process_index + 1)
        for global_batch in self.batch_sampler:
            if wrap_pool is None:
                wrap_pool = global_batch
            if len(global_batch) == self.batch_size:
                yield global_batch[lo:hi]
            else:
                # short tail batch
                if not self.even_batches:
                    mine = global_batch[lo:hi]
                    if len(mine) > 0:
                        yield mine
                else:
                    # wrap indices from the first batch until rectangular
                    padded = list(global_batch)
                    while len(padded) < self.batch_size:
                        padded += wrap_pool[: self.batch_size - len(padded)]
                    yield padded[lo:hi]

    def _iter_deal_dynamic(self):
        """Deal mode for VARYING batch sizes (no batch_size attribute): batch
        k goes to rank k % n verbatim; with even_batches the final incomplete
        round is padded with WHOLE batches cycled from the first round
        (reference: dynamic-batch support in BatchSamplerShard, pinned by
        tests/test_data_loader.py varying-batch-size cases); with drop_last
        only complete rounds are emitted."""
        initial: List[list] = []  # first n batches — the batch-granular pad pool
        held: List[list] = []  # current round, buffered when drop_last
        idx = -1
        for idx, batch in enumerate(self.batch_sampler):
            if len(initial) < self.num_processes:
                initial.append(list(batch))
            pos = idx % self.num_processes
            if self.drop_last:
                held.append(batch)
                if pos == self.num_processes - 1:
                    yield held[self.process_index]
                    held = []
            elif pos == self.process_index:
                yield list(batch)
        total = idx + 1
        if self.drop_last or total == 0 or not self.even_batches or total % self.num_processes == 0:
            return
        while len(initial) < self.num_processes:
            initial = initial + initial
        cycle = 0
        for pos in range(total % self.num_processes, self.num_processes):
            if pos == self.process_index:
                yield list(initial[cycle])
            cycle += 1

    def _iter_deal(self):
        if self.batch_size is None:
            yield from self._iter_deal_dynamic()
            return
        wrap_pool = []  # indices from the first full cycle, used for tail padding
        pending = []  # this rank's batch, held until the cycle completes
        idx = -1
        tail_batch = []
        for idx, batch in enumerate(self.batch_sampler):
            if not self.drop_last and idx < self.num_processes:
                wrap_pool += batch
            if idx % self.num_processes == self.process_index:
                pending = batch
            # only release once the LAST batch of the cycle is full — a short
            # batch there means some rank would starve (lookahead semantics)
            if idx % self.num_processes == self.num_processes - 1 and (
                self.batch_size is None or len(batch) == self.batch_size
            ):
                yield pending
                pending = []
            tail_batch = batch

        if self.drop_last or idx == -1 or len(wrap_pool) == 0:
            return
        if not self.even_batches:
            if len(pending) > 0:
                yield pending
            return

        # even_batches tail: complete the final cycle by wrapping indices
        if self.batch_size is not None and len(pending) == self.batch_size:
            # our batch was full but the cycle was incomplete: it was never
            # released above, release it now before padding the rest
            yield pending

        # make the wrap pool long enough for degenerate tiny datasets
        while len(wrap_pool) < self.num_processes * self.batch_size:
            wrap_pool += wrap_pool

        batch = tail_batch
        if len(batch) == self.batch_size:
            batch = []
            idx += 1
        cursor = 0
        while idx % self.num_processes != 0 or len(batch) > 0:
            take = cursor + self.batch_size - len(batch)
            batch = list(batch) + wrap_pool[cursor:take]
            if idx % self.num_processes == self.process_index:
                yield batch
            cursor = take
            batch = []
            idx += 1


class IterableDatasetShard(IterableDataset):
    """Shard an ``IterableDataset``: buffer ``batch_size × n`` items, yield the
    rank's slice; pad the tail from the first buffer when ``drop_last=False``
    (reference: data_loader.py:274-371)."""

    def __init__(
        self,
        dataset: IterableDataset,
        batch_size: int = 1,
        drop_last: bool = False,
        num_processes: int = 1,
        process_index: int = 0,
        split_batches: bool = False,
    ):
        if split_batches and batch_size > 1 and batch_size % num_processes != 0:
            raise ValueError(
                f"split_batches slices each fetched batch {num_processes} ways, so the batch size "
                f"must be divisible by the process count (got batch_size={batch_size}, "
                f"num_processes={num_processes})."
            )
        self.dataset = dataset
        self.batch_size = batch_size
        self.drop_last = drop_last
        self.num_processes = num_processes
        self.process_index = process_index
        self.split_batches = split_batches

    def set_epoch(self, epoch):
        self.epoch = epoch
        if hasattr(self.dataset, "set_epoch"):
            self.dataset.set_epoch(epoch)

    def __len__(self):
        # Will raise if the underlying dataset is not sized.
        if self.drop_last:
            return (len(self.dataset) // (self.batch_size * self.num_processes)) * self.batch_size
        else:
            return math.ceil(len(self.dataset) / (self.batch_size * self.num_processes)) * self.batch_size

    def __iter__(self):
        if (
            not hasattr(self.dataset, "set_epoch")
            and hasattr(self.dataset, "generator")
            and isinstance(self.dataset.generator, torch.Generator)
        ):
            self.dataset.generator.manual_seed(getattr(self, "epoch", 0))
        fetch_window = self.batch_size if self.split_batches else (self.batch_size * self.num_processes)
        per_rank = fetch_window // self.num_processes
        mine = slice(per_rank * self.process_index, per_rank * (self.process_index + 1))

        first_window = None  # recycled to pad a short tail window
        window = []
        for item in self.dataset:
            window.append(item)
            if len(window) < fetch_window:
                continue
            yield from window[mine]
            first_window = first_window or window
            window = []

        if self.drop_last or not window:
            return
        # short tail: top up from the first window (or the tail itself on
        # tiny datasets) until rectangular, then emit this rank's slice
        pad_source = first_window or list(window)
        while len(window) < fetch_window:
            window.extend(pad_source)
        yield from window[mine]


class DataLoaderStateMixin:
    """begin()/end() register the loader with GradientState so grad-accum can
    sync on the last batch (reference: data_loader.py:373-413)."""

    def __init_subclass__(cls, **kwargs):
        cls.end_of_dataloader = False
        cls.remainder = -1

    def reset(self):
        self.end_of_dataloader = False
        self.remainder = -1

    def begin(self):
        self.reset()
        with suppress(Exception):
            if not self._drop_last:
                length = getattr(self.dataset, "total_dataset_length", len(self.dataset))
                self.remainder = length % self.total_batch_size
        self.gradient_state._add_dataloader(self)

    def end(self):
        self.gradient_state._remove_dataloader(self)


class DataLoaderAdapter:
    """Delegation-based wrapper over ``torch.utils.data.DataLoader`` so the
    wrapped loader keeps its public surface (reference: data_loader.py:416-507)."""

    def __init__(self, dataset, use_stateful_dataloader=False, batch_sampler=None, **kwargs):
        self.use_stateful_dataloader = use_stateful_dataloader
        if use_stateful_dataloader:
            from torchdata.stateful_dataloader import StatefulDataLoader

            self.base_dataloader = StatefulDataLoader(dataset, batch_sampler=batch_sampler, **kwargs)
        else:
            self.base_dataloader = DataLoader(dataset, batch_sampler=batch_sampler, **kwargs)
        if hasattr(self.base_dataloader, "state_dict"):
            self.dl_state_dict = self.base_dataloader.state_dict()

    def __getattr__(self, name):
        # delegate attribute access to the underlying loader
        if name == "base_dataloader":
            raise AttributeError()
        return getattr(self.base_dataloader, name)

    @property
    def __class__(self):
        # spoof isinstance(dl, DataLoader) checks in user code
        return self.base_dataloader.__class__

    def __reduce__(self):
        # the __class__ spoof breaks default pickling (copyreg builds the
        # spoofed class); substitute the REAL type in the reconstructor
        # (reference: data_loader.py DataLoaderAdapter.__reduce__)
        args = super().__reduce__()
        return (args[0], (type(self),) + args[1][1:]) + args[2:]

    def __len__(self):
        return len(self.base_dataloader)

    def state_dict(self):
        return self.dl_state_dict

    def load_state_dict(self, state_dict):
        self.base_dataloader.load_state_dict(state_dict)
        self.dl_state_dict = state_dict

    def _update_state_dict(self):
        # Capture the loader state one batch BEFORE yielding: when resuming we
        # must not replay the batch the caller already consumed
        # (reference: data_loader.py:471-494).
        if hasattr(self.base_dataloader, "state_dict"):
            self.dl_state_dict = self.base_dataloader.state_dict()
            # decrement the yielded counter adjustments handled by torchdata itself


class DataLoaderShard(DataLoaderAdapter, DataLoaderStateMixin):
    """Per-rank loader: RNG sync at iteration start, one-batch-ahead prefetch
    so ``end_of_dataloader`` is known *before* the last batch is yielded, and
    async H2D placement (reference: data_loader.py:510-667)."""

    def __init__(
        self,
        dataset,
        device=None,
        rng_types=None,
        synchronized_generator=None,
        skip_batches=0,
        use_stateful_dataloader=False,
        _drop_last: bool = False,
        _non_blocking: bool = False,
        **kwargs,
    ):
        super().__init__(dataset, use_stateful_dataloader=use_stateful_dataloader, **kwargs)
        self.device = device
        self.rng_types = rng_types
        self.synchronized_generator = synchronized_generator
        self.skip_batches = skip_batches
        self.gradient_state = GradientState()
        self._drop_last = _drop_last
        self._non_blocking = _non_blocking
        self.iteration = 0

    def __iter__(self):
        if self.rng_types is not None:
            rng_types = [t for t in self.rng_types if t != "generator" or self.synchronized_generator is not None]
            if rng_types:
                synchronize_rng_states(rng_types, self.synchronized_generator)
        self.begin()
        self.set_epoch(self.iteration)
        dataloader_iter = self.base_dataloader.__iter__()
        # Prefetch one batch ahead so the last batch is flagged before yield.
        try:
            current_batch = next(dataloader_iter)
        except StopIteration:
            self.end()
            return

        batch_index = 0
        while True:
            try:
                # Snapshot loader state BEFORE fetching the lookahead batch: the
                # captured state then equals the number of batches the caller has
                # consumed, so a mid-epoch save/resume does not replay batches
                # (reference: data_loader.py:597).
                self._update_state_dict()
                # fetch next BEFORE yielding current (lookahead)
                next_batch = next(dataloader_iter)
                if batch_index >= self.skip_batches:
                    yield self._move(current_batch)
                batch_index += 1
                current_batch = next_batch
            except StopIteration:
                self.end_of_dataloader = True
                self._update_state_dict()
                if batch_index >= self.skip_batches:
                    yield self._move(current_batch)
                break
        self.iteration += 1
        self.end()

    def _move(self, batch):
        if self.device is not None:
            return send_to_device(batch, self.device, non_blocking=self._non_blocking)
        return batch

    def set_epoch(self, epoch: int):
        if self.iteration != epoch:
            self.iteration = epoch
        if hasattr(self.batch_sampler, "set_epoch"):
            self.batch_sampler.set_epoch(epoch)
        elif hasattr(self.batch_sampler, "batch_sampler") and hasattr(self.batch_sampler.batch_sampler, "sampler"):
            sampler = self.batch_sampler.batch_sampler.sampler
            if hasattr(sampler, "set_epoch"):
                sampler.set_epoch(epoch)
        if hasattr(self.batch_sampler, "sampler") and hasattr(self.batch_sampler.sampler, "set_epoch"):
            self.batch_sampler.sampler.set_epoch(epoch)
        elif hasattr(self.dataset, "set_epoch"):
            self.dataset.set_epoch(epoch)

    @property
    def total_batch_size(self):
        batch_sampler = self.sampler if isinstance(self.sampler, BatchSampler) else self.batch_sampler
        if hasattr(batch_sampler, "split_batches") and batch_sampler.split_batches:
            return batch_sampler.batch_size
        if hasattr(batch_sampler, "batch_size") and batch_sampler.batch_size is not None:
            n = getattr(batch_sampler, "num_processes", 1)
            return batch_sampler.batch_size * n
        return None

    @property
    def total_dataset_length(self):
        if hasattr(self.dataset, "total_length"):
            return self.dataset.total_length
        return len(self.dataset)

    def get_sampler(self):
        return get_sampler(self)

    def set_sampler(self, sampler):
        sampler_is_batch_sampler = isinstance(self.sampler, BatchSampler)
        if sampler_is_batch_sampler:
            self.sampler.sampler = sampler
        else:
            self.batch_sampler.sampler = sampler
            if hasattr(self.batch_sampler, "batch_sampler"):
                self.batch_sampler.batch_sampler.sampler = sampler


class DataLoaderDispatcher(DataLoaderAdapter, DataLoaderStateMixin):
    """Rank 0 reads ``num_processes`` batches, broadcasts the concatenated
    global batch, every rank keeps its slice (reference: data_loader.py:723-995).
    Used for IterableDataset worlds and ``dispatch_batches=True``."""

    def __init__(
        self,
        dataset,
        split_batches: bool = False,
        skip_batches=0,
        use_stateful_dataloader=False,
        _drop_last: bool = False,
        _non_blocking: bool = False,
        slice_fn=None,
        **kwargs,
    ):
        shuffle = False
        from torch.utils.data.datapipes.iter.combinatorics import ShufflerIterDataPipe

        if isinstance(dataset, ShufflerIterDataPipe):
            shuffle = dataset._shuffle_enabled
        super().__init__(dataset, use_stateful_dataloader=use_stateful_dataloader, **kwargs)
        self.split_batches = split_batches
        if shuffle:
            torch.utils.data.graph_settings.apply_shuffle_settings(dataset, shuffle=shuffle)
        self.gradient_state = GradientState()
        self.state = PartialState()
        self._drop_last = _drop_last
        self._non_blocking = _non_blocking
        self.skip_batches = skip_batches
        self.slice_fn = slice_tensors if slice_fn is None else slice_fn
        self.iteration = 0

    def _rank0_next(self, iterator, sink):
        """Rank 0 pulls one GLOBAL batch: num_processes local batches glued
        together (unless split_batches). Local batches land in ``sink`` as
        they arrive so a mid-glue StopIteration leaves the partial tail
        recoverable by the caller."""
        if self.split_batches:
            self._update_state_dict()
            return next(iterator)
        for _ in range(self.state.num_processes):
            self._update_state_dict()
            sink.append(next(iterator))
        try:
            return concatenate(sink, dim=0)
        except RuntimeError as e:
            raise RuntimeError(
                "Variable-size batches cannot be dispatched (`dispatch_batches=True` / "
                "IterableDataset). Use `dispatch_batches=False` so each rank fetches its own "
                "batch, or `split_batches=True` so rank 0 fetches one full batch and slices it."
            ) from e

    def _fetch_batches(self, iterator):
        """One lookahead step. Agrees a [structure, exhausted] header across
        ranks via broadcast_object_list; the payload itself travels later in
        __iter__ (a tensor broadcast). The collective COUNT here is identical
        on every rank — including the tail retry — or the job deadlocks."""
        batch, partial_tail = None, []
        if self.state.process_index == 0:
            try:
                batch = self._rank0_next(iterator, partial_tail)
                header = [get_data_structure(batch), False]
            except StopIteration:
                header = [None, True]
        else:
            header = [None, self._stop_iteration]
        broadcast_object_list(header)
        self._stop_iteration = header[1]
        if self._stop_iteration and not self.split_batches and not self._drop_last:
            # rank 0 may hold a PARTIAL tail (StopIteration mid-glue): agree
            # on whether those leftover rows exist with one more header round
            if self.state.process_index == 0 and len(partial_tail) > 0:
                batch = concatenate(partial_tail, dim=0)
                header = [get_data_structure(batch), False]
            else:
                header = [None, True]
            broadcast_object_list(header)
        return batch, header

    def __iter__(self):
        self.begin()
        self.set_epoch(self.iteration)
        # every rank drives the underlying loader (worker-side effects run
        # everywhere) but only rank 0's data is dispatched
        source = self.base_dataloader.__iter__()
        exhausted = False
        self._stop_iteration = False
        spare_head = None  # first num_processes rows, kept to pad the tail
        lookahead = self._fetch_batches(source)
        position = 0
        while not exhausted:
            current, header = lookahead
            if self.state.process_index != 0:
                current = initialize_tensors(header[0])  # right shapes, recv target
            current = send_to_device(current, self.state.device, non_blocking=self._non_blocking)
            current = broadcast(current, from_process=0)

            if current is None:
                raise ValueError("dispatched batch is empty before the agreed end of data")
            if spare_head is None and not self._drop_last:
                spare_head = self.slice_fn(
                    current,
                    slice(0, self.state.num_processes),
                    process_index=self.state.process_index,
                    num_processes=self.state.num_processes,
                )

            global_rows = find_batch_size(current)
            rows_per_rank = global_rows // self.state.num_processes

            exhausted = self._stop_iteration
            if not exhausted:
                # fetch ahead: the source may be done without us knowing —
                # a recovered partial tail still needs dispatching first
                lookahead = self._fetch_batches(source)
                if self._stop_iteration and lookahead[1][0] is None:
                    exhausted = True

            if exhausted and not self._drop_last and global_rows % self.state.num_processes != 0:
                # tail not divisible: top it up with the saved head rows so
                # every rank still gets rows_per_rank+1 (dedup is
                # gather_for_metrics' job via the remainder below)
                current = concatenate([current, spare_head], dim=0)
                rows_per_rank += 1

            mine = slice(self.state.process_index * rows_per_rank, (self.state.process_index + 1) * rows_per_rank)
            current = self.slice_fn(
                current, mine, process_index=self.state.process_index, num_processes=self.state.num_processes
            )

            if exhausted:
                self.end_of_dataloader = True
                self._update_state_dict()
                # Real (pre-padding) row count of the final global batch:
                # gather_for_metrics truncates its gathered tail to this many
                # rows (reference: data_loader.py:941 observed_batch_size).
                self.remainder = global_rows
            if position >= self.skip_batches:
                yield current
            position += 1
        self.iteration += 1
        self.end()

    def set_epoch(self, epoch: int):
        self.iteration = epoch
        inner_sampler = getattr(self.batch_sampler, "sampler", None)
        if hasattr(inner_sampler, "set_epoch"):
            inner_sampler.set_epoch(epoch)
        elif hasattr(self.dataset, "set_epoch"):
            self.dataset.set_epoch(epoch)

    def __len__(self):
        n_global = len(self.base_dataloader)
        if self.split_batches:
            return n_global
        div = self.state.num_processes
        return n_global // div if self._drop_last else math.ceil(n_global / div)

    @property
    def total_batch_size(self):
        per_fetch = self.dataset.batch_size
        return per_fetch if self.split_batches else per_fetch * self.dataset.num_processes

    @property
    def total_dataset_length(self):
        return len(self.dataset)

    def get_sampler(self):
        return get_sampler(self)

    def set_sampler(self, sampler):
        sampler_is_batch_sampler = isinstance(self.sampler, BatchSampler)
        if sampler_is_batch_sampler:
            self.sampler.sampler = sampler
        else:
            self.batch_sampler.sampler = sampler
            if hasattr(self.batch_sampler, "batch_sampler"):
                self.batch_sampler.batch_sampler.sampler = sampler


def get_sampler(dataloader):
    """Fish the innermost sampler out of a (possibly wrapped) dataloader."""
    sampler_is_batch_sampler = isinstance(getattr(dataloader, "sampler", None), BatchSampler)
    if sampler_is_batch_sampler:
        return getattr(dataloader.sampler, "sampler", None)
    else:
        batch_sampler = getattr(dataloader, "batch_sampler", None)
        if batch_sampler is None:
            return getattr(dataloader, "sampler", None)
        if hasattr(batch_sampler, "batch_sampler"):
            return getattr(batch_sampler.batch_sampler, "sampler", None)
        return getattr(batch_sampler, "sampler", None)


def prepare_data_loader(
    dataloader: DataLoader,
    device: Optional[torch.device] = None,
    num_processes: Optional[int] = None,
    process_index: Optional[int] = None,
    split_batches: bool = False,
    put_on_device: bool = False,
    rng_types: Optional[List[Union[str, RNGType]]] = None,
    dispatch_batches: Optional[bool] = None,
    even_batches: bool = True,
    slice_fn_for_dispatch: Optional[Callable] = None,
    use_seedable_sampler: bool = False,
    data_seed: Optional[int] = None,
    non_blocking: bool = False,
    use_stateful_dataloader: bool = False,
) -> DataLoader:
    """Wrap a user DataLoader for the current distributed world
    (reference: data_loader.py:1016-1329)."""
    state = PartialState()
    if num_processes is None:
        num_processes = state.num_processes
    if process_index is None:
        process_index = state.process_index

    if dispatch_batches is None:
        dispatch_batches = False if not put_on_device else isinstance(dataloader.dataset, IterableDataset)
    if dispatch_batches and not put_on_device:
        raise ValueError("Using `dispatch_batches=True` requires `put_on_device=True`.")

    if split_batches and dataloader.batch_size is not None and dataloader.batch_size % num_processes != 0:
        raise ValueError(
            f"split_batches slices each global batch {num_processes} ways, so the DataLoader's "
            f"batch size must be divisible by the process count (got batch_size="
            f"{dataloader.batch_size}, num_processes={num_processes})."
        )

    new_dataset = dataloader.dataset
    new_batch_sampler = dataloader.batch_sampler if not isinstance(new_dataset, IterableDataset) else None
    sampler_is_batch_sampler = isinstance(dataloader.sampler, BatchSampler)
    synchronized_generator = None

    sampler = get_sampler(dataloader)
    if isinstance(sampler, RandomSampler) and use_seedable_sampler:
        # When iterating through the dataloader during distributed processes
        # we want to ensure that on each process we are iterating through the same
        # samples in the same order: replace the RandomSampler by a seedable one.
        sampler = SeedableRandomSampler(
            data_source=sampler.data_source,
            replacement=sampler.replacement,
            num_samples=sampler._num_samples,
            generator=getattr(sampler, "generator", None),
            seed=data_seed,
        )
    if num_processes != 1 and not dispatch_batches:
        if isinstance(new_dataset, IterableDataset):
            if getattr(dataloader.dataset, "generator", None) is not None:
                synchronized_generator = dataloader.dataset.generator
            new_dataset = IterableDatasetShard(
                new_dataset,
                batch_size=dataloader.batch_size,
                drop_last=dataloader.drop_last,
                num_processes=num_processes,
                process_index=process_index,
                split_batches=split_batches,
            )
        else:
            if not use_seedable_sampler and hasattr(sampler, "generator"):
                if sampler.generator is None:
                    sampler.generator = torch.Generator()
                    sampler.generator.manual_seed(42)
                synchronized_generator = sampler.generator
            batch_sampler = dataloader.sampler if sampler_is_batch_sampler else dataloader.batch_sampler
            new_batch_sampler = BatchSamplerShard(
                batch_sampler,
                num_processes=num_processes,
                process_index=process_index,
                split_batches=split_batches,
                even_batches=even_batches,
            )

    # We ignore all of those since they are all dealt with by our new_batch_sampler
    ignore_kwargs = ["batch_size", "shuffle", "sampler", "batch_sampler", "drop_last"]
    kwargs = {
        k: getattr(dataloader, k, _PYTORCH_DATALOADER_KWARGS[k])
        for k in _PYTORCH_DATALOADER_KWARGS
        if k not in ignore_kwargs
    }
    # Need to provide batch_size as batch_sampler is None for Iterable dataset
    if new_batch_sampler is None:
        kwargs["drop_last"] = dataloader.drop_last
        kwargs["batch_size"] = (
            dataloader.batch_size // num_processes if split_batches and not dispatch_batches else dataloader.batch_size
        )
    if kwargs.get("prefetch_factor") is not None and kwargs.get("num_workers", 0) == 0:
        kwargs["prefetch_factor"] = None
    if isinstance(sampler, SeedableRandomSampler) and use_seedable_sampler:
        if sampler_is_batch_sampler:
            dataloader.sampler.sampler = sampler
        else:
            dataloader.batch_sampler.sampler = sampler
            if hasattr(dataloader.batch_sampler, "batch_sampler"):
                dataloader.batch_sampler.batch_sampler.sampler = sampler

    if dispatch_batches:
        kwargs.pop("generator", None)
        dataloader = DataLoaderDispatcher(
            new_dataset,
            split_batches=split_batches,
            batch_sampler=new_batch_sampler,
            _drop_last=dataloader.drop_last,
            _non_blocking=non_blocking,
            slice_fn=slice_fn_for_dispatch,
            use_stateful_dataloader=use_stateful_dataloader,
            **kwargs,
        )
    elif sampler_is_batch_sampler:
        dataloader = DataLoaderShard(
            new_dataset,
            device=device if put_on_device else None,
            sampler=new_batch_sampler,
            batch_size=dataloader.batch_size,
            rng_types=rng_types,
            _drop_last=dataloader.drop_last,
            _non_blocking=non_blocking,
            synchronized_generator=synchronized_generator,
            use_stateful_dataloader=use_stateful_dataloader,
            **kwargs,
        )
    else:
        dataloader = DataLoaderShard(
            new_dataset,
            device=device if put_on_device else None,
            batch_sampler=new_batch_sampler,
            rng_types=rng_types,
            synchronized_generator=synchronized_generator,
            _drop_last=dataloader.drop_last,
            _non_blocking=non_blocking,
            use_stateful_dataloader=use_stateful_dataloader,
            **kwargs,
        )

    return dataloader


class SkipBatchSampler(BatchSampler):
    """Yield batches of an inner batch sampler, skipping the first
    ``skip_batches`` (reference: data_loader.py:1332+)."""

    def __init__(self, batch_sampler, skip_batches=0):
        self.batch_sampler = batch_sampler
        self.sampler = getattr(batch_sampler, "sampler", None)
        self.skip_batches = skip_batches

    def __iter__(self):
        for index, samples in enumerate(self.batch_sampler):
            if index >= self.skip_batches:
                yield samples

    @property
    def total_length(self):
        return len(self.batch_sampler)

    def __len__(self):
        return len(self.batch_sampler) - self.skip_batches


class SkipDataLoader(DataLoaderAdapter, DataLoaderStateMixin):
    """DataLoader yielding everything after the first ``skip_batches``."""

    def __init__(self, dataset, skip_batches=0, use_stateful_dataloader=False, **kwargs):
        super().__init__(dataset, use_stateful_dataloader=use_stateful_dataloader, **kwargs)
        self.skip_batches = skip_batches
        self.gradient_state = GradientState()

    def __iter__(self):
        self.begin()
        for index, batch in enumerate(self.base_dataloader.__iter__()):
            if index >= self.skip_batches:
                self._update_state_dict()
                yield batch
        self.end()


def skip_first_batches(dataloader, num_batches=0):
    """Mid-epoch resume: a new loader that skips ``num_batches``
    (reference: data_loader.py:1395)."""
    state = PartialState()
    if state.distributed_type == DistributedType.FSDP:
        pass  # plain torch loaders below handle it

    dataset = dataloader.dataset
    sampler_is_batch_sampler = False
    if isinstance(dataset, IterableDataset):
        new_batch_sampler = None
    else:
        sampler_is_batch_sampler = isinstance(dataloader.sampler, BatchSampler)
        batch_sampler = dataloader.sampler if sampler_is_batch_sampler else dataloader.batch_sampler
        new_batch_sampler = SkipBatchSampler(batch_sampler, skip_batches=num_batches)

    # We ignore all of those since they are all dealt with by our new_batch_sampler
    ignore_kwargs = ["batch_size", "shuffle", "sampler", "batch_sampler", "drop_last"]
    kwargs = {
        k: getattr(dataloader, k, _PYTORCH_DATALOADER_KWARGS[k])
        for k in _PYTORCH_DATALOADER_KWARGS
        if k not in ignore_kwargs
    }
    if new_batch_sampler is None:
        kwargs["drop_last"] = dataloader.drop_last
        kwargs["batch_size"] = dataloader.batch_size
    if kwargs.get("prefetch_factor") is not None and kwargs.get("num_workers", 0) == 0:
        kwargs["prefetch_factor"] = None

    if isinstance(dataloader, DataLoaderDispatcher):
        if new_batch_sampler is None:
            # Need to manually skip batches in the dataloader
            kwargs["skip_batches"] = num_batches
        prior_iteration = dataloader.iteration
        dataloader = DataLoaderDispatcher(
            dataset,
            split_batches=dataloader.split_batches,
            batch_sampler=new_batch_sampler,
            _drop_last=dataloader._drop_last,
            **kwargs,
        )
        # mid-epoch resume must NOT rewind the epoch: iteration carries over
        # so __iter__'s set_epoch call replays the same sampler epoch
        dataloader.iteration = prior_iteration
    elif isinstance(dataloader, DataLoaderShard):
        if new_batch_sampler is None:
            # Need to manually skip batches in the dataloader
            kwargs["skip_batches"] = num_batches
        elif sampler_is_batch_sampler:
            kwargs["sampler"] = new_batch_sampler
            kwargs["batch_size"] = dataloader.batch_size
        else:
            kwargs["batch_sampler"] = new_batch_sampler
        prior_iteration = dataloader.iteration
        dataloader = DataLoaderShard(
            dataset,
            device=dataloader.device,
            rng_types=dataloader.rng_types,
            synchronized_generator=dataloader.synchronized_generator,
            **kwargs,
        )
        dataloader.iteration = prior_iteration
    else:
        if new_batch_sampler is None:
            # Need to manually skip batches in the dataloader
            dataloader = SkipDataLoader(dataset, skip_batches=num_batches, **kwargs)
        else:
            dataloader = DataLoader(dataset, batch_sampler=new_batch_sampler, **kwargs)

    return dataloader
