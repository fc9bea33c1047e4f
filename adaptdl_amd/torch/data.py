"""Elastic data loading with adaptive batch sizes.

Provides :class:`ElasticSampler` (deterministic strided partition of a
shuffled dataset across replicas, resumable from any sample index),
:class:`AdaptiveDataLoaderHelper` (the elastic core: batch-size
autoscaling via the goodput model, per-iteration exit-flag sync,
restart-safe loop bookkeeping), and :class:`AdaptiveDataLoader` (a drop-in
torch DataLoader).  The *semantics* — batch accounting, restart
replay-skip, the 1.05 hysteresis on batch-size changes, the exit-143
protocol — are the behavioral contract shared with the reference
(``/root/reference/adaptdl/adaptdl/torch/data.py``) and are encoded in
this repo's restart/property tests; the implementation is adaptdl_amd's
own.  The ``batch_size`` argument is the *total* batch size across
replicas; with autoscaling on, an epoch iterates until statistical
progress equals one non-adaptive epoch.
"""

from contextlib import contextmanager
import collections
import functools
import logging
import math
import pickle
import random

import numpy as np
import torch
from torch.utils.data import DataLoader, Sampler

import adaptdl_amd.checkpoint
import adaptdl_amd.collective
import adaptdl_amd.env
from adaptdl_amd.torch.epoch import current_epoch
from adaptdl_amd.torch._metrics import (
    profile_step_start, profile_step_commit,
    set_batch_size, get_goodput_fn, get_progress)
from adaptdl_amd._signal import get_exit_flag, get_rescale_request
from adaptdl_amd.torch import _rejoin

LOG = logging.getLogger(__name__)


def _control_reduce(a, b):
    """Elementwise reduce of the per-iteration control payload:
    (exit requested anywhere, rescale version seen by EVERYONE)."""
    return (a[0] or b[0], min(a[1], b[1]))


class ElasticSampler(Sampler):
    """Partitions (optionally shuffled) sample indices across replicas.

    Shuffling is a deterministic function of (epoch, pass index), so all
    replicas agree on the order without communication, and sampling can be
    resumed from an arbitrary global sample index after a rescale changes
    the replica count.
    """

    def __init__(self, dataset, shuffle=True):
        self.dataset = dataset
        self.shuffle = shuffle
        self.num_replicas = adaptdl_amd.env.num_replicas()
        self.rank = adaptdl_amd.env.replica_rank()
        self.epoch = 0
        self.index = 0  # global sample index to resume from

    def set_epoch(self, epoch, index=0):
        self.epoch = epoch
        self.index = index

    def _ordering(self):
        """Full index permutation for the current (epoch, data pass)."""
        n = len(self.dataset)
        if not self.shuffle:
            return list(range(n))
        g = torch.Generator()
        # Deterministic across processes (unlike builtin hash of str).
        g.manual_seed(self.epoch * 0x9E3779B1 + self.index // n)
        return torch.randperm(n, generator=g).tolist()

    def __len__(self):
        remaining = len(self.dataset) - self.index % len(self.dataset)
        return -(-remaining // self.num_replicas)  # ceil division

    def __iter__(self):
        order = self._ordering()
        first = self.index % len(self.dataset) + self.rank
        mine = order[first::self.num_replicas]
        # Pad so every replica yields the same number of samples.  The
        # modulo covers num_replicas > len(dataset) (a rank with no
        # samples of its own still yields one padding sample).
        want = len(self)
        while len(mine) < want:
            mine.append(order[self.rank % len(order)])
        assert len(mine) == want
        return iter(mine)


def current_dataloader():
    """The AdaptiveDataLoaderHelper currently being iterated, or None."""
    return AdaptiveDataLoaderHelper._current


class AdaptiveDataLoaderHelper(object):
    """Fine-grained control core for adaptive training loops."""

    # epoch -> number of dataloader loops completed in that epoch (all
    # dataloaders); used to identify loops for restart replay-skip.
    _position = collections.Counter()
    _training = None
    _current = None

    #: Set by loaders whose __iter__ can re-partition mid-pass (the
    #: in-place rescale raise is only safe there).
    _supports_inplace = False

    def __init__(self, batch_size=1):
        self.batch_size = batch_size
        self.future_exit = None
        # Autoscale config (off until autoscale_batch_size() is called).
        self._max_batch_size = None
        self._local_bsz_bounds = None
        self._gradient_accumulation = False
        self._speedup_threshold = 1.05
        self._accum_count = 0
        # Checkpointed loop position / batch-size choice.
        self._state = _AdaptiveDataLoaderState()
        adaptdl_amd.checkpoint.load_state(self._state)

    # ---- elastic loop position (checkpointed) ---------------------------

    @property
    def current_index(self):
        """Global number of samples processed so far in the current loop
        (None unless this loader is the one being iterated)."""
        if self._iterating:
            return self._state.current_index
        return None

    @current_index.setter
    def current_index(self, index):
        if self._iterating:
            self._state.current_index = index

    @property
    def end_index(self):
        return self._state.end_index

    @end_index.setter
    def end_index(self, index):
        self._state.end_index = index

    @property
    def _iterating(self):
        return AdaptiveDataLoaderHelper._current is self

    # ---- batch-size configuration ---------------------------------------

    @property
    def max_batch_size(self):
        return self._max_batch_size

    @property
    def local_bsz_bounds(self):
        return self._local_bsz_bounds

    @property
    def current_local_bsz(self):
        """Atomic (per-replica, per-accumulation-step) batch size."""
        return self._state.current_local_bsz

    @property
    def accumulation_steps(self):
        return self._state.accumulation_steps

    @property
    def current_batch_size(self):
        replicas = adaptdl_amd.env.num_replicas()
        return self.current_local_bsz * (self.accumulation_steps + 1) \
            * replicas

    def is_accum_step(self):
        return self._accum_count < self._state.accumulation_steps

    def is_optim_step(self):
        return not self.is_accum_step()

    @property
    def training(self):
        return self is AdaptiveDataLoaderHelper._training

    def train(self):
        """Mark this loader as the training loader (only one allowed)."""
        if AdaptiveDataLoaderHelper._training is None:
            AdaptiveDataLoaderHelper._training = self
        set_batch_size(self.batch_size, self.max_batch_size,
                       self.local_bsz_bounds, self._gradient_accumulation)

    def autoscale_batch_size(self, max_batch_size, local_bsz_bounds=None,
                             gradient_accumulation=False):
        """Enable goodput-driven adaptive batch sizes up to max_batch_size."""
        if not isinstance(max_batch_size, int) \
                or max_batch_size < self.batch_size:
            raise ValueError("invalid max_batch_size")
        if local_bsz_bounds is not None:
            lo, hi = local_bsz_bounds
            if (lo is not None and lo > self.batch_size) \
                    or (hi is not None and hi < self.batch_size):
                raise ValueError("invalid local_bsz_bounds")
        self._max_batch_size = max_batch_size
        self._local_bsz_bounds = local_bsz_bounds
        self._gradient_accumulation = gradient_accumulation
        self.train()

    def _sync_local_bsz(self):
        """Choose (atomic_bsz, accum_steps) for this pass and broadcast it.

        Keeps the current choice unless the goodput model predicts at least
        a ``_speedup_threshold`` speedup from changing (hysteresis).
        """
        state = self._state
        goodput_fn = get_goodput_fn()
        if self.max_batch_size is None or goodput_fn is None:
            # Autoscaling off (or model not fitted yet): split the target
            # batch size evenly and round up.
            replicas = adaptdl_amd.env.num_replicas()
            choice = (-(-self.batch_size // replicas), 0)
        else:
            topo = (adaptdl_amd.env.num_nodes(),
                    adaptdl_amd.env.num_replicas())
            proposed_goodput, atomic_bsz, accum_steps = goodput_fn.optimize(
                *topo, max_batch_size=self._max_batch_size,
                atomic_bsz_range=self._local_bsz_bounds,
                accumulation=self._gradient_accumulation)
            if not state.current_local_bsz:
                # First choice of the run: adopt the proposal outright.
                choice = (atomic_bsz, accum_steps)
            else:
                # Hysteresis: only adopt if predicted speedup over the
                # current setting is significant.
                held_goodput = goodput_fn(*topo, self.current_local_bsz,
                                          self.accumulation_steps)
                ratio = proposed_goodput / max(held_goodput, 1e-8)
                if ratio > self._speedup_threshold:
                    choice = (atomic_bsz, accum_steps)
                else:
                    choice = (state.current_local_bsz,
                              state.accumulation_steps)
        # All replicas adopt rank 0's choice.
        state.current_local_bsz, state.accumulation_steps = \
            adaptdl_amd.collective.broadcast(choice)
        return self.current_local_bsz

    # ---- per-iteration / per-loop contexts -------------------------------

    @contextmanager
    def profile(self, commit):
        """Wraps each iteration: exit-flag sync + step profiling.

        Must be entered the same number of times on every replica.  If a
        rescale/preemption signal was agreed on by all replicas, saves a
        checkpoint and exits with code 143.
        """
        if self.future_exit is not None:
            do_exit, rescale_ver = self.future_exit.result()
            self.future_exit = None
            if do_exit:
                adaptdl_amd.checkpoint.save_all_states()
                exit(143)
            if rescale_ver > _rejoin.applied_version() \
                    and self._supports_inplace and self.training \
                    and self._accum_count == 0:
                # Every replica agreed on the directive and is at an
                # optimizer-cycle boundary: hand control to __iter__,
                # which performs the in-place rejoin and re-partitions
                # the current pass (leavers never return from it).
                _, directive = get_rescale_request()
                if directive is not None:
                    raise _rejoin.InplaceRescale(directive)
        self.future_exit = adaptdl_amd.collective.allreduce_async(
            (get_exit_flag(), get_rescale_request()[0]), _control_reduce)
        profile_step_start(self.current_local_bsz)
        yield
        if commit:
            profile_step_commit(self.is_accum_step())
        if self.is_optim_step():
            self._accum_count = 0
        else:
            self._accum_count += 1

    @contextmanager
    def context(self):
        """All iterator loops run under this; tracks loop position."""
        epoch = current_epoch()
        try:
            if AdaptiveDataLoaderHelper._current is not None:
                raise RuntimeError(
                    "overlapping dataloader iterations detected")
            AdaptiveDataLoaderHelper._current = self
            yield
        finally:
            # Loop finished (or aborted): reset resume indices and record
            # this loop position as completed for restart replay-skip.
            self._state.current_index = 0
            self._state.end_index = 0
            self._state.last_position[epoch] = self._position[epoch]
            self._position[epoch] += 1
            AdaptiveDataLoaderHelper._current = None

    def skipdone(self):
        """True if this loop already completed before a restart (skip it)."""
        epoch = current_epoch()
        position = self._position[epoch]
        if position > self._state.last_position.get(epoch, -1):
            return False
        LOG.info("skipping dataloader loop at position %s in epoch %s",
                 position, epoch)
        self._position[epoch] += 1
        return True

    def to_tensorboard(self, writer, global_step, tag_prefix=""):
        if tag_prefix and not tag_prefix.endswith("/"):
            tag_prefix += "/"
        for tag, value in (("Total_Batch_Size", self.current_batch_size),
                           ("Local_Batch_Size", self.current_local_bsz),
                           ("Accumulation_Steps", self.accumulation_steps)):
            writer.add_scalar(tag_prefix + tag, value, global_step)


def _delegated(attr, only_while_iterating=False):
    """Property on the mixin that forwards to ``self._elastic``."""
    def getter(self):
        helper = self._elastic
        if only_while_iterating and not helper._iterating:
            return None
        return getattr(helper, attr)
    return property(getter)


class AdaptiveDataLoaderMixin(object):
    """Adds elasticity to custom DataLoader classes via ``self._elastic``."""

    def __init__(self, batch_size):
        self._elastic = AdaptiveDataLoaderHelper(batch_size)

    def autoscale_batch_size(self, *args, **kwargs):
        self._elastic.autoscale_batch_size(*args, **kwargs)

    def to_tensorboard(self, *args, **kwargs):
        self._elastic.to_tensorboard(*args, **kwargs)

    current_local_bsz = _delegated("current_local_bsz",
                                   only_while_iterating=True)
    current_batch_size = _delegated("current_batch_size",
                                    only_while_iterating=True)
    accumulation_steps = _delegated("accumulation_steps")
    training = _delegated("training")


def _worker_init_wrapper(worker_init_fn, num_workers):
    """Globally-unique python/numpy/torch seeds for each loader worker."""

    @functools.wraps(worker_init_fn)
    def wrapper(worker_id):
        seed = torch.initial_seed() \
            + adaptdl_amd.env.replica_rank() * max(num_workers or 1, 1)
        for seeder in (torch.manual_seed, random.seed,
                       lambda s: np.random.seed(s % 2 ** 32)):
            seeder(seed)
        if worker_init_fn is not None:
            return worker_init_fn(worker_id)
    return wrapper


class AdaptiveDataLoader(DataLoader, AdaptiveDataLoaderMixin):
    """Drop-in torch DataLoader with adaptive batch size + elasticity.

    Differences from a stock DataLoader: ``batch_size`` is the total across
    replicas; custom samplers are not supported; iteration must happen
    inside a :func:`remaining_epochs_until` loop.
    """

    def __init__(self, dataset, batch_size=1, shuffle=False, **kwargs):
        for forbidden in ("sampler", "batch_sampler"):
            if kwargs.get(forbidden) is not None:
                raise ValueError("AdaptiveDataLoader does not support "
                                 "custom 'sampler' or 'batch_sampler'")
        kwargs["sampler"] = ElasticSampler(dataset, shuffle=shuffle)
        kwargs["worker_init_fn"] = _worker_init_wrapper(
            kwargs.get("worker_init_fn"), kwargs.get("num_workers"))
        super().__init__(dataset, batch_size, shuffle=False, **kwargs)
        AdaptiveDataLoaderMixin.__init__(self, batch_size)

    def _epoch_target_reached(self, epoch):
        """Scale-invariant progress has covered ``epoch + 1`` epochs."""
        target = len(self.dataset) * (epoch + 1) / self.batch_size
        return get_progress() >= target

    def __iter__(self):
        """Iterate over batches; stops after one (statistical) epoch.

        Without autoscaling: one pass over the dataset split across
        replicas.  With autoscaling: iterates (possibly >1 data passes)
        until scale-invariant progress covers one epoch-equivalent.
        """
        epoch = current_epoch()
        adaptive = self._elastic.max_batch_size is not None
        self._elastic._supports_inplace = True
        with self._elastic.context():
            if self._elastic.skipdone():
                return
            while True:
                # Re-read per pass: an in-place rescale changes the
                # replica count without restarting this generator.
                replicas = adaptdl_amd.env.num_replicas()
                self.sampler.num_replicas = replicas
                self.sampler.rank = adaptdl_amd.env.replica_rank()
                self.sampler.set_epoch(
                    epoch, index=self._elastic.current_index)
                atomic_bsz = self._elastic._sync_local_bsz()
                self.batch_sampler.batch_size = atomic_bsz
                stop = False
                try:
                    inner = enumerate(super().__iter__())
                    for idx, batch in inner:
                        # Skip profiling the first batch of each pass
                        # (loader worker startup would pollute the perf
                        # model).
                        with self._elastic.profile(self.training
                                                   and idx >= 1):
                            yield batch
                            self._elastic.current_index += \
                                replicas * self.batch_sampler.batch_size
                            if adaptive and \
                                    self._epoch_target_reached(epoch):
                                stop = True
                                break
                except _rejoin.InplaceRescale as req:
                    # Survivors return with the world resized; leavers
                    # exit(143) inside.  The current pass then restarts
                    # from the exact global sample index with the new
                    # partitioning (same math as a restart resume).
                    _rejoin.perform(req.directive)
                    continue
                # Round current_index up to a multiple of the dataset size
                # (ends the data pass even if it stopped mid-way).
                self._elastic.current_index += \
                    -self._elastic.current_index % len(self.dataset)
                if stop or not adaptive:
                    return


class _AdaptiveDataLoaderState(adaptdl_amd.checkpoint.State):

    # Dataloaders must be initialized in the same order on every replica.
    init_count = collections.Counter()

    # Fields that survive a checkpoint-restart (the batch-size choice is
    # deliberately NOT persisted: it is re-chosen after a rescale).
    _SAVED = ("current_index", "end_index", "last_position")

    def __init__(self):
        if current_dataloader() is not None:
            raise RuntimeError("dataloader may not be initialized during "
                               "dataloader iteration")
        epoch = current_epoch()
        ordinal = self.init_count[epoch]
        self.init_count[epoch] += 1
        super().__init__(f"adaptdl-dataloader-epoch{epoch}-{ordinal}")
        self.current_index = 0
        self.end_index = 0
        self.last_position = {}
        self.current_local_bsz = 0
        self.accumulation_steps = 0

    def save(self, fileobj):
        pickle.dump({f: getattr(self, f) for f in self._SAVED}, fileobj)

    def load(self, fileobj):
        for f, v in pickle.load(fileobj).items():
            setattr(self, f, v)
