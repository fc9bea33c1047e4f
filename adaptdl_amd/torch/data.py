"""Elastic data loading with adaptive batch sizes.

Provides :class:`ElasticSampler` (deterministic strided partition of a
shuffled dataset across replicas, resumable from any sample index),
:class:`AdaptiveDataLoaderHelper` (the elastic core: batch-size
autoscaling via the goodput model, per-iteration exit-flag sync,
restart-safe loop bookkeeping), and :class:`AdaptiveDataLoader` (a drop-in
torch DataLoader).  Behavior mirrors the reference
(``/root/reference/adaptdl/adaptdl/torch/data.py``): the ``batch_size``
argument is the *total* batch size across replicas; with autoscaling on,
an epoch iterates until statistical progress equals one non-adaptive epoch.
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
from adaptdl_amd._signal import get_exit_flag

LOG = logging.getLogger(__name__)


class ElasticSampler(Sampler):
    """Partitions (optionally shuffled) sample indices across replicas.

    Shuffling is a deterministic function of (epoch, pass index), so all
    replicas agree on the order without communication, and sampling can be
    resumed from an arbitrary global sample index after a rescale changes
    the replica count (reference: data.py:41-111).
    """

    def __init__(self, dataset, shuffle=True):
        self.dataset = dataset
        self.shuffle = shuffle
        self.num_replicas = adaptdl_amd.env.num_replicas()
        self.rank = adaptdl_amd.env.replica_rank()
        self.epoch = 0
        self.index = 0  # global sample index to resume from

    def __iter__(self):
        if self.shuffle:
            g = torch.Generator()
            # Deterministic across processes (unlike builtin hash of str).
            g.manual_seed(self.epoch * 0x9E3779B1
                          + self.index // len(self.dataset))
            indices = torch.randperm(len(self.dataset), generator=g).tolist()
        else:
            indices = list(range(len(self.dataset)))
        base_index = self.index % len(self.dataset)
        local_indices = indices[base_index + self.rank::self.num_replicas]
        # Pad so every replica yields the same number of samples.  The
        # modulo covers num_replicas > len(dataset) (a rank with no
        # samples of its own still yields one padding sample).
        if len(local_indices) < len(self):
            local_indices.append(indices[self.rank % len(indices)])
        assert len(local_indices) == len(self)
        return iter(local_indices)

    def __len__(self):
        base_index = self.index % len(self.dataset)
        return math.ceil((len(self.dataset) - base_index) / self.num_replicas)

    def set_epoch(self, epoch, index=0):
        self.epoch = epoch
        self.index = index


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

    def __init__(self, batch_size=1):
        self._max_batch_size = None
        self._local_bsz_bounds = None
        self._state = _AdaptiveDataLoaderState()
        adaptdl_amd.checkpoint.load_state(self._state)
        self.batch_size = batch_size
        self.future_exit = None
        self._gradient_accumulation = False
        self._speedup_threshold = 1.05
        self._accum_count = 0

    @property
    def current_index(self):
        """Global number of samples processed so far in the current loop."""
        if AdaptiveDataLoaderHelper._current is not self:
            return None
        return self._state.current_index

    @current_index.setter
    def current_index(self, index):
        if AdaptiveDataLoaderHelper._current is not self:
            return
        self._state.current_index = index

    @property
    def end_index(self):
        return self._state.end_index

    @end_index.setter
    def end_index(self, index):
        self._state.end_index = index

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

    def is_accum_step(self):
        return self._accum_count < self._state.accumulation_steps

    def is_optim_step(self):
        return not self.is_accum_step()

    def train(self):
        """Mark this loader as the training loader (only one allowed)."""
        if AdaptiveDataLoaderHelper._training is None:
            AdaptiveDataLoaderHelper._training = self
        set_batch_size(self.batch_size, self.max_batch_size,
                       self.local_bsz_bounds, self._gradient_accumulation)

    def autoscale_batch_size(self, max_batch_size, local_bsz_bounds=None,
                             gradient_accumulation=False):
        """Enable goodput-driven adaptive batch sizes up to max_batch_size."""
        if not isinstance(max_batch_size, int) or \
                max_batch_size < self.batch_size:
            raise ValueError("invalid max_batch_size")
        if local_bsz_bounds is not None and (
                local_bsz_bounds[0] is not None and
                local_bsz_bounds[0] > self.batch_size or
                local_bsz_bounds[1] is not None and
                local_bsz_bounds[1] < self.batch_size):
            raise ValueError("invalid local_bsz_bounds")
        self._max_batch_size = max_batch_size
        self._local_bsz_bounds = local_bsz_bounds
        self._gradient_accumulation = gradient_accumulation
        self.train()

    def _sync_local_bsz(self):
        """Choose (atomic_bsz, accum_steps) for this pass and broadcast it.

        Keeps the current choice unless the goodput model predicts at least
        a 5% speedup from changing (hysteresis; reference data.py:270-305).
        """
        goodput_fn = get_goodput_fn()
        if self.max_batch_size is None or goodput_fn is None:
            self._state.current_local_bsz = math.ceil(
                self.batch_size / adaptdl_amd.env.num_replicas())
            self._state.accumulation_steps = 0
        elif not self._state.current_local_bsz:
            _, atomic_bsz, accum_steps = goodput_fn.optimize(
                adaptdl_amd.env.num_nodes(), adaptdl_amd.env.num_replicas(),
                max_batch_size=self._max_batch_size,
                atomic_bsz_range=self._local_bsz_bounds,
                accumulation=self._gradient_accumulation)
            self._state.current_local_bsz = atomic_bsz
            self._state.accumulation_steps = accum_steps
        else:
            suggest_goodput, atomic_bsz, accum_steps = goodput_fn.optimize(
                adaptdl_amd.env.num_nodes(), adaptdl_amd.env.num_replicas(),
                max_batch_size=self._max_batch_size,
                atomic_bsz_range=self._local_bsz_bounds,
                accumulation=self._gradient_accumulation)
            current_goodput = goodput_fn(
                adaptdl_amd.env.num_nodes(), adaptdl_amd.env.num_replicas(),
                self.current_local_bsz, self.accumulation_steps)
            speedup = suggest_goodput / max(current_goodput, 1e-8)
            if speedup > self._speedup_threshold:
                self._state.current_local_bsz = atomic_bsz
                self._state.accumulation_steps = accum_steps
        self._state.current_local_bsz, self._state.accumulation_steps = \
            adaptdl_amd.collective.broadcast(
                (self._state.current_local_bsz,
                 self._state.accumulation_steps))
        return self.current_local_bsz

    @property
    def training(self):
        return self is AdaptiveDataLoaderHelper._training

    @contextmanager
    def profile(self, commit):
        """Wraps each iteration: exit-flag sync + step profiling.

        Must be entered the same number of times on every replica.  If a
        rescale/preemption signal was agreed on by all replicas, saves a
        checkpoint and exits with code 143.
        """
        if self.future_exit is not None and self.future_exit.result():
            adaptdl_amd.checkpoint.save_all_states()
            exit(143)
        self.future_exit = adaptdl_amd.collective.allreduce_async(
            get_exit_flag(), lambda a, b: a or b)
        profile_step_start(self.current_local_bsz)
        yield
        if commit:
            profile_step_commit(self.is_accum_step())
        self._accum_count = (0 if self.is_optim_step()
                             else self._accum_count + 1)

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
            self._state.current_index = 0
            self._state.end_index = 0
            self._state.last_position[epoch] = self._position[epoch]
            self._position[epoch] += 1
            AdaptiveDataLoaderHelper._current = None

    @property
    def current_batch_size(self):
        return (self.current_local_bsz * (self.accumulation_steps + 1) *
                adaptdl_amd.env.num_replicas())

    def skipdone(self):
        """True if this loop already completed before a restart (skip it)."""
        epoch = current_epoch()
        position = self._position[epoch]
        if position <= self._state.last_position.get(epoch, -1):
            LOG.info("skipping dataloader loop at position %s in epoch %s",
                     position, epoch)
            self._position[epoch] += 1
            return True
        return False

    def to_tensorboard(self, writer, global_step, tag_prefix=""):
        if tag_prefix and not tag_prefix.endswith("/"):
            tag_prefix += "/"
        writer.add_scalar(tag_prefix + "Total_Batch_Size",
                          self.current_batch_size, global_step)
        writer.add_scalar(tag_prefix + "Local_Batch_Size",
                          self.current_local_bsz, global_step)
        writer.add_scalar(tag_prefix + "Accumulation_Steps",
                          self.accumulation_steps, global_step)


class AdaptiveDataLoaderMixin(object):
    """Adds elasticity to custom DataLoader classes via ``self._elastic``."""

    def __init__(self, batch_size):
        self._elastic = AdaptiveDataLoaderHelper(batch_size)

    def autoscale_batch_size(self, max_batch_size, local_bsz_bounds=None,
                             gradient_accumulation=False):
        self._elastic.autoscale_batch_size(max_batch_size, local_bsz_bounds,
                                           gradient_accumulation)

    @property
    def current_local_bsz(self):
        if AdaptiveDataLoaderHelper._current is not self._elastic:
            return None
        return self._elastic.current_local_bsz

    @property
    def accumulation_steps(self):
        return self._elastic.accumulation_steps

    @property
    def training(self):
        return self._elastic.training

    @property
    def current_batch_size(self):
        if AdaptiveDataLoaderHelper._current is not self._elastic:
            return None
        return self._elastic.current_batch_size

    def to_tensorboard(self, writer, global_step, tag_prefix=""):
        self._elastic.to_tensorboard(writer, global_step, tag_prefix)


def _worker_init_wrapper(worker_init_fn, num_workers):
    """Globally-unique python/numpy/torch seeds for each loader worker."""

    @functools.wraps(worker_init_fn)
    def wrapper(worker_id):
        nonlocal num_workers
        num_workers = num_workers or 1
        seed = torch.initial_seed() \
            + adaptdl_amd.env.replica_rank() * num_workers
        torch.manual_seed(seed)
        np.random.seed(seed % 2 ** 32)
        random.seed(seed)
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
        if kwargs.get("batch_sampler") is not None \
                or kwargs.get("sampler") is not None:
            raise ValueError("AdaptiveDataLoader does not support "
                             "custom 'sampler' or 'batch_sampler'")
        kwargs["sampler"] = ElasticSampler(dataset, shuffle=shuffle)
        kwargs["worker_init_fn"] = _worker_init_wrapper(
            kwargs.get("worker_init_fn"), kwargs.get("num_workers"))
        super().__init__(dataset, batch_size, shuffle=False, **kwargs)
        AdaptiveDataLoaderMixin.__init__(self, batch_size)

    def __iter__(self):
        """Iterate over batches; stops after one (statistical) epoch.

        Without autoscaling: one pass over the dataset split across
        replicas.  With autoscaling: iterates (possibly >1 data passes)
        until scale-invariant progress covers one epoch-equivalent.
        """
        epoch = current_epoch()
        num_replicas = adaptdl_amd.env.num_replicas()
        with self._elastic.context():
            if self._elastic.skipdone():
                return
            done = False
            while not done:
                self.sampler.set_epoch(epoch,
                                       index=self._elastic.current_index)
                self.batch_sampler.batch_size = \
                    self._elastic._sync_local_bsz()
                for idx, batch in enumerate(super().__iter__()):
                    # Skip profiling the first batch of each pass (loader
                    # worker startup would pollute the perf model).
                    with self._elastic.profile(self.training and idx >= 1):
                        yield batch
                        self._elastic.current_index += \
                            num_replicas * self.batch_sampler.batch_size
                        if self._elastic.max_batch_size is not None and \
                                get_progress() >= len(self.dataset) * \
                                (epoch + 1) / self.batch_size:
                            done = True
                            break
                if self._elastic.max_batch_size is None:
                    done = True
                # Round current_index up to a multiple of the dataset size.
                self._elastic.current_index -= \
                    self._elastic.current_index % -len(self.dataset)


class _AdaptiveDataLoaderState(adaptdl_amd.checkpoint.State):

    # Dataloaders must be initialized in the same order on every replica.
    init_count = collections.Counter()

    def __init__(self):
        if current_dataloader() is not None:
            raise RuntimeError("dataloader may not be initialized during "
                               "dataloader iteration")
        epoch = current_epoch()
        count = _AdaptiveDataLoaderState.init_count[epoch]
        super().__init__("adaptdl-dataloader-epoch{}-{}".format(epoch, count))
        _AdaptiveDataLoaderState.init_count[epoch] += 1
        self.current_index = 0
        self.end_index = 0
        self.last_position = {}
        self.current_local_bsz = 0
        self.accumulation_steps = 0

    def save(self, fileobj):
        pickle.dump((self.current_index, self.end_index,
                     self.last_position), fileobj)

    def load(self, fileobj):
        self.current_index, self.end_index, self.last_position = \
            pickle.load(fileobj)
