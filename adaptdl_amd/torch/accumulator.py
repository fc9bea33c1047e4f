"""Restart-safe distributed metric accumulator.

Dict-like object with two modes: *accumulation* (default; ``+=``/``-=``
updates are captured locally) and *synchronized* (inside
:meth:`Accumulator.synchronized`; pending updates are summed across replicas
and results readable).  Results of synchronizations are recorded in a
per-epoch history so that replayed program sections after a restart observe
identical values.  API and semantics follow the reference
(``/root/reference/adaptdl/adaptdl/torch/accumulator.py``).
"""

import collections
import contextlib
import copy
import pickle

import adaptdl_amd.checkpoint
import adaptdl_amd.collective
from adaptdl_amd.torch.epoch import current_epoch


class Accumulator(collections.abc.MutableMapping):
    """See module docstring.  Typical use::

        accum = Accumulator()
        for epoch in remaining_epochs_until(60):
            for batch in loader:
                accum["loss_sum"] += ...
                accum["total"] += ...
            with accum.synchronized():
                print(accum["loss_sum"] / accum["total"])
                accum.clear()
    """

    def __init__(self, *args, **kwargs):
        self._sync_count = collections.Counter()
        self._synchronized = None
        self._state = _AccumulatorState(*args, **kwargs)
        adaptdl_amd.checkpoint.load_state(self._state)

    @contextlib.contextmanager
    def synchronized(self):
        """Enter synchronized mode (a distributed synchronization point)."""
        if self._synchronized is not None:
            yield self
            return
        epoch = current_epoch()
        # Results of epochs that finished are never replayed; drop them.
        for key in list(self._state.results_history.keys()):
            if key is not None and epoch is not None and key < epoch:
                self._state.results_history.pop(key)
        count = self._sync_count[epoch]
        self._sync_count[epoch] += 1
        results_list = self._state.results_history[epoch]
        assert count <= len(results_list)
        if count < len(results_list):
            # Replaying: reuse recorded results, discard pending updates.
            self._synchronized = results_list[count]
            self._state.updates.clear()
        else:
            self._state.sync()
            from adaptdl_amd.torch.data import current_dataloader
            if current_dataloader() is None:
                # Inside dataloader iterations code is not replayed, so
                # only record history outside of them.
                results_list.append(copy.deepcopy(self._state.results))
            self._synchronized = self._state.results
        try:
            yield self
        finally:
            # Back to accumulation mode.
            self._synchronized = None

    def update(self, *args, **kwargs):
        """Additively apply a collection of key-update pairs."""
        for key, val in dict(*args, **kwargs).items():
            self[key] += val

    def subtract(self, *args, **kwargs):
        """Subtract a collection of key-update pairs."""
        for key, val in dict(*args, **kwargs).items():
            self[key] -= val

    def __iadd__(self, other):
        self.update(other)
        return self

    def __isub__(self, other):
        self.subtract(other)
        return self

    def __getitem__(self, key):
        if self._synchronized is not None:
            return self._synchronized.__getitem__(key)
        # Accumulation mode: return a proxy capturing += / -= updates.
        return _Value(self, key)

    def __setitem__(self, key, value):
        if self._synchronized is not None:
            self._synchronized[key] = value
            return
        if not isinstance(value, _Value):
            raise TypeError("invalid value type: {}".format(type(value)))
        if value.accum is not self:
            raise ValueError("incompatible accumulator")
        if key != value.key:
            raise ValueError("incompatible key: {}".format(value.key))
        self._state.updates.setdefault(key, 0)
        self._state.updates[key] += value.update

    def __contains__(self, key):
        if self._synchronized is not None:
            return key in self._synchronized
        return False

    def __delitem__(self, key):
        if self._synchronized is not None:
            del self._synchronized[key]

    def __iter__(self):
        if self._synchronized is not None:
            return iter(self._synchronized)
        return iter(())

    def __len__(self):
        if self._synchronized is not None:
            return len(self._synchronized)
        return 0

    def __repr__(self):
        if self._synchronized is not None:
            return repr(self._synchronized)
        return "{}"


class _Value(object):
    """Accumulation-mode proxy: ``accum[k] += v`` reads this out of
    __getitem__, applies +/-, and hands it back to __setitem__, which
    records the delta."""

    __slots__ = ["accum", "key", "update"]

    def __init__(self, accum, key):
        self.accum = accum
        self.key = key
        self.update = 0

    def _apply(self, delta):
        if isinstance(delta, _Value):
            raise TypeError(
                "invalid update type: {}".format(type(delta)))
        self.update += delta
        return self

    def __add__(self, delta):
        return self._apply(delta)

    def __sub__(self, delta):
        return self._apply(-delta)


def _dict_iadd(a, b):
    for k, v in b.items():
        if k in a:
            a[k] += v
        else:
            a[k] = v
    return a


class _AccumulatorState(adaptdl_amd.checkpoint.State):

    # Accumulators must be initialized in the same order on every replica;
    # name them by (epoch, creation index within epoch).
    init_count = collections.Counter()

    def __init__(self, *args, **kwargs):
        from adaptdl_amd.torch.data import current_dataloader
        if current_dataloader() is not None:
            raise RuntimeError("accumulator may not be initialized during "
                               "dataloader iteration")
        epoch = current_epoch()
        count = _AccumulatorState.init_count[epoch]
        super().__init__("adaptdl-accumulator-epoch{}-{}".format(epoch, count))
        _AccumulatorState.init_count[epoch] += 1
        self.results_history = collections.defaultdict(list)
        self.results = dict(*args, **kwargs)
        self.updates = {}

    def save(self, fileobj):
        pickle.dump((self.results_history, self.results), fileobj)

    def load(self, fileobj):
        self.results_history, self.results = pickle.load(fileobj)

    def sync(self):
        updates = adaptdl_amd.collective.allreduce(self.updates, _dict_iadd)
        _dict_iadd(self.results, updates)
        self.updates.clear()
