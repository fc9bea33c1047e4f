"""Named-state checkpoint registry (checkpoint-restart elasticity).

Keeps the reference's on-disk checkpoint format (directory of named state
files inside ``checkpoint-<num_restarts>`` directories, written via an
atomically-renamed ``_checkpoint`` staging dir; see
``/root/reference/adaptdl/adaptdl/checkpoint.py:97-206``) so checkpoints are
interchangeable with reference jobs.  The implementation is new: states are
kept in a single registry dict, save/load are rank-0-writes +
all-rank-sync, and the latest checkpoint dir is chosen by the highest K.
"""

import os
import logging
import shutil

from adaptdl_amd import env

LOG = logging.getLogger(__name__)

CKPT_DIR_PREFIX = "checkpoint-"

_REGISTRY = {}  # name -> State


class State(object):
    """A named piece of state saved/loaded as part of a checkpoint.

    Subclass and override :meth:`save`, :meth:`load`, and optionally
    :meth:`sync` (invoked on all replicas before rank 0 writes).
    """

    def __init__(self, name):
        if name in _REGISTRY:
            raise ValueError("State '{}' already exists".format(name))
        _REGISTRY[name] = self
        self._adaptdl_name = name

    @property
    def name(self):
        return self._adaptdl_name

    def save(self, fileobj):
        pass

    def load(self, fileobj):
        pass

    def sync(self):
        pass


def _staging_dir(root):
    d = os.path.join(root, "_checkpoint")
    os.makedirs(d, exist_ok=True)
    return d


def save_all_states(cold=False):
    """Save every registered State; returns the checkpoint root on rank 0.

    The rescale hot path writes to the RAM-backed warm root when the
    controller provisioned one (``ADAPTDL_WARM_CHECKPOINT_PATH``,
    normally tmpfs): model/optimizer/GNS state then round-trips through
    host memory, never disk, on an elastic rescale.  ``cold=True`` (or
    no warm root) writes to the on-disk ``ADAPTDL_CHECKPOINT_PATH`` in
    the reference-compatible checkpoint-<K> format for crash recovery.
    """
    root = None if cold else env.warm_checkpoint_path()
    if root is None:
        root = env.checkpoint_path()
    for state in list(_REGISTRY.values()):
        save_state(state, root)
    if env.replica_rank() == 0 and root is not None:
        final = os.path.join(root, CKPT_DIR_PREFIX + str(env.num_restarts()))
        if os.path.isdir(final):
            shutil.rmtree(final)
        os.rename(_staging_dir(root), final)  # atomic publish
        for name in os.listdir(root):
            path = os.path.join(root, name)
            if name.startswith(CKPT_DIR_PREFIX) and path != final:
                shutil.rmtree(path, ignore_errors=True)
        return root


def save_state(state, root, sync=True):
    """Sync a State across replicas, then write it on rank 0."""
    if sync:
        state.sync()
    if env.replica_rank() == 0 and root is not None:
        path = os.path.join(_staging_dir(root), state.name)
        with open(path, "wb") as f:
            state.save(f)


def _latest_ckpt_dir(root):
    """(dir, K) of the highest checkpoint-K under root, or (None, -1)."""
    best = None
    best_k = -1
    if root is None or not os.path.isdir(root):
        return best, best_k
    for name in os.listdir(root):
        if name.startswith(CKPT_DIR_PREFIX):
            try:
                k = int(name[len(CKPT_DIR_PREFIX):])
            except ValueError:
                continue
            if k > best_k:
                best_k, best = k, os.path.join(root, name)
    return best, best_k


def load_state(state):
    """Load a State from the most recent checkpoint dir, if one exists.

    Both roots are considered — the RAM-backed warm root (rescale hot
    path) and the on-disk cold root — and the highest restart number
    wins, the warm root breaking ties (it is at least as recent).

    Returns True iff the state file was found and ``state.load`` invoked.
    """
    warm_dir, warm_k = _latest_ckpt_dir(env.warm_checkpoint_path())
    cold_dir, cold_k = _latest_ckpt_dir(env.checkpoint_path())
    ckpt_dir = warm_dir if warm_k >= cold_k and warm_dir is not None \
        else cold_dir
    if ckpt_dir is None:
        return False
    path = os.path.join(ckpt_dir, state.name)
    if not os.path.isfile(path):
        LOG.warning("no state file %s in %s", state.name, ckpt_dir)
        return False
    with open(path, "rb") as f:
        state.load(f)
    return True
