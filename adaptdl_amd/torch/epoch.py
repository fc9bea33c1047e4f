"""Restart-safe epoch loops.

Training programs written against adaptdl_amd must be *idempotent up to
epoch granularity*: after a checkpoint-restart the program is re-executed
from the top, and :func:`remaining_epochs_until` skips epochs that already
finished, while dataloader/accumulator state replays position within the
current epoch.  Semantics match the reference
(``/root/reference/adaptdl/adaptdl/torch/epoch.py:96-178``).
"""

import logging
import pickle

import adaptdl_amd.checkpoint

LOG = logging.getLogger(__name__)


def remaining_epochs_until(epoch):
    """Iterate over the epochs in ``[finished_epochs(), epoch)``.

    After a checkpoint-restart, previously finished epochs are skipped.

    Raises:
        RuntimeError: If invoked before a previous epoch loop has ended.
    """
    state = _epoch_state()
    if state.current_epoch is not None:
        raise RuntimeError("overlapping epoch loops detected")
    LOG.info("starting at epoch %s" if state.finished_epochs < epoch
             else "skipping all epochs up to %s",
             state.finished_epochs if state.finished_epochs < epoch
             else epoch)
    while state.finished_epochs < epoch:
        state.current_epoch = state.finished_epochs
        try:
            yield state.current_epoch
        finally:
            # Catches breaks and exceptions escaping the epoch body too.
            state.finished_epochs += 1
            state.current_epoch = None


def current_epoch():
    """The current epoch, or None outside a remaining_epochs_until loop."""
    return _epoch_state().current_epoch


def finished_epochs():
    """Number of epochs finished across all restarts."""
    return _epoch_state().finished_epochs


class _EpochState(adaptdl_amd.checkpoint.State):
    """Checkpointed epoch counter; current_epoch is deliberately NOT
    persisted (a restart resumes AT the interrupted epoch via the
    dataloader's position replay, not inside it)."""

    def __init__(self):
        super().__init__(".adaptdl-epoch")
        self.finished_epochs = 0
        self.current_epoch = None

    def save(self, fileobj):
        pickle.dump(self.finished_epochs, fileobj)

    def load(self, fileobj):
        self.finished_epochs = pickle.load(fileobj)


_EPOCH_STATE = None


def _epoch_state():
    global _EPOCH_STATE
    if _EPOCH_STATE is None:
        _EPOCH_STATE = _EpochState()
        adaptdl_amd.checkpoint.load_state(_EPOCH_STATE)
    return _EPOCH_STATE
