"""SIGTERM/SIGINT handling for graceful checkpoint-and-exit.

Semantics follow the reference (``/root/reference/adaptdl/adaptdl/_signal.py``):
the first signal sets a flag polled once per iteration by the data loader
(which then checkpoints and exits with code 143); a second SIGINT force-exits.
"""

import logging
import signal
import sys
import threading

LOG = logging.getLogger(__name__)

_EXIT_FLAG = False
_INSTALLED = False


def _handler(signum, frame):
    global _EXIT_FLAG
    if _EXIT_FLAG and signum == signal.SIGINT:
        LOG.warning("second SIGINT received, exiting immediately")
        sys.exit(1)
    LOG.info("signal %s received, will checkpoint and exit", signum)
    _EXIT_FLAG = True


def install_signal_handlers():
    global _INSTALLED
    if _INSTALLED or threading.current_thread() is not threading.main_thread():
        return
    signal.signal(signal.SIGTERM, _handler)
    signal.signal(signal.SIGINT, _handler)
    _INSTALLED = True


def get_exit_flag():
    """True once a SIGTERM/SIGINT (or test-injected exit) was received."""
    return _EXIT_FLAG


def set_exit_flag(value=True):
    """Programmatic trigger used by the in-process allocator on rescale."""
    global _EXIT_FLAG
    _EXIT_FLAG = value
