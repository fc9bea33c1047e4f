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
_RESCALE_SEEN = False  # SIGUSR2: an in-place rescale directive exists


def _handler(signum, frame):
    global _EXIT_FLAG
    if _EXIT_FLAG and signum == signal.SIGINT:
        LOG.warning("second SIGINT received, exiting immediately")
        sys.exit(1)
    LOG.info("signal %s received, will checkpoint and exit", signum)
    _EXIT_FLAG = True


def _usr2_handler(signum, frame):
    global _RESCALE_SEEN
    _RESCALE_SEEN = True


def install_signal_handlers():
    global _INSTALLED
    if _INSTALLED or threading.current_thread() is not threading.main_thread():
        return
    signal.signal(signal.SIGTERM, _handler)
    signal.signal(signal.SIGINT, _handler)
    signal.signal(signal.SIGUSR2, _usr2_handler)
    _INSTALLED = True
    # Readiness marker: the controller only sends SIGUSR2 (whose default
    # disposition TERMINATES a process) to workers that have installed
    # the handler — a directive racing process startup must not kill the
    # worker it is addressed to.
    try:
        import os
        from adaptdl_amd import env
        root = env.checkpoint_path()
        if root is not None and os.path.isdir(root):
            open(os.path.join(root, ".sigusr2-ready-{}".format(
                env.replica_rank())), "w").close()
    except OSError:
        pass


def get_rescale_request():
    """(version, directive) of a pending in-place rescale, or (0, None).

    The directive file is written by the controller before SIGUSR2, so
    reading it after the flag is race-free.  Versions already applied
    in this process (tracked by torch._rejoin) read as not-pending.
    """
    if not _RESCALE_SEEN:
        return 0, None
    import json
    import os
    from adaptdl_amd import env
    from adaptdl_amd.torch import _rejoin
    root = env.checkpoint_path()
    if root is None:
        return 0, None
    path = os.path.join(root, "rescale-inplace.json")
    try:
        with open(path) as f:
            directive = json.load(f)
    except (OSError, ValueError):
        return 0, None
    version = int(directive.get("version", 0))
    if version <= _rejoin.applied_version():
        return 0, None
    return version, directive


def get_exit_flag():
    """True once a SIGTERM/SIGINT (or test-injected exit) was received."""
    return _EXIT_FLAG


def set_exit_flag(value=True):
    """Programmatic trigger used by the in-process allocator on rescale."""
    global _EXIT_FLAG
    _EXIT_FLAG = value
