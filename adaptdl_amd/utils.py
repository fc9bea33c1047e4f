"""Small shared helpers."""

import functools
import logging
import traceback

LOG = logging.getLogger(__name__)


def print_exc(func):
    """Log exceptions escaping autograd hooks/callbacks (which swallow them).

    Mirrors the debugging aid of the reference
    (``/root/reference/adaptdl/adaptdl/utils.py``).
    """

    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        try:
            return func(*args, **kwargs)
        except Exception:
            LOG.error("exception in %s:\n%s", func.__qualname__,
                      traceback.format_exc())
            raise

    return wrapper
