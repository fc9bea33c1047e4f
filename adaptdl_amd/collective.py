"""Control-plane object collectives (allreduce / broadcast of Python objects).

This is the small-message control plane used for coordination that must work
*before* (and independently of) the RCCL data plane: broadcasting the
rendezvous port, per-iteration exit-flag OR-allreduce, batch-size broadcast,
and Accumulator dict reduction.  Mirrors the API of the reference
(``/root/reference/adaptdl/adaptdl/collective.py:37-144`` and ``reducer.py``)
but is a new implementation: a single selector-driven server thread on rank 0
folds values in rank order per sequence number and replies to every rank,
so there is no reply-ordering/GIL-deadlock workaround, and each client owns
one receiver thread that resolves futures strictly in sequence order.

Stays on the CPU/TCP path deliberately: one tiny message per iteration must
not touch the GPU streams or RCCL (which would serialize with gradient
all-reduce on xGMI).
"""

import logging
import pickle
import selectors
import socket
import struct
import threading
import time

from adaptdl_amd import env

LOG = logging.getLogger(__name__)

_HDR = struct.Struct("!Q")


def default_reduce_fn(a, b):
    return a + b


class Future(object):
    """Result placeholder for an asynchronous collective."""

    def __init__(self):
        self._event = threading.Event()
        self._result = None
        self._exc = None

    def set(self, result, exc=None):
        self._result = result
        self._exc = exc
        self._event.set()

    def result(self, timeout=None):
        if not self._event.wait(timeout):
            raise TimeoutError("collective result timed out")
        if self._exc is not None:
            raise self._exc
        return self._result


def _send_msg(sock, obj):
    data = pickle.dumps(obj)
    sock.sendall(_HDR.pack(len(data)) + data)


def _recv_exactly(sock, n):
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("collective peer closed connection")
        buf += chunk
    return bytes(buf)


def _recv_msg(sock):
    (n,) = _HDR.unpack(_recv_exactly(sock, _HDR.size))
    return pickle.loads(_recv_exactly(sock, n))


class _Server(threading.Thread):
    """Rank-0 server: folds per-sequence values in rank order, replies all.

    The fold for sequence ``seq`` uses the reduce_fn supplied by rank 0's own
    call for that sequence (all replicas run the same program, so the
    functions agree); folding is deferred until rank 0's value arrives.
    """

    def __init__(self, port_sock, num_replicas):
        super().__init__(daemon=True, name="adaptdl-collective-server")
        self._listen = port_sock
        self._num = num_replicas
        self._socks = {}          # rank -> socket (excluding rank 0)
        self._pending = {}        # seq -> {rank: value}
        self._fns = {}            # seq -> reduce_fn (from rank 0)
        self._local = {}          # seq -> Future for rank 0
        self._lock = threading.Lock()
        # Replies may be sent from the server thread or from rank 0's calling
        # thread (whichever completes a sequence last); serialize socket use.
        self._send_lock = threading.Lock()
        self._sel = selectors.DefaultSelector()

    def submit_local(self, seq, value, reduce_fn, future):
        with self._lock:
            self._fns[seq] = reduce_fn
            self._local[seq] = future
            self._pending.setdefault(seq, {})[0] = value
        self._maybe_complete(seq)

    def _maybe_complete(self, seq):
        with self._lock:
            vals = self._pending.get(seq)
            if vals is None or len(vals) < self._num or seq not in self._fns:
                return
            fn = self._fns.pop(seq)
            self._pending.pop(seq)
            future = self._local.pop(seq)
            socks = dict(self._socks)
        result = vals[0]
        for rank in range(1, self._num):
            result = fn(result, vals[rank])
        with self._send_lock:
            for rank, sock in socks.items():
                try:
                    _send_msg(sock, (seq, result))
                except OSError:
                    LOG.warning("failed replying to rank %d", rank)
        future.set(result)

    def run(self):
        try:
            # Accept one connection per non-zero rank.
            while len(self._socks) < self._num - 1:
                sock, _ = self._listen.accept()
                sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                rank = _recv_msg(sock)
                self._socks[rank] = sock
                self._sel.register(sock, selectors.EVENT_READ, rank)
            self._listen.close()
            while True:
                for key, _ in self._sel.select():
                    rank = key.data
                    try:
                        seq, value = _recv_msg(key.fileobj)
                    except (ConnectionError, OSError):
                        self._sel.unregister(key.fileobj)
                        return
                    with self._lock:
                        self._pending.setdefault(seq, {})[rank] = value
                    self._maybe_complete(seq)
        except Exception:  # noqa: BLE001 - server dies with the process
            LOG.exception("collective server terminated")


class _Client(threading.Thread):
    """Receiver thread on ranks > 0: resolves futures in sequence order."""

    def __init__(self, sock):
        super().__init__(daemon=True, name="adaptdl-collective-client")
        self._sock = sock
        self._futures = {}
        self._lock = threading.Lock()

    def register(self, seq, future):
        with self._lock:
            self._futures[seq] = future

    def run(self):
        try:
            while True:
                seq, result = _recv_msg(self._sock)
                with self._lock:
                    future = self._futures.pop(seq, None)
                if future is not None:
                    future.set(result)
        except (ConnectionError, OSError):
            with self._lock:
                for f in self._futures.values():
                    f.set(None, ConnectionError("collective server gone"))


class Coordinator(object):

    def __init__(self, rank, num_replicas, master_addr, master_port,
                 timeout=120.0):
        self._rank = rank
        self._num = num_replicas
        self._seq = 0
        self._seq_lock = threading.Lock()
        self._server = None
        self._client = None
        self._sock = None
        if num_replicas == 1:
            return
        if rank == 0:
            listener = socket.create_server(("0.0.0.0", master_port))
            self._server = _Server(listener, num_replicas)
            self._server.start()
        else:
            deadline = time.monotonic() + timeout
            while True:
                try:
                    self._sock = socket.create_connection(
                        (master_addr, master_port), timeout=5)
                    break
                except OSError:
                    if time.monotonic() > deadline:
                        raise
                    time.sleep(0.25)
            self._sock.settimeout(None)  # connect timeout must not persist
            self._sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            _send_msg(self._sock, rank)
            self._client = _Client(self._sock)
            self._client.start()
        # Barrier: initialize() must not return until every replica has
        # joined, so no rank can finish a collective-free program section and
        # exit (killing the rank-0 server) before the others connect.
        self.allreduce(None, lambda a, b: a)

    def allreduce_async(self, value, reduce_fn):
        future = Future()
        if self._num == 1:
            future.set(value)
            return future
        with self._seq_lock:
            seq = self._seq
            self._seq += 1
            if self._rank == 0:
                self._server.submit_local(seq, value, reduce_fn, future)
            else:
                self._client.register(seq, future)
                _send_msg(self._sock, (seq, value))
        return future

    def allreduce(self, value, reduce_fn):
        return self.allreduce_async(value, reduce_fn).result()

    def broadcast(self, value):
        return self.allreduce(value, lambda a, b: a)

    def close(self):
        # Best-effort exit barrier: lets in-flight matched collectives drain
        # and keeps the rank-0 server alive until every replica is done.
        if self._num > 1:
            try:
                self.allreduce_async(None, lambda a, b: a).result(timeout=60)
            except Exception:  # noqa: BLE001
                LOG.warning("collective teardown barrier failed")
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass


_COORD = None


def initialize(master_addr=None, master_port=None, replica_rank=None,
               num_replicas=None):
    """Initialize the control plane; blocks until all replicas connect."""
    global _COORD
    if _COORD is not None:
        raise RuntimeError("collective is already initialized")
    _COORD = Coordinator(
        env.replica_rank() if replica_rank is None else replica_rank,
        env.num_replicas() if num_replicas is None else num_replicas,
        master_addr if master_addr is not None else env.master_addr(),
        master_port if master_port is not None else env.master_port())


def initialized():
    return _COORD is not None


def teardown():
    global _COORD
    if _COORD is not None:
        _COORD.close()
        _COORD = None


def _coord():
    if _COORD is None:
        raise RuntimeError("adaptdl_amd.collective is not initialized")
    return _COORD


def allreduce(value, reduce_fn=default_reduce_fn):
    """Reduce ``value`` across all replicas; every replica gets the result."""
    return _coord().allreduce(value, reduce_fn)


def allreduce_async(value, reduce_fn=default_reduce_fn):
    """Asynchronous :func:`allreduce`; returns a :class:`Future`."""
    return _coord().allreduce_async(value, reduce_fn)


def broadcast(value):
    """Broadcast ``value`` from replica 0 to all replicas."""
    return _coord().broadcast(value)
