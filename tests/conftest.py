import multiprocessing as mp
import functools
import os
import signal
import socket
import tempfile

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def pick_unused_port():
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def elastic_multiprocessing(func):
    """Run ``func`` in N forked replica processes with a full ADAPTDL_* env,
    restarting the whole group with K replicas whenever it returns K.

    This simulates rescale-via-checkpoint-restart entirely on localhost/CPU,
    mirroring the behavior of the reference test harness
    (/root/reference/adaptdl/adaptdl/conftest.py:25-100).
    """

    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        num_restarts = 0
        num_replicas = 1
        ctx = mp.get_context("fork")
        with tempfile.TemporaryDirectory() as tmpdir:
            while num_replicas:
                assert isinstance(num_replicas, int)
                master_port = pick_unused_port()
                queue = ctx.Queue()

                def run(rank):
                    # Fork-safety: the parent may have used torch's OpenMP
                    # pool already; a forked child inheriting it deadlocks
                    # on its first parallel region.  Run single-threaded.
                    import torch
                    torch.set_num_threads(1)
                    # The child inherits any module-level singletons other
                    # tests created in the pytest process; reset them so
                    # State registration/loading behaves like a fresh
                    # replica process.
                    import sys as _sys
                    if "adaptdl_amd.checkpoint" in _sys.modules:
                        _sys.modules["adaptdl_amd.checkpoint"] \
                            ._REGISTRY.clear()
                    if "adaptdl_amd.torch._metrics" in _sys.modules:
                        _sys.modules["adaptdl_amd.torch._metrics"] \
                            ._reset_for_tests()
                    if "adaptdl_amd.collective" in _sys.modules:
                        # A main-process test may have initialized the
                        # coordinator; the fork must start fresh (its
                        # socket fds are the parent's).
                        _sys.modules["adaptdl_amd.collective"] \
                            ._COORD = None
                    if "adaptdl_amd.torch.epoch" in _sys.modules:
                        _sys.modules["adaptdl_amd.torch.epoch"] \
                            ._EPOCH_STATE = None
                    if "adaptdl_amd.torch.data" in _sys.modules:
                        data_mod = _sys.modules["adaptdl_amd.torch.data"]
                        data_mod.AdaptiveDataLoaderHelper._current = None
                        data_mod.AdaptiveDataLoaderHelper._training = None
                        data_mod.AdaptiveDataLoaderHelper._position.clear()
                        data_mod._AdaptiveDataLoaderState \
                            .init_count.clear()
                    os.environ["ADAPTDL_CHECKPOINT_PATH"] = str(tmpdir)
                    os.environ["ADAPTDL_JOB_ID"] = "tmpjob"
                    os.environ["ADAPTDL_MASTER_ADDR"] = "127.0.0.1"
                    os.environ["ADAPTDL_MASTER_PORT"] = str(master_port)
                    os.environ["ADAPTDL_REPLICA_RANK"] = str(rank)
                    os.environ["ADAPTDL_NUM_REPLICAS"] = str(num_replicas)
                    os.environ["ADAPTDL_NUM_NODES"] = "1"
                    os.environ["ADAPTDL_NUM_RESTARTS"] = str(num_restarts)
                    os.environ["MASTER_ADDR"] = "127.0.0.1"
                    ret = None
                    try:
                        ret = func(*args, **kwargs)
                    finally:
                        queue.put((rank, ret))

                procs = [ctx.Process(target=run, args=(rank,))
                         for rank in range(num_replicas)]
                for proc in procs:
                    proc.start()
                try:
                    for i in range(num_replicas):
                        rank, ret = queue.get(timeout=180)
                        procs[rank].join()
                        assert procs[rank].exitcode == 0, \
                            "rank {} exited {}".format(
                                rank, procs[rank].exitcode)
                        if i == 0:
                            num_replicas = ret
                        assert num_replicas == ret
                finally:
                    for proc in procs:
                        if proc.is_alive():
                            os.kill(proc.pid, signal.SIGKILL)
                        proc.join()
                    queue.close()
                num_restarts += 1

    return wrapper


@pytest.fixture
def tmp_ckpt_env(tmp_path, monkeypatch):
    """Single-replica local env with a tmp checkpoint dir."""
    monkeypatch.setenv("ADAPTDL_CHECKPOINT_PATH", str(tmp_path))
    monkeypatch.setenv("ADAPTDL_JOB_ID", "tmpjob")
    monkeypatch.setenv("ADAPTDL_REPLICA_RANK", "0")
    monkeypatch.setenv("ADAPTDL_NUM_REPLICAS", "1")
    monkeypatch.setenv("ADAPTDL_NUM_NODES", "1")
    monkeypatch.setenv("ADAPTDL_NUM_RESTARTS", "0")
    yield tmp_path
