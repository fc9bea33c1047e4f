"""Environment configuration for adaptdl_amd.

All runtime configuration of a training replica comes from ``ADAPTDL_*``
environment variables, kept name-compatible with the reference
(``/root/reference/adaptdl/adaptdl/env.py:23-173``) so existing job specs and
the test harness carry over unchanged.  Unlike the reference (which is
configured by a Kubernetes controller), these variables are normally set by
the in-process node allocator (``adaptdl_amd.sched``) or by
``torch.distributed.run`` when launched under torchrun on one 8-GPU MI355X
node.
"""

import os


def checkpoint_path():
    """Directory where checkpoint-<K> directories are written (shared fs)."""
    return os.getenv("ADAPTDL_CHECKPOINT_PATH")


def warm_checkpoint_path():
    """RAM-backed checkpoint root for elastic rescales (tmpfs, e.g. a
    /dev/shm directory provisioned by the controller).  When set, the
    SIGTERM-rescale checkpoint round-trips through memory instead of
    disk; ``checkpoint_path`` remains the cold (crash-recovery) path."""
    return os.getenv("ADAPTDL_WARM_CHECKPOINT_PATH")


def share_path():
    """Directory shared between all replicas (e.g. for dataset caches)."""
    return os.getenv("ADAPTDL_SHARE_PATH")


def job_id():
    return os.getenv("ADAPTDL_JOB_ID", "local")


def master_addr():
    # torchrun compatibility: fall back to MASTER_ADDR.
    return os.getenv("ADAPTDL_MASTER_ADDR",
                     os.getenv("MASTER_ADDR", "127.0.0.1"))


def master_port():
    """Port used by the control-plane object collectives (rank 0 listens)."""
    port = os.getenv("ADAPTDL_MASTER_PORT")
    if port is not None:
        return int(port)
    # Under torchrun, derive a deterministic side port from MASTER_PORT
    # (which is used by the torch.distributed store itself).
    tr_port = os.getenv("MASTER_PORT")
    if tr_port is not None:
        return int(tr_port) + 17
    return 0


def replica_rank():
    rank = os.getenv("ADAPTDL_REPLICA_RANK", os.getenv("RANK"))
    return int(rank) if rank is not None else 0


def num_replicas():
    n = os.getenv("ADAPTDL_NUM_REPLICAS", os.getenv("WORLD_SIZE"))
    return int(n) if n is not None else 1


def num_nodes():
    n = os.getenv("ADAPTDL_NUM_NODES")
    if n is not None:
        return int(n)
    # Under torchrun: WORLD_SIZE // LOCAL_WORLD_SIZE when both present.
    ws = os.getenv("WORLD_SIZE")
    lws = os.getenv("LOCAL_WORLD_SIZE")
    if ws is not None and lws is not None and int(lws) > 0:
        return max(1, int(ws) // int(lws))
    return 1


def num_restarts():
    return int(os.getenv("ADAPTDL_NUM_RESTARTS", "0"))


def local_rank():
    lr = os.getenv("ADAPTDL_LOCAL_RANK", os.getenv("LOCAL_RANK"))
    if lr is not None:
        return int(lr)
    # Single-node in-process allocator: replica rank IS the GPU index.
    return replica_rank()


def supervisor_url():
    """URL of the (optional) local supervisor; None => fully in-process."""
    return os.getenv("ADAPTDL_SUPERVISOR_URL")


def sched_version():
    return os.getenv("ADAPTDL_SCHED_VERSION")


def from_ray():
    """Ray backend is not part of the MI355X build; always False."""
    return False
