"""adaptdl_amd.torch: the adaptive training API.

Drop-in surface of the reference ``adaptdl.torch`` package
(``/root/reference/adaptdl/adaptdl/torch/__init__.py:132-142``):
``init_process_group``, ``AdaptiveDataParallel``, ``AdaptiveDataLoader``,
``ElasticSampler``, ``Accumulator``, epoch helpers, and scaling rules —
rebuilt MI355X-first (RCCL over xGMI data plane, fused HIP statistics).
"""

import logging
import os
import socket

import torch.distributed

import adaptdl_amd.collective
import adaptdl_amd.env
from adaptdl_amd._signal import install_signal_handlers

from adaptdl_amd.torch.epoch import (current_epoch, finished_epochs,  # noqa
                                     remaining_epochs_until)
from adaptdl_amd.torch.data import (AdaptiveDataLoader,  # noqa
                                    AdaptiveDataLoaderHelper,
                                    AdaptiveDataLoaderMixin,  # noqa: F401
                                    ElasticSampler, current_dataloader)
from adaptdl_amd.torch.parallel import AdaptiveDataParallel  # noqa
from adaptdl_amd.torch.accumulator import Accumulator  # noqa
from adaptdl_amd.torch.iterator import AdaptiveBPTTIterator  # noqa
from adaptdl_amd.torch.optim import (FusedSGD, FusedAdam,  # noqa
                                     FusedAdamW)
from adaptdl_amd.torch.graph_step import (GraphedStepper,  # noqa
                                          maybe_graphed_stepper)

LOG = logging.getLogger(__name__)

__all__ = [
    "init_process_group",
    "current_epoch",
    "finished_epochs",
    "remaining_epochs_until",
    "current_dataloader",
    "AdaptiveDataLoader",
    "AdaptiveDataLoaderHelper",
    "AdaptiveDataLoaderMixin",
    "ElasticSampler",
    "AdaptiveDataParallel",
    "Accumulator",
    "AdaptiveBPTTIterator",
    "FusedSGD",
    "FusedAdam",
    "FusedAdamW",
    "GraphedStepper",
    "maybe_graphed_stepper",
]


def version_check():
    """Warn when the scheduler that launched us is a different version
    (reference torch/__init__.py:43-48 parity)."""
    import adaptdl_amd
    sched = adaptdl_amd.env.sched_version()
    if sched is not None and sched != adaptdl_amd.__version__:
        LOG.warning(
            "adaptdl_amd version %s differs from scheduler version %s",
            adaptdl_amd.__version__, sched)


def _pick_free_port():
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("0.0.0.0", 0))
        return s.getsockname()[1]


def init_process_group(backend=None, init_method=None, world_size=None,
                       rank=None):
    """Initialize the control plane and the torch.distributed data plane.

    On MI355X nodes use backend="nccl" (RCCL over xGMI); on CPU use "gloo".
    With no arguments, the backend is chosen automatically.  Replica
    rank/count and the rendezvous address come from the ``ADAPTDL_*`` (or
    torchrun) environment (reference flow: torch/__init__.py:51-127 minus
    the Kubernetes supervisor long-poll, which the in-process allocator
    replaces).
    """
    install_signal_handlers()
    version_check()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    num_replicas = world_size if world_size is not None \
        else adaptdl_amd.env.num_replicas()
    replica_rank = rank if rank is not None \
        else adaptdl_amd.env.replica_rank()
    master_addr = adaptdl_amd.env.master_addr()

    if not adaptdl_amd.collective.initialized():
        adaptdl_amd.collective.initialize(master_addr)

    if torch.distributed.is_initialized():
        return

    if init_method is None:
        if os.getenv("TORCHELASTIC_RUN_ID") is not None:
            # Under torchrun the agent process hosts the c10d store and
            # every worker is a store CLIENT (TORCHELASTIC_USE_AGENT_STORE)
            # — a fresh tcp:// port would have no server.  Use the
            # torchrun-provided env:// rendezvous directly.
            init_method = "env://"
        else:
            # Broadcast a fresh rendezvous port from rank 0 over the
            # control plane (a new port every (re)start avoids TIME_WAIT
            # collisions).
            port = adaptdl_amd.collective.broadcast(_pick_free_port())
            init_method = "tcp://{}:{}".format(master_addr, port)
    if backend == "nccl":
        torch.cuda.set_device(adaptdl_amd.env.local_rank()
                              % max(torch.cuda.device_count(), 1))
    torch.distributed.init_process_group(
        backend, init_method=init_method,
        world_size=num_replicas, rank=replica_rank)
    LOG.info("initialized process group: backend=%s world_size=%d rank=%d",
             backend, num_replicas, replica_rank)
