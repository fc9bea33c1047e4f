"""In-process scale-DOWN: surviving replicas rejoin without restarting.

The checkpoint-restart path (SIGTERM -> save -> exit(143) -> respawn)
is the general rescale mechanism; for pure scale-downs it is wasteful:
data-parallel state is fully replicated, so the surviving processes
already hold everything in HBM.  This module implements the in-place
alternative (north star: "elastic rescale via in-HBM checkpoint and
rejoin"):

- the controller writes a directive file (``rescale-inplace.json`` in
  the job dir) and sends SIGUSR2,
- every worker learns of it via ``_signal.get_rescale_request`` and the
  dataloader's per-iteration control-plane allreduce (so all replicas
  agree on the exact optimizer-cycle boundary at which to act),
- at that boundary, leavers (rank >= new world) simply exit(143) —
  their state lives on in the survivors — and survivors tear down the
  control plane + process group, re-initialize both at the new size,
  and rebind every live GradSyncEngine/GNS to the new world,
- the dataloader then re-partitions the CURRENT pass from the exact
  global sample index (the same mid-epoch resume math a restart uses),
  and training continues: model, optimizer, and GNS tensors never left
  device memory and no process was restarted.

Scale-UPS (and workers that cannot reach a safe point, e.g. BPTT
iterators or eval-only phases) keep the checkpoint-restart path; the
controller escalates to it if the in-place directive is not acted on
within a timeout.
"""

import logging
import os
import sys
import weakref

import torch
import torch.distributed

import adaptdl_amd.collective as collective
import adaptdl_amd.env

LOG = logging.getLogger(__name__)

# GNS instances to rebind on a world-size change (weakrefs: ADP/GNS
# lifetime belongs to the training script).
_LIVE_GNS = []

# Highest directive version already applied in this process.
_applied_version = 0


def register_gns(gns):
    _LIVE_GNS.append(weakref.ref(gns))


def applied_version():
    return _applied_version


class InplaceRescale(Exception):
    """Raised by the dataloader's profile() at a safe cycle boundary;
    caught by AdaptiveDataLoader.__iter__, which performs the rejoin
    and re-partitions the current pass."""

    def __init__(self, directive):
        super().__init__("in-place rescale to {}".format(
            directive.get("world")))
        self.directive = directive


def _pick_free_port():
    import socket
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("0.0.0.0", 0))
        return s.getsockname()[1]


def perform(directive):
    """Execute the agreed in-place scale-down.  Called at an optimizer-
    cycle boundary with all replicas lockstep.  Leavers do not return.
    """
    global _applied_version
    world = int(directive["world"])
    port = int(directive["master_port"])
    rank = adaptdl_amd.env.replica_rank()
    old_world = adaptdl_amd.env.num_replicas()
    backend = (torch.distributed.get_backend()
               if torch.distributed.is_initialized() else None)
    LOG.info("in-place rescale v%s: %d -> %d replicas (rank %d %s)",
             directive.get("version"), old_world, world, rank,
             "leaving" if rank >= world else "staying")

    # All old-group members arrive here together: the teardown barrier
    # and group destruction are collective over the OLD membership.
    collective.teardown()
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()

    if rank >= world:
        # Leaver: state is replicated in the survivors; nothing to save.
        sys.stdout.flush()
        sys.stderr.flush()
        os._exit(143)

    _applied_version = int(directive["version"])
    os.environ["ADAPTDL_NUM_REPLICAS"] = str(world)
    os.environ["ADAPTDL_MASTER_PORT"] = str(port)

    collective.initialize()  # new control plane at the directive's port
    if backend is not None and world >= 1:
        rendezvous = collective.broadcast(_pick_free_port())
        torch.distributed.init_process_group(
            backend, init_method="tcp://{}:{}".format(
                adaptdl_amd.env.master_addr(), rendezvous),
            world_size=world, rank=rank)

    for ref in list(_LIVE_GNS):
        gns = ref()
        if gns is None:
            _LIVE_GNS.remove(ref)
            continue
        gns._rebind_world(world)
    LOG.info("in-place rescale complete: now rank %d of %d", rank, world)
