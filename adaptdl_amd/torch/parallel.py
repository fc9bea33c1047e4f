"""AdaptiveDataParallel: elastic adaptive-batch-size data parallelism.

API-compatible with the reference class of the same name
(``/root/reference/adaptdl/adaptdl/torch/parallel.py``), but NOT built on
torch DistributedDataParallel: gradient synchronization, bucket management,
GNS statistics, and sync-time measurement are owned by adaptdl_amd's
GradSyncEngine (RCCL over xGMI + fused HIP statistic kernels on MI355X),
which makes the hook/callback ordering the reference depends on (DDP
internals) explicit.

Responsibilities:
- broadcast initial model/buffer state from rank 0,
- toggle gradient sync off during accumulation steps (driven by the
  training dataloader's (atomic_bsz, accum_steps) choice),
- after each synchronized step: record gain / lr factor / progress and
  feed grad params to the goodput model,
- save/restore model + optimizer + lr scheduler + AMP scaler state on
  checkpoint-restart.
"""

import logging
import pickle

import numpy as np
import torch
import torch.distributed

import adaptdl_amd.checkpoint
import adaptdl_amd.env
from adaptdl_amd.torch.data import current_dataloader
from adaptdl_amd.torch.scaling_rules import (AdaScale, AdamScale,
                                             ScalingRuleBase)
from adaptdl_amd.torch.gradient_noise_scale import (GradientNoiseScale,
                                                    AdamGradientNoiseScale)
from adaptdl_amd.torch._metrics import (update_grad_params, update_progress)
from adaptdl_amd.utils import print_exc

LOG = logging.getLogger(__name__)


class AdaptiveDataParallel(torch.nn.Module):
    """Wraps a model for elastic adaptive data-parallel training.

    Arguments:
        model (torch.nn.Module): model to train.
        optimizer (torch.optim.Optimizer): optimizer; patched with the
            chosen scaling rule.
        lr_scheduler: optional LR scheduler (checkpointed).
        mp_scaler: optional torch.amp GradScaler for fp16 training.
        scaling_rule: optional ScalingRuleBase; defaults to AdaScale
            (AdamScale for Adam/AdamW optimizers).
        name: unique name, needed if multiple instances exist.
    """

    def __init__(self, model, optimizer, lr_scheduler=None, mp_scaler=None,
                 scaling_rule=None, name="adaptdl-dataparallel", **kwargs):
        super().__init__()
        self.module = model
        self._key = name

        is_adam = isinstance(optimizer, (torch.optim.Adam,
                                         torch.optim.AdamW))
        if not scaling_rule and is_adam:
            self.scaling_rule = AdamScale()
        else:
            self.scaling_rule = scaling_rule or AdaScale()

        if isinstance(self.scaling_rule, AdamScale):
            self.gns = AdamGradientNoiseScale(self, optimizer,
                                              mp_scaler=mp_scaler)
        else:
            self.gns = GradientNoiseScale(self, optimizer,
                                          mp_scaler=mp_scaler)
        self.scaling_rule.initialize(self, optimizer, patch_optimizer=True)
        if hasattr(optimizer, "attach_engine"):
            # Fused flat-bucket optimizer: flatten parameters into the
            # engine's bucket layout (one fused kernel per bucket).
            optimizer.attach_engine(self.gns.engine)
        self.require_backward_grad_sync = True

        self._sync_module_states()

        self._state = _AdaptiveDataParallelState(
            model, optimizer, lr_scheduler, mp_scaler, name)
        adaptdl_amd.checkpoint.load_state(self._state)

    def _sync_module_states(self):
        """Broadcast params+buffers from rank 0 (replicas start equal)."""
        if not torch.distributed.is_initialized() or \
                torch.distributed.get_world_size() == 1:
            return
        with torch.no_grad():
            for tensor in list(self.module.parameters()) + \
                    list(self.module.buffers()):
                torch.distributed.broadcast(tensor.data, src=0)

    def forward(self, *args, **kwargs):
        dataloader = current_dataloader()
        if dataloader is not None and dataloader.training:
            self.require_backward_grad_sync = dataloader.is_optim_step()
            accum_scale = (dataloader.current_local_bsz *
                           adaptdl_amd.env.num_replicas() /
                           dataloader.batch_size)
            self.gns.set_accum_scale(accum_scale)
        self.gns.engine.require_sync = self.require_backward_grad_sync
        return self.module(*args, **kwargs)

    @print_exc
    def _after_sync(self):
        """Invoked by the GNS once gradients are synchronized each step."""
        dataloader = current_dataloader()
        if dataloader is None:
            raise RuntimeError("backpropagation outside AdaptiveDataLoader")
        dataloader.train()
        scale = dataloader.current_batch_size / dataloader.batch_size
        self._state.gain = self.gns.gain(scale)
        self._state.lr_factor = \
            float(np.average(self.scaling_rule.scale_lr(scale)))
        update_progress(self.gns.get_progress())
        if dataloader.max_batch_size and \
                dataloader.max_batch_size > dataloader.batch_size:
            update_grad_params(self._key, self.gns.sqr_avg(),
                               self.gns.var_avg())

    def zero_grad(self, *args, **kwargs):
        import warnings
        warnings.warn("zero_grad has no effect with AdaptiveDataParallel; "
                      "use optimizer.zero_grad()")

    @property
    def gain(self):
        """Current estimate of the AdaScale gain (r_t)."""
        return self._state.gain

    def to_tensorboard(self, writer, global_step, tag_prefix=""):
        """Write GNS/scaling metrics to a TensorBoard SummaryWriter."""
        if tag_prefix and not tag_prefix.endswith("/"):
            tag_prefix += "/"
        writer.add_scalar(tag_prefix + "Gradient_Norm_Sqr",
                          self.gns.sqr_avg(), global_step)
        writer.add_scalar(tag_prefix + "Gradient_Variance",
                          self.gns.var_avg(), global_step)
        writer.add_scalar(tag_prefix + "Gain", self._state.gain, global_step)
        writer.add_scalar(tag_prefix + "Learning_Rate_Factor",
                          self._state.lr_factor, global_step)
        writer.add_scalar(tag_prefix + "Accum_Scale",
                          self.gns.accum_scale, global_step)
        if self.gns.accum_count > 0:
            writer.add_scalar(tag_prefix + "Accum_Count",
                              self.gns.accum_count, global_step)
        writer.add_scalar(tag_prefix + "Progress",
                          self.gns.get_progress(), global_step)


class _AdaptiveDataParallelState(adaptdl_amd.checkpoint.State):
    """Checkpoints (model, optimizer, lr_scheduler, scaler) state_dicts in
    the same torch.save layout as the reference (parallel.py:218-239)."""

    def __init__(self, model, optimizer, lr_scheduler, mp_scaler,
                 name="adaptdl-dataparallel"):
        super().__init__(name)
        self.model = model
        self.optimizer = optimizer
        self.lr_scheduler = lr_scheduler
        self.mp_scaler = mp_scaler
        self.gain = 1.0
        self.lr_factor = 1.0

    def save(self, fileobj):
        # The optimizer state's "gns" entry holds numpy arrays; keep the
        # torch.save container format of the reference.
        state_dicts = [self.model.state_dict(),
                       _optimizer_state_dict(self.optimizer),
                       self.lr_scheduler.state_dict()
                       if self.lr_scheduler is not None else None,
                       self.mp_scaler.state_dict()
                       if self.mp_scaler is not None else None]
        torch.save((state_dicts, self.gain, self.lr_factor), fileobj)

    def load(self, fileobj):
        state_dicts, self.gain, self.lr_factor = \
            torch.load(fileobj, weights_only=False)
        self.model.load_state_dict(state_dicts[0])
        _load_optimizer_state_dict(self.optimizer, state_dicts[1])
        if state_dicts[2] is not None:
            self.lr_scheduler.load_state_dict(state_dicts[2])
        if state_dicts[3] is not None:
            self.mp_scaler.load_state_dict(state_dicts[3])


def _optimizer_state_dict(optimizer):
    """Optimizer state_dict including the "gns" entry.

    torch's Optimizer.state_dict() only covers per-param state; the GNS
    state lives under the string key "gns" and is serialized alongside.
    """
    sd = optimizer.state_dict()
    gns = optimizer.state.get("gns")
    return {"torch": sd, "gns": pickle.dumps(gns)}


def _load_optimizer_state_dict(optimizer, sd):
    if isinstance(sd, dict) and "torch" in sd:
        optimizer.load_state_dict(sd["torch"])
        gns = pickle.loads(sd["gns"])
        if gns is not None:
            optimizer.state["gns"] = gns
    else:
        optimizer.load_state_dict(sd)
