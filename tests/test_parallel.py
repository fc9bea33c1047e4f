"""End-to-end convergence through the full stack on CPU/gloo.

Mirrors the reference's linear-regression gate
(/root/reference/adaptdl/adaptdl/torch/parallel_test.py:40-68): train
y = 3x1 + 4x2 through AdaptiveDataParallel + AdaptiveDataLoader +
remaining_epochs_until with checkpoint-restarts, and assert the learned
weights converge to (3, 4).
"""

import numpy as np
import torch

import adaptdl_amd.collective as collective
import adaptdl_amd.checkpoint as checkpoint
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@elastic_multiprocessing
def _run_linear_regression():
    import adaptdl_amd.torch as adl
    collective.initialize()
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())

    torch.manual_seed(42)
    true_w = torch.tensor([[3.0], [4.0]])
    xs = torch.randn(128, 2)
    ys = xs @ true_w + 0.01 * torch.randn(128, 1)
    dataset = torch.utils.data.TensorDataset(xs, ys)

    model = torch.nn.Linear(2, 1, bias=False)
    with torch.no_grad():
        model.weight.zero_()
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=16, shuffle=True)

    for epoch in adl.remaining_epochs_until(30):
        for x, y in loader:
            optim.zero_grad()
            loss = ((adp(x) - y) ** 2).mean()
            loss.backward()
            optim.step()
        if env.num_restarts() == 0 and epoch == 5:
            checkpoint.save_all_states()
            collective.teardown()
            torch.distributed.destroy_process_group()
            return 2
        if env.num_restarts() == 1 and epoch == 15:
            checkpoint.save_all_states()
            collective.teardown()
            torch.distributed.destroy_process_group()
            return 3

    w = model.weight.detach().numpy().ravel()
    assert np.allclose(w, [3.0, 4.0], atol=0.1), w
    assert adp.gain >= 1.0
    collective.teardown()
    torch.distributed.destroy_process_group()
    return 0


def test_linear_regression_convergence():
    _run_linear_regression()
