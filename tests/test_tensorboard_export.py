"""TensorBoard exporter surfaces (ADP + dataloader) with a fake writer
(the tensorboard package is not installed in this environment; the
exporters only need the add_scalar protocol)."""

import torch

import adaptdl_amd.collective as collective
import adaptdl_amd.torch as adl


class FakeWriter:
    def __init__(self):
        self.scalars = {}

    def add_scalar(self, tag, value, global_step):
        self.scalars[tag] = (float(value), global_step)


def test_adp_and_loader_exporters(tmp_ckpt_env):
    collective.initialize(master_addr="127.0.0.1")
    model = torch.nn.Linear(4, 2)
    optim = torch.optim.SGD(model.parameters(), lr=0.1)
    adp = adl.AdaptiveDataParallel(model, optim, name="tb-test")
    dataset = torch.utils.data.TensorDataset(torch.randn(32, 4),
                                             torch.randn(32, 2))
    loader = adl.AdaptiveDataLoader(dataset, batch_size=8)

    for epoch in adl.remaining_epochs_until(1):
        for x, y in loader:
            optim.zero_grad()
            torch.nn.functional.mse_loss(adp(x), y).backward()
            optim.step()

    w = FakeWriter()
    adp.to_tensorboard(w, global_step=1, tag_prefix="AdaptDL")
    loader.to_tensorboard(w, global_step=1, tag_prefix="AdaptDL")
    tags = set(w.scalars)
    assert "AdaptDL/Gradient_Norm_Sqr" in tags
    assert "AdaptDL/Gradient_Variance" in tags
    assert "AdaptDL/Gain" in tags
    assert any("Batch" in t for t in tags), tags
    assert all(torch.isfinite(torch.tensor(v)) for v, _ in
               w.scalars.values())
    collective.teardown()   # forked tests re-initialize their own
