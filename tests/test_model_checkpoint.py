"""Toy_Net layout/init/forward parity + checkpoint layout conformance
(reference utils/model.py)."""
import os

import torch
import torch.nn as nn

from ddp_tricks_amd import Toy_Net, same_seeds

EXPECTED_KEYS = [
    "conv.0.weight", "conv.0.bias",
    "conv.1.weight", "conv.1.bias", "conv.1.running_mean",
    "conv.1.running_var", "conv.1.num_batches_tracked",
    "conv.3.weight", "conv.3.bias",
    "conv.4.weight", "conv.4.bias", "conv.4.running_mean",
    "conv.4.running_var", "conv.4.num_batches_tracked",
    "conv.7.weight", "conv.7.bias",
    "conv.8.weight", "conv.8.bias", "conv.8.running_mean",
    "conv.8.running_var", "conv.8.num_batches_tracked",
    "conv.10.weight", "conv.10.bias",
    "conv.11.weight", "conv.11.bias", "conv.11.running_mean",
    "conv.11.running_var", "conv.11.num_batches_tracked",
    "dense.1.weight", "dense.1.bias",
    "dense.2.weight", "dense.2.bias", "dense.2.running_mean",
    "dense.2.running_var", "dense.2.num_batches_tracked",
    "dense.4.weight", "dense.4.bias",
]


class RefToyNet(nn.Module):
    """Plain-torch replica of the reference architecture (oracle)."""

    def __init__(self):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(1, 64, 3), nn.BatchNorm2d(64), nn.ReLU(True),
            nn.Conv2d(64, 128, 3), nn.BatchNorm2d(128), nn.ReLU(True),
            nn.MaxPool2d(2),
            nn.Conv2d(128, 256, 3), nn.BatchNorm2d(256), nn.ReLU(True),
            nn.Conv2d(256, 512, 3), nn.BatchNorm2d(512), nn.ReLU(True),
            nn.MaxPool2d(2))
        self.dense = nn.Sequential(
            nn.Flatten(), nn.Linear(8192, 512), nn.BatchNorm1d(512),
            nn.ReLU(True), nn.Linear(512, 10))

    def forward(self, x):
        return self.dense(self.conv(x))


def test_state_dict_layout():
    sd = Toy_Net().state_dict()
    assert list(sd.keys()) == EXPECTED_KEYS
    assert sum(v.numel() for k, v in sd.items() if "weight" in k or "bias" in k
               or "running" in k) > 0
    n_params = sum(p.numel() for p in Toy_Net().parameters())
    assert n_params == 5752714


def test_init_and_forward_parity():
    same_seeds(42)
    ours = Toy_Net()
    same_seeds(42)
    ref = RefToyNet()
    x = torch.randn(8, 1, 28, 28)
    ours.eval(), ref.eval()
    assert torch.equal(ours(x), ref(x))


def test_checkpoint_loads_into_torch_definition(tmp_path):
    m = Toy_Net()
    path = os.path.join(tmp_path, "ckpt.pt")
    torch.save(m.state_dict(), path)
    ref = RefToyNet()
    ref.load_state_dict(torch.load(path))  # strict=True: byte-compatible keys


def test_backward_parity_fp32():
    same_seeds(1)
    ours = Toy_Net()
    same_seeds(1)
    ref = RefToyNet()
    ours.train(), ref.train()
    x = torch.randn(16, 1, 28, 28)
    t = torch.randint(0, 10, (16,))
    lo = torch.nn.functional.cross_entropy(ours(x), t)
    lr = torch.nn.functional.cross_entropy(ref(x), t)
    lo.backward(), lr.backward()
    assert torch.allclose(lo, lr)
    for (ka, pa), (kb, pb) in zip(ours.named_parameters(), ref.named_parameters()):
        assert torch.allclose(pa.grad, pb.grad, atol=1e-6), ka
