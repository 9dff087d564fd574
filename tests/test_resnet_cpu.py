"""ResNet model family on CPU: shapes, torchvision-compatible state_dict
layout, registry, residual BN fallback semantics (BASELINE configs 4-5)."""
import torch

from ddp_tricks_amd.models import build_model, resnet18, resnet50


def test_resnet18_cifar_shapes():
    m = resnet18(num_classes=10, cifar_stem=True)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_resnet50_imagenet_shapes():
    m = resnet50()
    y = m(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)


def test_param_counts_match_torchvision():
    # canonical torchvision counts for these architectures
    assert sum(p.numel() for p in resnet18().parameters()) == 11_689_512
    assert sum(p.numel() for p in resnet50().parameters()) == 25_557_032


def test_state_dict_keys_torchvision_layout():
    sd = resnet18().state_dict()
    for k in ("conv1.weight", "bn1.weight", "bn1.running_mean",
              "layer1.0.conv1.weight", "layer2.0.downsample.0.weight",
              "layer2.0.downsample.1.running_var", "layer4.1.bn2.bias",
              "fc.weight", "fc.bias"):
        assert k in sd, k
    sd50 = resnet50().state_dict()
    for k in ("layer1.0.conv3.weight", "layer1.0.bn3.weight",
              "layer1.0.downsample.0.weight", "layer3.5.conv2.weight"):
        assert k in sd50, k


def test_registry():
    m = build_model("resnet18", num_classes=10, cifar_stem=True)
    assert m(torch.randn(1, 3, 32, 32)).shape == (1, 10)


def test_residual_bn_cpu_semantics():
    """CPU fallback of batch_norm(residual=...) == bn -> +res -> relu."""
    from ddp_tricks_amd.ops.functional import batch_norm
    torch.manual_seed(0)
    x = torch.randn(4, 8, 5, 5)
    res = torch.randn(4, 8, 5, 5)
    w = torch.rand(8) + 0.5
    b = torch.randn(8)
    rm = torch.zeros(8)
    rv = torch.ones(8)
    y = batch_norm(x, rm, rv, w, b, True, 0.1, 1e-5, fuse_relu=True,
                   residual=res)
    ref = (torch.nn.functional.batch_norm(
        x, torch.zeros(8), torch.ones(8), w, b, True, 0.1, 1e-5) + res).relu()
    assert torch.allclose(y, ref, atol=1e-6)


def test_maxpool_padding_cpu():
    from ddp_tricks_amd.ops.functional import max_pool2d
    x = torch.randn(2, 4, 9, 9)
    y = max_pool2d(x, 3, 2, 1)
    ref = torch.nn.functional.max_pool2d(x, 3, 2, 1)
    assert torch.equal(y, ref)


def test_global_avgpool_cpu():
    from ddp_tricks_amd.ops.functional import global_avg_pool2d
    x = torch.randn(2, 4, 7, 7)
    assert torch.allclose(global_avg_pool2d(x), x.mean(dim=(2, 3)))


def test_train_resnet18_cifar10_cpu(tmp_path, monkeypatch):
    """train(args) with --model resnet18 --dataset cifar10 (config 4 path)."""
    import os
    import types

    import torch.distributed as dist

    from ddp_tricks_amd import amp
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "128")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29682")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("gloo", rank=0, world_size=1)
    amp._state.__init__()
    from ddp_tricks_amd.utils.train import train
    args = types.SimpleNamespace(
        exp_name="R18", learning_rate=0.05, batch_size=64, epochs=1,
        warmup_epochs=1, warmup_type="linear", seed_num=42,
        data_path="/nonexistent", model_path=str(tmp_path), local_rank=0,
        model="resnet18", dataset="cifar10")
    train(args)
    import torch
    sd = torch.load(os.path.join(tmp_path, "R18.pt"), weights_only=True)
    assert "layer4.1.bn2.running_var" in sd and sd["fc.weight"].shape == (10, 512)


def test_flatten_cpu_fallback_semantics():
    """Flatten's CPU path (and any non-dispatch GPU case) must equal
    torch.flatten on channels_last input — NCHW semantic order."""
    import torch
    from ddp_tricks_amd.ops.modules import Flatten
    x = torch.randn(3, 5, 2, 4).contiguous(memory_format=torch.channels_last)
    y = Flatten()(x)
    assert torch.equal(y, torch.flatten(x, 1))


def test_vgg16_cpu_forward_backward():
    """VGG-16-BN zoo extension: shapes, torchvision-style keys, grads."""
    import torch

    from ddp_tricks_amd.models import build_model
    for kw, shape, ncls in [({"num_classes": 7}, (2, 3, 224, 224), 7),
                            ({"num_classes": 10, "cifar_head": True},
                             (2, 3, 32, 32), 10)]:
        m = build_model("vgg16", **kw)
        y = m(torch.randn(*shape))
        assert y.shape == (2, ncls)
        y.pow(2).mean().backward()
        assert all(p.grad is not None for p in m.parameters())
    keys = m.state_dict()
    assert "features.0.weight" in keys and "features.1.running_mean" in keys
    assert "classifier.6.weight" in keys   # torchvision vgg16_bn layout


def test_vgg16_dropout_slot():
    import torch

    from ddp_tricks_amd.models import build_model
    m = build_model("vgg16", num_classes=10, cifar_head=True, dropout=0.5)
    m.train()
    torch.manual_seed(0)
    y1 = m(torch.ones(4, 3, 32, 32))
    y2 = m(torch.ones(4, 3, 32, 32))
    assert not torch.equal(y1, y2)   # dropout active in train mode
    m.eval()
    with torch.no_grad():
        z1 = m(torch.ones(4, 3, 32, 32))
        z2 = m(torch.ones(4, 3, 32, 32))
    assert torch.equal(z1, z2)
