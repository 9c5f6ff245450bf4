import pytest
import torch

from turboprune_amd.config import compose
from turboprune_amd.models import build_model, resnet18, resnet50
from turboprune_amd.models.deit import (local_deit_small_patch16_224,
                                        local_deit_tiny_distilled_patch16_224)
from turboprune_amd.models.vgg import vgg16_bn
from turboprune_amd.ops.mask_layers import masked_modules


def test_resnet18_cifar_forward():
    m = resnet18(num_classes=10, cifar_stem=True)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_resnet50_imagenet_forward():
    m = resnet50(num_classes=1000)
    y = m(torch.randn(2, 3, 64, 64))  # small spatial for CPU speed
    assert y.shape == (2, 1000)


def test_resnet50_masked_layer_count():
    m = resnet50()
    convs = [n for n, _ in masked_modules(m)]
    # 53 convs + fc (as Conv1dMask) = 54 masked layers in ResNet50
    assert len(convs) == 54


def test_resnet_state_dict_naming_torchvision_compatible():
    sd = resnet50().state_dict()
    for key in ("conv1.weight", "bn1.weight", "layer1.0.conv1.weight",
                "layer1.0.downsample.0.weight", "layer4.2.bn3.running_mean",
                "fc.weight", "fc.bias", "layer1.0.conv1.mask", "conv1.mask"):
        assert key in sd, key
    assert sd["fc.weight"].shape == (1000, 2048, 1)  # Conv1dMask shape


def test_vgg16_forward():
    m = vgg16_bn(num_classes=10, cifar_stem=True)
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_deit_small_forward():
    m = local_deit_small_patch16_224(num_classes=1000)
    y = m(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)
    names = [n for n, _ in masked_modules(m)]
    assert "patch_embed.proj" in names
    assert "blocks.0.attn.qkv" in names
    assert "blocks.11.mlp.fc2" in names
    assert "head" in names


def test_deit_distilled_train_eval():
    m = local_deit_tiny_distilled_patch16_224(num_classes=10)
    x = torch.randn(1, 3, 224, 224)
    m.train()
    out = m(x)
    assert isinstance(out, tuple) and len(out) == 2
    m.eval()
    out = m(x)
    assert out.shape == (1, 10)


def test_build_model_from_cfg():
    cfg = compose("cifar10_er_erk")
    pm = build_model(cfg)
    assert pm(torch.randn(1, 3, 32, 32)).shape == (1, 10)
    # CIFAR stem surgery applied
    assert pm.model.conv1.kernel_size == (3, 3)
    assert isinstance(pm.model.maxpool, torch.nn.Identity)

    cfg = compose("imagenet_er_balanced")
    pm = build_model(cfg)
    assert pm.model.conv1.kernel_size == (7, 7)


def test_prunemodel_sparsity_and_reset():
    cfg = compose("cifar10_er_erk")
    pm = build_model(cfg)
    assert pm.get_overall_sparsity() == 0.0
    for _, m in masked_modules(pm.model):
        m.mask.zero_()
    assert pm.get_overall_sparsity() == pytest.approx(100.0)
    pm.reset_masks()
    assert pm.get_overall_sparsity() == 0.0


def test_build_model_unknown_raises():
    cfg = compose("cifar10_er_erk", ["model_params.model_name=nonexistent"])
    with pytest.raises(ValueError, match="unknown model"):
        build_model(cfg)


def test_resnet152_and_wide_variants():
    """Extended zoo: torchvision-named deep/wide ResNets (torchvision is
    not installed in the image; the native zoo is the model surface)."""
    import torch

    from turboprune_amd.models import available_models, build_model
    from turboprune_amd.config import compose
    names = available_models()
    assert {"resnet152", "wide_resnet50_2", "wide_resnet101_2"} <= set(names)

    cfg = compose("bench_resnet50_imagenet",
                  ["model_params.model_name=wide_resnet50_2"])
    pm = build_model(cfg, num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = pm(x)
    assert y.shape == (2, 10)
    # wide: bottleneck conv2 of layer1 has 128 channels (64 * 128/64)
    w = pm.model.layer1[0].conv2.weight
    assert w.shape[0] == 128 and w.shape[1] == 128
    # resnet50 unchanged by the width plumbing
    cfg50 = compose("bench_resnet50_imagenet")
    pm50 = build_model(cfg50, num_classes=10)
    assert pm50.model.layer1[0].conv2.weight.shape[0] == 64
