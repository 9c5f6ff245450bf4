"""Checkpoint format + rewind semantics (reference:
utils/harness_utils.py:354-365, custom_models.py:112-166)."""

import os

import torch

from turboprune_amd.config import compose
from turboprune_amd.models import build_model
from turboprune_amd.ops.mask_layers import masked_modules
from turboprune_amd.utils.experiment import (gen_expt_dir, save_model,
                                             unwrap_inner_model)


def _cfg(tmp_path, **overrides):
    ovs = [f"experiment_params.base_dir={tmp_path}/experiments"]
    ovs += [f"{k}={v}" for k, v in overrides.items()]
    return compose("cifar10_er_erk", ovs)


def test_expt_dir_layout_and_prefix(tmp_path):
    cfg = _cfg(tmp_path)
    prefix, expt_dir = gen_expt_dir(cfg)
    assert prefix.startswith("CIFAR10_model_resnet18_trainingtype_at_init"
                             "_prunemethod_er_erk_target_0.90_seed_0")
    assert "_lr_0.200_mom_0.9_wd_0.0005_sched_TriangularSchedule" in prefix
    base = os.path.basename(expt_dir)
    parts = base.split("__")
    assert len(parts) == 3 and parts[0] == prefix
    assert len(parts[1]) == 6  # uuid6
    for sub in ("checkpoints", "metrics", "metrics/level_wise_metrics",
                "artifacts"):
        assert os.path.isdir(os.path.join(expt_dir, sub))


def test_save_model_inner_state_dict(tmp_path):
    cfg = _cfg(tmp_path)
    pm = build_model(cfg)
    path = tmp_path / "model.pt"
    save_model(pm, str(path), distributed=False)
    sd = torch.load(path, weights_only=True)
    # no wrapper prefix — keys are the inner architecture's
    assert "conv1.weight" in sd
    assert "conv1.mask" in sd
    assert not any(k.startswith(("model.", "module.")) for k in sd)
    assert sd["conv1.mask"].dtype == torch.float32


def test_reset_weights_imp_keeps_masks(tmp_path):
    cfg = _cfg(tmp_path, **{"pruning_params.training_type": "imp"})
    prefix, expt_dir = gen_expt_dir(cfg)
    pm = build_model(cfg)
    save_model(pm, os.path.join(expt_dir, "checkpoints", "model_init.pt"),
               False)
    init_weights = {n: m.weight.detach().clone()
                    for n, m in masked_modules(pm.model)}
    # train-ish mutation + prune-ish mask rewrite
    with torch.no_grad():
        for _, m in masked_modules(pm.model):
            m.weight.add_(1.0)
            m.mask.bernoulli_(0.5)
    masks = {n: m.mask.clone() for n, m in masked_modules(pm.model)}
    pm.reset_weights(cfg, expt_dir)
    for n, m in masked_modules(pm.model):
        assert torch.allclose(m.weight.detach(), init_weights[n])  # rewound
        assert torch.equal(m.mask, masks[n])  # masks untouched


def test_reset_weights_lrr_noop(tmp_path):
    cfg = _cfg(tmp_path, **{"pruning_params.training_type": "lrr"})
    prefix, expt_dir = gen_expt_dir(cfg)
    pm = build_model(cfg)
    with torch.no_grad():
        for _, m in masked_modules(pm.model):
            m.weight.add_(1.0)
    before = {n: m.weight.detach().clone()
              for n, m in masked_modules(pm.model)}
    pm.reset_weights(cfg, expt_dir)  # no checkpoint needed: no-op
    for n, m in masked_modules(pm.model):
        assert torch.equal(m.weight.detach(), before[n])


def test_load_only_masks(tmp_path):
    cfg = _cfg(tmp_path)
    pm1 = build_model(cfg)
    with torch.no_grad():
        for _, m in masked_modules(pm1.model):
            m.mask.bernoulli_(0.3)
    path = tmp_path / "donor.pt"
    save_model(pm1, str(path), False)

    pm2 = build_model(cfg)
    w_before = {n: m.weight.detach().clone()
                for n, m in masked_modules(pm2.model)}
    pm2.load_only_masks(str(path))
    for (n, m1), (_, m2) in zip(masked_modules(pm1.model),
                                masked_modules(pm2.model)):
        assert torch.equal(m1.mask, m2.mask)
        assert torch.equal(m2.weight.detach(), w_before[n])  # weights kept


def test_unwrap_inner_model():
    cfg = compose("cifar10_er_erk")
    pm = build_model(cfg)
    assert unwrap_inner_model(pm) is pm.model


def test_reset_optimizer_roundtrip(tmp_path):
    import torch.nn as nn

    from turboprune_amd.utils.console import reset_optimizer
    cfg = _cfg(tmp_path)
    prefix, expt_dir = gen_expt_dir(cfg)
    model = nn.Linear(4, 4)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    model(torch.randn(2, 4)).sum().backward()
    opt.step()
    torch.save(opt.state_dict(),
               os.path.join(expt_dir, "artifacts", "optimizer_init.pt"))
    # perturb state then rewind
    model(torch.randn(2, 4)).sum().backward()
    opt.step()
    reset_optimizer(expt_dir, opt, "imp")
    ref = torch.load(os.path.join(expt_dir, "artifacts",
                                  "optimizer_init.pt"), weights_only=False)
    assert str(opt.state_dict()["state"]) == str(ref["state"])


def test_display_training_info_smoke(capsys):
    from turboprune_amd.utils.console import display_training_info
    display_training_info({"device": "cuda:0", "expt_dir": "/a/b/c"},
                          {"lr": 0.2}, cycle_info={"cycle": 1},
                          training_info={"epochs": 10})
