import pytest

from turboprune_amd.config import compose, validate_config
from turboprune_amd.config.compose import parse_cli
from turboprune_amd.config.schema import ConfigError


def test_compose_defaults_groups():
    cfg = compose("cifar10_er_erk")
    assert cfg.dataset_params.dataset_name == "CIFAR10"
    assert cfg.model_params.model_name == "resnet18"
    assert cfg.pruning_params.prune_method == "er_erk"
    assert cfg.optimizer_params.scheduler_type == "TriangularSchedule"
    assert cfg.cyclic_training.num_cycles == 1


def test_compose_imagenet_headline():
    cfg = compose("imagenet_er_balanced")
    assert cfg.model_params.model_name == "resnet50"
    assert cfg.dataset_params.total_batch_size == 512
    assert cfg.experiment_params.distributed is True
    validate_config(cfg)


def test_float_coercion():
    cfg = compose("imagenet_imp")
    assert isinstance(cfg.optimizer_params.weight_decay, float)
    assert cfg.optimizer_params.weight_decay == pytest.approx(1e-4)


def test_dotted_override():
    cfg = compose("cifar10_er_erk", ["optimizer_params.lr=0.05",
                                     "experiment_params.distributed=true"])
    assert cfg.optimizer_params.lr == 0.05
    assert cfg.experiment_params.distributed is True


def test_group_override():
    cfg = compose("cifar10_er_erk", ["pruning_params=iterative_imp"])
    assert cfg.pruning_params.prune_method == "mag"
    assert cfg.pruning_params.training_type == "imp"


def test_add_override():
    cfg = compose("cifar10_er_erk", ["+dataset_params.synthetic_size=128"])
    assert cfg.dataset_params.synthetic_size == 128


def test_parse_cli():
    name, ovs = parse_cli(["--config-name=imagenet_imp", "a.b=1", "+c.d=2"])
    assert name == "imagenet_imp"
    assert ovs == ["a.b=1", "+c.d=2"]


def test_validation_catches_bad_method():
    cfg = compose("cifar10_er_erk", ["pruning_params.prune_method=bogus"])
    with pytest.raises(ConfigError, match="prune_method"):
        validate_config(cfg)


def test_validation_requires_prune_rate_for_mag():
    cfg = compose("cifar10_er_erk", ["pruning_params.prune_method=mag"])
    with pytest.raises(ConfigError, match="prune_rate"):
        validate_config(cfg)


def test_all_shipped_composites_validate():
    for name in ("cifar10_er_erk", "cifar10_er_balanced", "cifar10_er_snip",
                 "cifar10_er_synflow", "cifar100_er_erk",
                 "cifar100_er_balanced", "cifar100_er_snip",
                 "cifar100_er_synflow", "imagenet_er_balanced",
                 "imagenet_imp", "imagenet_synflow", "imagenet_deit_lrr",
                 "bench_resnet50_imagenet", "config"):
        validate_config(compose(name))
