"""Multi-process distributed training over gloo (world_size=2, CPU)."""
import pytest

from maggy_amd import experiment
from maggy_amd.config import TorchDistributedConfig
from tests import _dist_fns as fns


@pytest.mark.timeout(180)
def test_ddp_training_gloo(exp_dir):
    cfg = TorchDistributedConfig(
        module=fns.TinyNet, hparams={"hidden": 16}, num_gpus=2,
        name="ddp-gloo")
    res = experiment.lagom(fns.dist_train_fn, cfg)
    assert res["world_size"] == 2
    assert res["final_metric_avg"] is not None
    assert set(res["per_rank"].keys()) == {"0", "1"}


@pytest.mark.timeout(180)
def test_patched_dataloader_shards(exp_dir):
    cfg = TorchDistributedConfig(
        module=fns.TinyNet, num_gpus=2, name="dl-gloo")
    res = experiment.lagom(fns.dist_dataloader_fn, cfg)
    assert res["final_metric_avg"] == 1.0


@pytest.mark.timeout(180)
def test_zero_sharded_optimizer_gloo(exp_dir):
    cfg = TorchDistributedConfig(
        module=fns.TinyNet, num_gpus=2, zero_lvl=1, name="zero-gloo")
    res = experiment.lagom(fns.dist_zero_fn, cfg)
    assert res["final_metric_avg"] is not None


@pytest.mark.timeout(180)
def test_dist_rank_failure_raises(exp_dir):
    from maggy_amd.exceptions import WorkerCrashError

    cfg = TorchDistributedConfig(
        module=fns.TinyNet, num_gpus=2, name="crash-gloo")
    with pytest.raises(WorkerCrashError):
        experiment.lagom(fns.dist_crashing_fn, cfg)


@pytest.mark.timeout(180)
def test_zero2_grad_sharding_gloo(exp_dir):
    cfg = TorchDistributedConfig(
        module=fns.TinyNet, num_gpus=2, zero_lvl=2, name="zero2-gloo")
    res = experiment.lagom(fns.dist_zero2_fn, cfg)
    assert res["final_metric_avg"] is not None


@pytest.mark.timeout(180)
def test_mixed_precision_autocast_gloo(exp_dir):
    cfg = TorchDistributedConfig(
        module=fns.TinyNet, num_gpus=2, mixed_precision=True,
        name="amp-gloo")
    res = experiment.lagom(fns.dist_autocast_fn, cfg)
    assert res["final_metric_avg"] == 1.0


def test_zero_lvl_3_rejected():
    with pytest.raises(ValueError, match="zero_lvl"):
        TorchDistributedConfig(module=fns.TinyNet, zero_lvl=3)


@pytest.mark.timeout(300)
def test_llama_tiny_ddp_8rank_gloo(exp_dir):
    """8-rank data-parallel Llama-tiny on CPU/gloo: the same world size the
    driver's SCALE run uses on 8 GPUs."""
    cfg = TorchDistributedConfig(
        module=fns.TinyLlama, hparams={"vocab": 256}, num_gpus=8,
        name="llama8-gloo")
    res = experiment.lagom(fns.dist_llama_fn, cfg)
    assert res["world_size"] == 8
    assert len(res["per_rank"]) == 8
    assert res["final_metric_avg"] is not None
