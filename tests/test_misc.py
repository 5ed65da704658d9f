"""Smaller parity/robustness cases across the API surface."""
import json
import os

import pytest
import torch

import maggy_amd
from maggy_amd import Searchspace, util
from maggy_amd.config import TorchDistributedConfig
from maggy_amd.core.environment import Environment


def test_package_surface():
    # the reference's tests import Trial from the package root; the
    # reference itself forgot to export it
    assert hasattr(maggy_amd, "Trial")
    assert hasattr(maggy_amd, "Searchspace")
    assert callable(maggy_amd.experiment.lagom)
    assert maggy_amd.__version__


def test_torch_dist_config_validation():
    import torch.nn as nn

    with pytest.raises(ValueError):
        TorchDistributedConfig(module=nn.Linear, backend="deepspeed")
    with pytest.raises(ValueError):
        TorchDistributedConfig(module=nn.Linear, zero_lvl=7)
    cfg = TorchDistributedConfig(module=nn.Linear, zero_lvl=2)
    assert cfg.backend == "torch"


def test_handle_return_val_errors(tmp_path):
    from maggy_amd.exceptions import MetricTypeError, ReturnTypeError

    with pytest.raises(ReturnTypeError):
        util.handle_return_val(None, str(tmp_path), "Metric")
    with pytest.raises(ReturnTypeError):
        util.handle_return_val("nope", str(tmp_path), "Metric")
    with pytest.raises(KeyError):
        util.handle_return_val({"other": 1.0}, str(tmp_path), "Metric")
    with pytest.raises(MetricTypeError):
        util.handle_return_val({"Metric": "str"}, str(tmp_path), "Metric")
    with pytest.raises(ValueError):
        util.handle_return_val(1.0, str(tmp_path), None)
    # happy path writes both artifacts
    v = util.handle_return_val({"Metric": 2.5, "x": 1}, str(tmp_path),
                               "Metric")
    assert v == 2.5
    assert json.load(open(tmp_path / ".outputs.json"))["x"] == 1
    assert json.load(open(tmp_path / ".metric")) == 2.5


def test_environment_run_ids(tmp_path):
    env = Environment(base_dir=str(tmp_path))
    app = env.get_app_id()
    assert app.startswith("application_")
    r1 = env.next_run_id(app)
    env.get_logdir(app, r1)
    r2 = env.next_run_id(app)
    assert r2 == r1 + 1


def test_searchspace_dict_list_errors():
    sp = Searchspace(a=("DOUBLE", [0.0, 1.0]), b=("INTEGER", [0, 3]))
    with pytest.raises(ValueError):
        sp.list_to_dict([1.0])  # wrong arity
    with pytest.raises(KeyError):
        sp.dict_to_list({"a": 0.5})  # missing name


def test_dataloader_single_process(tmp_path):
    from torch.utils.data import TensorDataset

    from maggy_amd.parallel.data import MaggyDataLoader

    ds = TensorDataset(torch.arange(10).float())
    dl = MaggyDataLoader(ds, batch_size=4)
    batches = [b for (b,) in dl]
    assert sum(len(b) for b in batches) == 10


def test_metric_helpers_cpu():
    from maggy_amd.ops import grad_l2norm, metric_mean, metric_sum
    from maggy_amd.ops.reduce import metric_max

    t = torch.arange(6).float()
    assert metric_sum(t) == 15.0
    assert metric_mean(t) == 2.5
    assert metric_max(t) == 5.0
    p = torch.nn.Parameter(torch.ones(4))
    p.grad = torch.full((4,), 2.0)
    assert grad_l2norm([p]) == pytest.approx(4.0)


def test_reporter_tensor_metric_cpu():
    from maggy_amd.core.reporter import Reporter

    rep = Reporter()
    rep.broadcast(torch.tensor(1.5), 0)          # 0-d tensor
    assert rep.metric == 1.5
    rep.broadcast(torch.tensor([1.0, 3.0]), 1)   # mean-reduced
    assert rep.metric == 2.0


def test_experiment_json_running_state(exp_dir):
    from maggy_amd.config import HyperparameterOptConfig
    from maggy_amd.core.driver import OptimizationDriver

    sp = Searchspace(lr=("DOUBLE", [0.0, 1.0]))
    d = OptimizationDriver(HyperparameterOptConfig(
        num_trials=2, optimizer="randomsearch", searchspace=sp,
        es_policy="none", num_workers=1, name="meta"))
    d.job_start = 1000.0
    meta = json.loads(d.experiment_json())
    assert meta["status"] == "RUNNING"
    assert meta["executors"] == 1
    assert os.path.isdir(meta["logdir"])
