"""End-to-end experiment engine on a real GPU (1x MI355X)."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs an AMD GPU", allow_module_level=True)


def test_hpo_on_gpu_pool(exp_dir):
    from maggy_amd import Searchspace, experiment
    from maggy_amd.config import HyperparameterOptConfig
    from tests._gpu_train_fns import gpu_trial_fn

    sp = Searchspace(lr=("DOUBLE", [1e-3, 1e-2]))
    cfg = HyperparameterOptConfig(
        num_trials=3, optimizer="randomsearch", searchspace=sp,
        direction="min", es_policy="none", num_workers=1, name="gpu-hpo")
    res = experiment.lagom(gpu_trial_fn, cfg)
    assert res["num_trials"] == 3
    assert res["best_val"] is not None
    # artifacts written
    app = [d for d in os.listdir(exp_dir)
           if d.startswith("application_")][0]
    run_dir = os.path.join(exp_dir, app,
                           sorted(os.listdir(os.path.join(exp_dir, app)))[0])
    assert os.path.exists(os.path.join(run_dir, "result.json"))


def test_native_extension_is_loaded():
    """The HIP extension must be the in-tree .so (no silent fallback)."""
    from maggy_amd import ops

    ext = ops.require_ext()
    assert "_build/_maggy_hip.so" in ext.__file__
