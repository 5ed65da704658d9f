"""The examples/ scripts stay runnable (they double as living docs)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, tmp_path, timeout=240):
    env = dict(os.environ, MAGGY_LOG_DIR=str(tmp_path))
    p = subprocess.run([sys.executable] + args, cwd=REPO, env=env,
                       capture_output=True, text=True, timeout=timeout)
    assert p.returncode == 0, p.stdout[-2000:] + p.stderr[-2000:]
    return p.stdout


@pytest.mark.timeout(300)
def test_example_random_search(tmp_path):
    out = _run(["examples/hpo_random_search.py", "--trials", "4"], tmp_path)
    assert "best config:" in out


@pytest.mark.timeout(300)
def test_example_asha(tmp_path):
    out = _run(["examples/hpo_asha_hyperband.py", "--mode", "asha"],
               tmp_path)
    assert "best:" in out


@pytest.mark.timeout(300)
def test_example_hyperband(tmp_path):
    out = _run(["examples/hpo_asha_hyperband.py", "--mode", "hyperband"],
               tmp_path)
    assert "best:" in out


@pytest.mark.timeout(300)
def test_example_ablation(tmp_path):
    out = _run(["examples/ablation_transformer.py"], tmp_path)
    assert "trials: 5" in out


@pytest.mark.timeout(300)
def test_example_dist(tmp_path):
    out = _run(["examples/dist_llama_ddp.py", "--workers", "2"], tmp_path)
    assert "world: 2" in out
