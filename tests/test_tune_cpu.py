"""Tune integration tests — mirrors reference tests/test_tune.py:
trial iteration counts, checkpointing, resource math."""
import os

import pytest

from ray_lightning_amd import RayStrategy, Trainer
from ray_lightning_amd import tune
from ray_lightning_amd.tune import (TuneReportCallback,
                                    TuneReportCheckpointCallback,
                                    get_tune_resources)

from utils import BoringModel


def _train_fn(config, checkpoint=False):
    model = BoringModel()
    callbacks = [
        TuneReportCheckpointCallback(
            metrics={"loss": "x"}, on="validation_end") if checkpoint
        else TuneReportCallback(metrics={"loss": "x"},
                                on="validation_end")]
    trainer = Trainer(
        default_root_dir=config["root"],
        max_epochs=config["max_epochs"],
        strategy=RayStrategy(num_workers=config.get("num_workers", 1)),
        callbacks=callbacks,
        limit_train_batches=4, limit_val_batches=2,
        num_sanity_val_steps=0, enable_checkpointing=False)
    trainer.fit(model)


def _train_fn_ckpt(config):
    _train_fn(config, checkpoint=True)


def test_tune_resources_math():
    """Trial cost = 1 head CPU + N worker bundles
    (reference tune.py:32-56, README formula)."""
    pg = get_tune_resources(num_workers=3, num_cpus_per_worker=2,
                            use_gpu=False)
    assert pg.required_cpus == 1 + 3 * 2
    assert pg.required_gpus == 0
    pg = get_tune_resources(num_workers=2, use_gpu=True)
    assert pg.required_gpus == 2


def test_trial_runs_expected_iterations(tmp_path):
    """training_iteration == max_epochs (one report per validation end,
    reference test_tune.py:41-65)."""
    analysis = tune.run(
        _train_fn,
        config={"root": str(tmp_path),
                "max_epochs": tune.grid_search([1, 2])},
        local_dir=str(tmp_path / "tune"),
        metric="loss", mode="min")
    results = [t.last_result for t in analysis.trials]
    iters = sorted(r["training_iteration"] for r in results)
    assert iters == [1, 2]
    assert all("loss" in r for r in results)


def test_best_checkpoint_exists(tmp_path):
    """reference test_tune.py:68-92."""
    analysis = tune.run(
        _train_fn_ckpt,
        config={"root": str(tmp_path), "max_epochs": 2},
        local_dir=str(tmp_path / "tune"),
        metric="loss", mode="min")
    best = analysis.best_checkpoint
    assert best is not None and os.path.exists(best)
    files = os.listdir(best)
    assert "checkpoint" in files


def test_trial_with_two_distributed_workers(tmp_path):
    """Nested: a tune trial launching a 2-worker distributed fit
    (reference test_tune.py fixture shape: trial driver + 2 workers)."""
    analysis = tune.run(
        _train_fn,
        config={"root": str(tmp_path), "max_epochs": 1, "num_workers": 2},
        local_dir=str(tmp_path / "tune"),
        metric="loss", mode="min")
    assert analysis.trials[0].last_result["training_iteration"] == 1


def test_search_space_sampling(tmp_path):
    rng_space = {"lr": tune.loguniform(1e-4, 1e-1),
                 "b": tune.choice([1, 2, 3])}
    configs = tune._expand_configs(rng_space, num_samples=5, seed=0)
    assert len(configs) == 5
    for c in configs:
        assert 1e-4 <= c["lr"] <= 1e-1
        assert c["b"] in (1, 2, 3)
    grid = tune._expand_configs(
        {"x": tune.grid_search([1, 2]), "y": tune.grid_search([3, 4])},
        num_samples=1, seed=0)
    assert len(grid) == 4


def test_report_outside_session_raises():
    with pytest.raises(RuntimeError, match="outside a Tune session"):
        tune.report(loss=1.0)


def test_checkpoint_dir_outside_session_raises():
    with pytest.raises(RuntimeError, match="outside a Tune session"):
        with tune.checkpoint_dir(step=0):
            pass


def test_tune_callback_invalid_on_raises():
    with pytest.raises(ValueError, match="Invalid `on`"):
        TuneReportCallback(on="nonsense_hook")


def test_tune_gpu_overcommit_raises_fast(tmp_path, monkeypatch):
    """A per-trial GPU demand that exceeds node capacity must raise
    immediately instead of spinning with every trial pending
    (r01 deadlock: need_gpus=2 on a 1-GPU node hung forever)."""
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "0")
    with pytest.raises(RuntimeError, match="only 1.0 are visible"):
        tune.run(
            _train_fn,
            name="overcommit",
            config={"max_epochs": 1, "root": str(tmp_path)},
            num_samples=2,
            local_dir=str(tmp_path),
            resources_per_trial=get_tune_resources(
                num_workers=2, use_gpu=True))


def test_tune_gpu_zero_visible_raises(tmp_path, monkeypatch):
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "")
    with pytest.raises(RuntimeError, match="none are visible"):
        tune.run(
            _train_fn,
            name="nogpu",
            config={"max_epochs": 1, "root": str(tmp_path)},
            num_samples=1,
            local_dir=str(tmp_path),
            resources_per_trial=get_tune_resources(
                num_workers=1, use_gpu=True))


def test_tune_degrades_gracefully_without_subsystem(tmp_path):
    """Optional-dependency degradation (reference tune.py:13-27 +
    the CI job that breaks the import, test.yaml:197-226): with the
    Tune subsystem unavailable the callbacks become ``Unavailable``
    and core RayStrategy training still works."""
    import subprocess
    import sys
    script = r"""
import ray_lightning_amd.tune as tune
assert tune.TUNE_INSTALLED is False
assert tune.is_session_enabled() is False
try:
    tune.TuneReportCallback(metrics={"loss": "x"})
except RuntimeError:
    pass
else:
    raise AssertionError("Unavailable should raise on instantiation")

# core training is unaffected (exercises the launcher's
# is_session_enabled hook in degraded mode)
from utils import BoringModel
from ray_lightning_amd import RayStrategy, Trainer
trainer = Trainer(max_epochs=1, limit_train_batches=2,
                  limit_val_batches=0, num_sanity_val_steps=0,
                  enable_checkpointing=False,
                  strategy=RayStrategy(num_workers=2))
trainer.fit(BoringModel())
assert trainer.state.finished
print("DEGRADED-OK")
"""
    tests_dir = os.path.dirname(os.path.abspath(__file__))
    repo_root = os.path.dirname(tests_dir)
    env = dict(os.environ, RLA_DISABLE_TUNE="1",
               PYTHONPATH=os.pathsep.join(
                   [repo_root, tests_dir,
                    os.environ.get("PYTHONPATH", "")]))
    out = subprocess.run([sys.executable, "-c", script], env=env,
                         capture_output=True, text=True, timeout=180,
                         cwd=str(tmp_path))
    assert out.returncode == 0, out.stderr
    assert "DEGRADED-OK" in out.stdout
