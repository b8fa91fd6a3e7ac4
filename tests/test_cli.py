"""CLI arg resolution (reference tests/test_lightning_cli.py:11-27)."""
import pytest

from ray_lightning_amd import RayShardedStrategy, RayStrategy
from ray_lightning_amd.cli import LightningCLI, _coerce

from utils import BoringModel, XORDataModule, XORModel


def test_strategy_args_resolved():
    cli = LightningCLI(
        BoringModel,
        args=["--trainer.strategy=RayStrategy",
              "--trainer.strategy.num_workers=2",
              "--trainer.strategy.use_gpu=false",
              "--trainer.strategy.bucket_cap_mb=50",
              "--trainer.max_epochs=1"],
        run=False)
    s = cli.trainer.strategy
    assert isinstance(s, RayStrategy)
    assert s.num_workers == 2
    assert s.use_gpu is False
    # unknown ctor kwargs flow into **ddp_kwargs
    assert s._ddp_kwargs["bucket_cap_mb"] == 50
    assert cli.trainer.max_epochs == 1


def test_registry_nickname():
    cli = LightningCLI(
        BoringModel,
        args=["--trainer.strategy=ddp_sharded_ray",
              "--trainer.strategy.num_workers=1"],
        run=False)
    assert isinstance(cli.trainer.strategy, RayShardedStrategy)


def test_cli_fit_runs(tmp_path):
    cli = LightningCLI(
        XORModel, XORDataModule,
        args=["fit",
              "--trainer.strategy=RayStrategy",
              "--trainer.strategy.num_workers=2",
              "--trainer.max_epochs=1",
              "--trainer.enable_checkpointing=false",
              "--trainer.num_sanity_val_steps=0",
              f"--trainer.default_root_dir={tmp_path}"],
        run=True)
    assert cli.trainer.state.finished
    assert float(cli.trainer.callback_metrics["avg_val_loss"]) == \
        pytest.approx(0.3)


def test_coerce_types():
    assert _coerce("2") == 2
    assert _coerce("0.5") == 0.5
    assert _coerce("true") is True
    assert _coerce("None") is None
    assert _coerce("[1, 2]") == [1, 2]
    assert _coerce("hello") == "hello"
