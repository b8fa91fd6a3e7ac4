"""RayStrategy on GPU: actor-launched training with device binding
(reference tests/test_ddp_gpu.py — train on GPU :38-63, model-on-CUDA
assert :66-79, fractional GPUs with gloo :84-123, downsized to the
1-GPU CI box)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from ray_lightning_amd import RayStrategy, Trainer  # noqa: E402

from utils import BoringModel, get_trainer, train_test  # noqa: E402


def test_train_one_gpu_worker(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayStrategy(num_workers=1,
                                               use_gpu=True))
    train_test(trainer, model)


class _DeviceAssertModel(BoringModel):
    def training_step(self, batch, batch_idx):
        # runs inside the GPU actor: model must be bound to a CUDA
        # device (reference test_ddp_gpu.py:66-79)
        assert self.layer.weight.is_cuda
        assert batch.is_cuda
        return super().training_step(batch, batch_idx)


def test_model_on_cuda_inside_worker(tmp_path):
    model = _DeviceAssertModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayStrategy(num_workers=1,
                                               use_gpu=True))
    trainer.fit(model)
    assert trainer.state.finished


def test_fractional_gpus_share_device(tmp_path):
    """Two workers at 0.5 GPU each share one device; gradient comm
    falls back to the gloo data plane (reference test_ddp_gpu.py:84-123
    semantics)."""
    import warnings
    model = BoringModel()
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")  # fractional-GPU warning
        strategy = RayStrategy(num_workers=2, use_gpu=True,
                               resources_per_worker={"GPU": 0.5})
    trainer = get_trainer(str(tmp_path), strategy=strategy,
                          limit_train_batches=4, limit_val_batches=2)
    train_test(trainer, model)


def test_metrics_transported_from_gpu_worker(tmp_path):
    from utils import XORDataModule, XORModel
    model = XORModel()
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=1,
                      strategy=RayStrategy(num_workers=1, use_gpu=True),
                      enable_checkpointing=False, num_sanity_val_steps=0)
    trainer.fit(model, datamodule=XORDataModule())
    assert float(trainer.callback_metrics["avg_val_loss"]) == \
        pytest.approx(0.3)


def test_sharded_strategy_fractional_gpu(tmp_path):
    """RayShardedStrategy with two workers sharing one device: OSS
    partition + reduce-to-owner + param broadcast on real HIP tensors
    (gloo data plane)."""
    import warnings
    from ray_lightning_amd import RayShardedStrategy
    model = BoringModel()
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        strategy = RayShardedStrategy(num_workers=2, use_gpu=True,
                                      resources_per_worker={"GPU": 0.5})
    trainer = get_trainer(str(tmp_path), strategy=strategy,
                          limit_train_batches=4, limit_val_batches=2)
    train_test(trainer, model)


def test_tune_trial_on_gpu(tmp_path):
    """One Tune trial running a GPU worker end-to-end (reference
    test_tune.py:95-116 downsized to 1 GPU)."""
    from ray_lightning_amd import tune
    from ray_lightning_amd.tune import (TuneReportCallback,
                                        get_tune_resources)

    def train_fn(config):
        model = BoringModel()
        trainer = Trainer(
            default_root_dir=config["root"], max_epochs=1,
            strategy=RayStrategy(num_workers=1, use_gpu=True),
            callbacks=[TuneReportCallback({"loss": "x"},
                                          on="validation_end")],
            limit_train_batches=4, limit_val_batches=2,
            num_sanity_val_steps=0, enable_checkpointing=False)
        trainer.fit(model)

    analysis = tune.run(
        train_fn, config={"root": str(tmp_path)},
        resources_per_trial=get_tune_resources(num_workers=1,
                                               use_gpu=True),
        local_dir=str(tmp_path / "tune"), metric="loss", mode="min")
    assert analysis.trials[0].last_result["training_iteration"] == 1


def test_tune_two_concurrent_fractional_gpu_trials(tmp_path):
    """TWO Tune trials concurrently, each a 2-worker RayStrategy at
    GPU=0.25/worker, all sharing one MI355X (BASELINE config 5's
    trial-concurrency shape downsized to one device; gloo data plane
    for the fractional workers)."""
    import time

    from ray_lightning_amd import tune
    from ray_lightning_amd.tune import (TuneReportCallback,
                                        get_tune_resources)

    def train_fn(config):
        model = BoringModel()
        trainer = Trainer(
            default_root_dir=config["root"], max_epochs=1,
            strategy=RayStrategy(
                num_workers=2,
                resources_per_worker={"GPU": 0.25}),
            callbacks=[TuneReportCallback({"loss": "x"},
                                          on="validation_end")],
            limit_train_batches=4, limit_val_batches=2,
            num_sanity_val_steps=0, enable_checkpointing=False)
        trainer.fit(model)

    t0 = time.time()
    analysis = tune.run(
        train_fn,
        config={"root": str(tmp_path),
                "trial": tune.grid_search([0, 1])},
        resources_per_trial=get_tune_resources(
            num_workers=2, use_gpu=True,
            resources_per_worker={"GPU": 0.25}),
        local_dir=str(tmp_path / "tune"), metric="loss", mode="min")
    assert len(analysis.trials) == 2
    for t in analysis.trials:
        assert t.last_result["training_iteration"] == 1
    assert time.time() - t0 < 300


def test_horovod_strategy_on_gpu(tmp_path):
    """HorovodRayStrategy trains on a GPU worker (the third strategy's
    GPU tier; allreduce routes through the same engine)."""
    from ray_lightning_amd.strategies.ray_horovod import \
        HorovodRayStrategy

    model = BoringModel()
    trainer = Trainer(
        default_root_dir=str(tmp_path), max_epochs=1,
        strategy=HorovodRayStrategy(num_workers=1, use_gpu=True),
        limit_train_batches=4, limit_val_batches=2,
        num_sanity_val_steps=0, enable_checkpointing=False)
    trainer.fit(model)
    assert trainer.state.finished
