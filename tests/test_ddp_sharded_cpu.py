"""RayShardedStrategy integration tests on CPU — mirrors reference
tests/test_ddp_sharded.py coverage."""
import pytest
import torch

from ray_lightning_amd import RayShardedStrategy, Trainer

from utils import BoringModel, get_trainer


def test_strategy_recognized():
    s = RayShardedStrategy(num_workers=2)
    assert s.strategy_name == "ddp_sharded_ray"
    assert s.nickname == "ddp_sharded_ray"
    assert s.num_workers == 2


def test_train_two_workers(tmp_path):
    model = BoringModel()
    before = torch.cat([p.flatten() for p in model.parameters()]).clone()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    trainer.fit(model)
    after = torch.cat([p.flatten() for p in model.parameters()])
    assert torch.norm(before - after) > 0.1


def test_checkpoint_param_equality(tmp_path):
    """Saved checkpoint params match the in-memory model after fit
    (reference test_ddp_sharded.py:46-63)."""
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    trainer.fit(model)
    ckpt = torch.load(trainer.checkpoint_callback.best_model_path,
                      map_location="cpu", weights_only=False)
    for name, p in model.state_dict().items():
        assert torch.allclose(p.cpu(), ckpt["state_dict"][name],
                              atol=1e-6), name


def test_finetune_restart(tmp_path):
    """Train, load checkpoint into a fresh model, train again
    (reference test_ddp_sharded.py:66-80)."""
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    trainer.fit(model)
    ckpt_path = trainer.checkpoint_callback.best_model_path

    model2 = BoringModel.load_from_checkpoint(ckpt_path)
    trainer2 = get_trainer(str(tmp_path),
                           strategy=RayShardedStrategy(num_workers=2))
    trainer2.fit(model2)
    assert trainer2.state.finished


def test_resume_from_checkpoint(tmp_path):
    """reference test_ddp_sharded.py:83-104."""
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    trainer.fit(model)
    ckpt_path = trainer.checkpoint_callback.best_model_path

    model2 = BoringModel()
    trainer2 = get_trainer(str(tmp_path), max_epochs=2,
                           strategy=RayShardedStrategy(num_workers=2),
                           resume_from_checkpoint=ckpt_path)
    trainer2.fit(model2)
    assert trainer2.state.finished
    assert trainer2.current_epoch == 2


def test_test_without_fit(tmp_path):
    """reference test_ddp_sharded.py:107-115."""
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    out = trainer.test(model)
    assert "y" in out[0]


def test_resume_with_fewer_workers(tmp_path):
    """2-worker checkpoint resumes on 1 worker
    (reference test_ddp_sharded.py:118-137)."""
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=RayShardedStrategy(num_workers=2))
    trainer.fit(model)
    ckpt_path = trainer.checkpoint_callback.best_model_path

    model2 = BoringModel()
    trainer2 = get_trainer(str(tmp_path), max_epochs=2,
                           strategy=RayShardedStrategy(num_workers=1),
                           resume_from_checkpoint=ckpt_path)
    trainer2.fit(model2)
    assert trainer2.state.finished
