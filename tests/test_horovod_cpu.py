"""HorovodRayStrategy CPU tests — mirrors reference tests/test_horovod.py
(the strategy keeps the reference's ctor surface but routes allreduce
through the same RCCL/gloo engine)."""
import torch

from ray_lightning_amd import HorovodRayStrategy

from utils import BoringModel, LightningMNISTClassifier, get_trainer, \
    load_test, predict_test, train_test
from ray_lightning_amd import seed_everything


def test_ctor_surface():
    s = HorovodRayStrategy(num_workers=2, num_cpus_per_worker=1,
                           use_gpu=False)
    assert s.num_workers == 2
    assert s.cpus_per_worker == 1
    assert s.executor is None  # reference API surface (ray_horovod.py:86)
    assert s.strategy_name == "horovod_ray"


def test_train_two_workers(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=HorovodRayStrategy(num_workers=2))
    train_test(trainer, model)


def test_load(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path),
                          strategy=HorovodRayStrategy(num_workers=2))
    load_test(trainer, model)


def test_predict(tmp_path):
    seed_everything(44)
    model = LightningMNISTClassifier()
    trainer = get_trainer(str(tmp_path), max_epochs=2,
                          strategy=HorovodRayStrategy(num_workers=2))
    predict_test(trainer, model)
