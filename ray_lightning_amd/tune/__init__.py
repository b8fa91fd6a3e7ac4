"""Tune integration surface with optional-dependency degradation.

The reference guards its Tune surface behind ``TUNE_INSTALLED``
(reference tune.py:13-27: ``ray.tune`` importable -> real callbacks,
else ``Unavailable`` placeholders) and proves the degradation with a CI
job that breaks the import (reference .github/workflows/test.yaml:
197-226). Here the Tune subsystem is self-contained, so the failure
modes are a broken sub-install or an explicit opt-out
(``RLA_DISABLE_TUNE=1``, the CI-job equivalent): either way the core
training stack keeps working and the Tune names degrade to
``Unavailable``.
"""
from __future__ import annotations

import os

from ..util import Unavailable

try:
    if os.environ.get("RLA_DISABLE_TUNE") == "1":
        raise ImportError("Tune disabled via RLA_DISABLE_TUNE=1")
    from ._impl import _expand_configs  # noqa: F401 — test surface
    from ._impl import (ExperimentAnalysis, Trial, TuneCallback,
                        TuneReportCallback, TuneReportCheckpointCallback,
                        checkpoint_dir, choice, get_tune_resources,
                        grid_search, is_session_enabled, loguniform,
                        report, run, uniform)
    TUNE_INSTALLED = True
except ImportError:
    TUNE_INSTALLED = False

    # Core-path hook: the launcher asks whether a Tune session wraps the
    # current fit; without Tune the answer is simply no (reference
    # ray_launcher.py:96-99 behaves the same when tune is absent).
    def is_session_enabled() -> bool:
        return False

    TuneCallback = Unavailable
    TuneReportCallback = Unavailable
    TuneReportCheckpointCallback = Unavailable
    Trial = Unavailable
    ExperimentAnalysis = Unavailable

    def _unavailable(*_args, **_kwargs):
        raise RuntimeError(
            "ray_lightning_amd.tune is not available "
            "(RLA_DISABLE_TUNE=1 or broken sub-install); "
            "core training is unaffected.")

    get_tune_resources = _unavailable
    run = _unavailable
    report = _unavailable
    checkpoint_dir = _unavailable
    grid_search = _unavailable
    choice = _unavailable
    uniform = _unavailable
    loguniform = _unavailable
