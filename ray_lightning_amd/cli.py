"""LightningCLI-style command line interface.

Parity target: the reference is driven through jsonargparse's
``LightningCLI`` and pins strategy-kwarg resolution from the command
line (reference tests/test_lightning_cli.py:11-27:
``--trainer.strategy=RayStrategy --trainer.strategy.num_workers=2
--trainer.strategy.bucket_cap_mb=50``). This implements that dotted
``--trainer.* / --model.* / --data.*`` surface over this framework's
Trainer/strategies, including typed strategy constructor args and
DDP-kwargs pass-through.
"""
from __future__ import annotations

import ast
import sys
from typing import Any, Dict, List, Optional, Type

from .strategies import STRATEGY_REGISTRY
from .trainer import LightningDataModule, LightningModule, Trainer


def _coerce(value: str) -> Any:
    """Best-effort typed parse: int/float/bool/None/list/dict, else str."""
    try:
        return ast.literal_eval(value)
    except (ValueError, SyntaxError):
        low = value.lower()
        if low in ("true", "yes"):
            return True
        if low in ("false", "no"):
            return False
        if low in ("none", "null"):
            return None
        return value


def _parse_dotted(args: List[str]) -> Dict[str, Dict[str, Any]]:
    """``--a.b=v`` / ``--a.b v`` -> {"a": {"b": v}} (nested keys keep
    their remaining dots: ``--trainer.strategy.num_workers`` ->
    trainer["strategy.num_workers"])."""
    out: Dict[str, Dict[str, Any]] = {}
    i = 0
    while i < len(args):
        arg = args[i]
        if not arg.startswith("--"):
            raise SystemExit(f"unrecognized argument: {arg}")
        body = arg[2:]
        if "=" in body:
            key, value = body.split("=", 1)
        else:
            key = body
            i += 1
            if i >= len(args):
                raise SystemExit(f"missing value for --{key}")
            value = args[i]
        if "." not in key:
            raise SystemExit(
                f"expected dotted key (--trainer.x / --model.x): --{key}")
        section, rest = key.split(".", 1)
        out.setdefault(section, {})[rest] = _coerce(value)
        i += 1
    return out


class LightningCLI:
    """Build a Trainer + model (+ datamodule) from dotted CLI args and
    optionally run a subcommand (``fit``/``validate``/``test``/
    ``predict``)."""

    def __init__(self,
                 model_class: Type[LightningModule],
                 datamodule_class: Optional[
                     Type[LightningDataModule]] = None,
                 args: Optional[List[str]] = None,
                 trainer_defaults: Optional[Dict[str, Any]] = None,
                 run: bool = True):
        argv = list(sys.argv[1:] if args is None else args)
        self.subcommand: Optional[str] = None
        if argv and not argv[0].startswith("-"):
            self.subcommand = argv.pop(0)

        parsed = _parse_dotted(argv)
        trainer_cfg = dict(trainer_defaults or {})
        trainer_cfg.update({k: v for k, v in
                            parsed.get("trainer", {}).items()
                            if not k.startswith("strategy")})
        strategy_cfg = {k: v for k, v in parsed.get("trainer", {}).items()
                        if k.startswith("strategy")}
        model_cfg = parsed.get("model", {})
        data_cfg = parsed.get("data", {})

        strategy = self._build_strategy(strategy_cfg)
        if strategy is not None:
            trainer_cfg["strategy"] = strategy

        self.model = model_class(**model_cfg)
        self.datamodule = (datamodule_class(**data_cfg)
                           if datamodule_class is not None else None)
        self.trainer = Trainer(**trainer_cfg)

        if run and self.subcommand:
            fn = getattr(self.trainer, self.subcommand)
            fn(self.model, datamodule=self.datamodule)

    @staticmethod
    def _build_strategy(cfg: Dict[str, Any]):
        """cfg keys: "strategy" = class name or registry nickname,
        "strategy.<kwarg>" = constructor args (unknown kwargs flow into
        ``**ddp_kwargs``, reference test_lightning_cli.py:15)."""
        name = cfg.pop("strategy", None)
        kwargs = {k.split(".", 1)[1]: v for k, v in cfg.items()}
        if name is None:
            return None
        if isinstance(name, str):
            from . import strategies as _s
            cls = getattr(_s, name, None) or STRATEGY_REGISTRY.get(name)
            if cls is None:
                raise SystemExit(f"unknown strategy: {name}")
            return cls(**kwargs)
        return name
