"""WandB logging shim with a tensorboard-like API
(reference megatron/wandb_logger.py:12-162)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class WandBConfig:
    project: Optional[str] = None
    entity: Optional[str] = None
    name: Optional[str] = None
    id: Optional[str] = None
    api_key: Optional[str] = None

    @classmethod
    def from_config(cls, cfg):
        return cls(
            project=cfg.wandb_project, entity=cfg.wandb_entity,
            name=cfg.wandb_name, id=cfg.wandb_id, api_key=cfg.wandb_api_key,
        )


class WandbTBShim:
    """add_scalar/add_text/flush_all accumulating per-step dicts."""

    def __init__(self, config: WandBConfig):
        import wandb

        self._wandb = wandb
        kwargs = {}
        if config.project:
            kwargs["project"] = config.project
        if config.entity:
            kwargs["entity"] = config.entity
        if config.name:
            kwargs["name"] = config.name
        if config.id:
            kwargs["id"] = config.id
            kwargs["resume"] = "allow"
        if config.api_key:
            wandb.login(key=config.api_key)
        self._run = wandb.init(**kwargs)
        self._pending = {}
        self._pending_step = None

    def add_scalar(self, name, value, step):
        if self._pending_step is not None and step != self._pending_step:
            self.flush_all()
        self._pending[name] = value
        self._pending_step = step

    def add_text(self, name, value, step=None):
        self._wandb.log({name: self._wandb.Html(value)})

    def flush_all(self):
        if self._pending:
            self._wandb.log(self._pending, step=self._pending_step)
            self._pending = {}
            self._pending_step = None
