"""Checkpoint/resume (SURVEY.md §5 "Checkpoint / resume"): byte-compatible
with the reference contract — per-expert `torch.save` maps
{'model': state_dict, 'optimizer': state_dict, ...} written to
checkpoint_dir/<expert_uid>/checkpoint_<timestamp>.pt plus checkpoint_last.pt;
`load_experts` restores the latest on server start."""
from __future__ import annotations

import os
import shutil
import threading
import time
from pathlib import Path
from typing import Dict

import torch

from ..runtime import ExpertBackend
from ..utils import get_logger

logger = get_logger(__name__)


def store_experts(experts: Dict[str, ExpertBackend], checkpoint_dir: os.PathLike) -> None:
    checkpoint_dir = Path(checkpoint_dir)
    timestamp = time.strftime("%Y%m%d_%H%M%S")
    for uid, backend in experts.items():
        expert_dir = checkpoint_dir / uid
        expert_dir.mkdir(parents=True, exist_ok=True)
        tmp_path = expert_dir / f".tmp_checkpoint_{timestamp}.pt"
        final_path = expert_dir / f"checkpoint_{timestamp}.pt"
        torch.save(backend.state_dict(), tmp_path)
        os.replace(tmp_path, final_path)  # atomic publish
        shutil.copyfile(final_path, expert_dir / "checkpoint_last.pt")


def load_experts(experts: Dict[str, ExpertBackend], checkpoint_dir: os.PathLike) -> int:
    """Restore each expert from its latest checkpoint; returns #restored."""
    checkpoint_dir = Path(checkpoint_dir)
    restored = 0
    for uid, backend in experts.items():
        latest = checkpoint_dir / uid / "checkpoint_last.pt"
        if latest.exists():
            state = torch.load(latest, map_location="cpu", weights_only=False)
            backend.load_state_dict(state)
            restored += 1
    return restored


class CheckpointSaver(threading.Thread):
    """Background thread: every `update_period` seconds, snapshot all experts."""

    def __init__(self, expert_backends: Dict[str, ExpertBackend], checkpoint_dir: os.PathLike,
                 update_period: float = 300.0):
        super().__init__(daemon=True, name="CheckpointSaver")
        self.expert_backends = expert_backends
        self.checkpoint_dir = Path(checkpoint_dir)
        self.update_period = update_period
        self._stop = threading.Event()

    def run(self) -> None:
        while not self._stop.wait(self.update_period):
            try:
                store_experts(self.expert_backends, self.checkpoint_dir)
            except Exception as e:  # noqa: BLE001
                logger.warning(f"checkpoint save failed: {e!r}")

    def stop(self) -> None:
        self._stop.set()
