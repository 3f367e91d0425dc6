"""Expert checkpointing (reference hivemind/moe/server/checkpoints.py:36-75):
a background thread saving each expert's {module, optimizer, scheduler} state
to ``checkpoint_<timestamp>.pt`` with a ``checkpoint_last.pt`` symlink;
``load_experts`` restores at server start."""

from __future__ import annotations

import threading
from datetime import datetime, timezone
from pathlib import Path
from shutil import copy2
from tempfile import TemporaryDirectory
from typing import Dict

import torch

from ...utils.logging import get_logger
from .module_backend import ModuleBackend

logger = get_logger(__name__)


def is_directory(directory: Path) -> bool:
    assert directory is not None
    assert directory.exists()
    assert directory.is_dir()
    return True


def copy_tree(src: str, dst: str):
    src_path, dst_path = Path(src), Path(dst)
    if not dst_path.exists():
        dst_path.mkdir(exist_ok=True, parents=True)
    for child in src_path.iterdir():
        if child.is_file():
            copy2(child, dst_path / child.name)
        else:
            copy_tree(str(child), str(dst_path / child.name))


class CheckpointSaver(threading.Thread):
    def __init__(self, module_backends: Dict[str, ModuleBackend], checkpoint_dir: Path, update_period: float):
        super().__init__(name="moe-checkpoints", daemon=True)
        assert is_directory(checkpoint_dir)
        self.module_backends = module_backends
        self.update_period = update_period
        self.checkpoint_dir = checkpoint_dir
        self.stop_event = threading.Event()
        store_experts(self.module_backends, self.checkpoint_dir)

    def run(self) -> None:
        while not self.stop_event.wait(self.update_period):
            try:
                store_experts(self.module_backends, self.checkpoint_dir)
            except Exception as e:
                logger.warning(f"checkpoint save failed: {e!r}")

    def shutdown(self):
        self.stop_event.set()


def store_experts(experts: Dict[str, ModuleBackend], checkpoint_dir: Path):
    logger.debug(f"storing {len(experts)} expert checkpoints to {checkpoint_dir}")
    assert is_directory(checkpoint_dir)
    timestamp = datetime.now(timezone.utc).strftime("%Y_%m_%d_%H_%M_%S")
    with TemporaryDirectory() as tmpdirname:
        for expert_name, backend in experts.items():
            expert_dir = Path(tmpdirname) / expert_name
            expert_dir.mkdir()
            checkpoint_name = expert_dir / f"checkpoint_{timestamp}.pt"
            torch.save(backend.state_dict(), checkpoint_name)
            symlink = expert_dir / "checkpoint_last.pt"
            if symlink.exists() or symlink.is_symlink():
                symlink.unlink()
            symlink.symlink_to(checkpoint_name.name)
        copy_tree(tmpdirname, str(checkpoint_dir))


def load_experts(experts: Dict[str, ModuleBackend], checkpoint_dir: Path):
    assert is_directory(checkpoint_dir)
    for expert_name, backend in experts.items():
        checkpoints_folder = checkpoint_dir / expert_name
        latest_checkpoint = checkpoints_folder / "checkpoint_last.pt"
        if latest_checkpoint.exists():
            backend.load_state_dict(torch.load(latest_checkpoint, weights_only=False))
            logger.debug(f"restored expert {expert_name} from {latest_checkpoint}")
        else:
            logger.warning(f"no checkpoint found for expert {expert_name}")
