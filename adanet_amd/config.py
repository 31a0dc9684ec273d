"""RunConfig: runtime settings (device, seeds, checkpoints, cluster).

The reference rides tf.estimator.RunConfig + the TF_CONFIG env JSON for
cluster topology. MI355X-native: a plain dataclass;
cluster topology comes from the torchrun-style env (RANK / WORLD_SIZE /
LOCAL_RANK / MASTER_ADDR / MASTER_PORT), one process per GPU over RCCL.
"""

from __future__ import annotations

import dataclasses
import os
from typing import Optional

import torch


@dataclasses.dataclass
class RunConfig:
    model_dir: Optional[str] = None
    tf_random_seed: Optional[int] = None  # name kept for API parity
    save_checkpoints_steps: Optional[int] = None
    keep_checkpoint_max: int = 5
    log_step_count_steps: int = 100
    device: Optional[str] = None  # "cuda:<local_rank>" | "cpu" | None=auto

    @property
    def random_seed(self):
        return self.tf_random_seed

    @property
    def world_size(self) -> int:
        return int(os.environ.get("WORLD_SIZE", "1"))

    @property
    def rank(self) -> int:
        return int(os.environ.get("RANK", "0"))

    @property
    def local_rank(self) -> int:
        return int(os.environ.get("LOCAL_RANK", str(self.rank)))

    @property
    def is_chief(self) -> bool:
        return self.rank == 0

    @property
    def num_worker_replicas(self) -> int:
        return self.world_size

    def resolve_device(self) -> torch.device:
        if self.device is not None:
            return torch.device(self.device)
        if torch.cuda.is_available():
            return torch.device("cuda", self.local_rank % max(
                1, torch.cuda.device_count()))
        return torch.device("cpu")
