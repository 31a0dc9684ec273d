"""_Architecture: serializable record of an ensemble's composition.

JSON format is byte-compatible with the reference's architecture-<t>.json
(reference adanet/core/architecture.py:24-173, serialize :132-153):
{"ensemble_candidate_name", "iteration_number", "global_step",
 "ensembler_name", "subnetworks": [{"iteration_number", "builder_name"}...],
 "replay_indices"} — sort_keys=True.
"""

from __future__ import annotations

import copy
import json
from typing import List, Optional, Tuple


class _Architecture(object):

    def __init__(self, ensemble_candidate_name: Optional[str],
                 ensembler_name: Optional[str], global_step: int = 0,
                 replay_indices: Optional[List[int]] = None):
        self.ensemble_candidate_name = ensemble_candidate_name
        self.ensembler_name = ensembler_name
        self.global_step = global_step
        self._subnets: List[Tuple[int, str]] = []
        self._replay_indices: List[int] = list(replay_indices or [])

    @property
    def subnetworks(self) -> Tuple[Tuple[int, str], ...]:
        return tuple(self._subnets)

    @property
    def replay_indices(self) -> List[int]:
        return self._replay_indices

    @property
    def subnetworks_grouped_by_iteration(self):
        grouped = {}
        for it, name in self._subnets:
            grouped.setdefault(it, []).append(name)
        return tuple((i, tuple(grouped[i])) for i in sorted(grouped))

    def add_subnetwork(self, iteration_number: int, builder_name: str):
        self._subnets.append((iteration_number, builder_name))

    def add_replay_index(self, index: int):
        self._replay_indices.append(index)

    def set_replay_indices(self, indices):
        self._replay_indices = copy.copy(list(indices))

    def serialize(self, iteration_number: int, global_step: int) -> str:
        assert global_step is not None
        arch = {
            "ensemble_candidate_name": self.ensemble_candidate_name,
            "iteration_number": int(iteration_number),
            "global_step": int(global_step),
            "ensembler_name": self.ensembler_name,
            "subnetworks": [{
                "iteration_number": int(it),
                "builder_name": name,
            } for it, name in self._subnets],
            "replay_indices": self._replay_indices,
        }
        return json.dumps(arch, sort_keys=True)

    @staticmethod
    def deserialize(serialized: str) -> "_Architecture":
        arch = json.loads(serialized)
        out = _Architecture(arch["ensemble_candidate_name"],
                            arch["ensembler_name"], arch["global_step"],
                            arch["replay_indices"])
        for sub in arch["subnetworks"]:
            out.add_subnetwork(sub["iteration_number"], sub["builder_name"])
        return out
