"""Phases: composable stages of a ModelFlow pipeline.

Reference: adanet/experimental/phases/ — Phase ABC (phase.py),
InputPhase (input_phase.py:25), KerasTrainerPhase
(keras_trainer_phase.py:28), KerasTunerPhase (keras_tuner_phase.py:29 —
here a built-in random-search TunerPhase since keras-tuner doesn't exist
on this stack), AutoEnsemblePhase with its own GrowStrategy/AllStrategy/
RandomKStrategy (autoensemble_phase.py:54-135), RepeatPhase
(repeat_phase.py:26).
"""

from __future__ import annotations

import abc
import random
from typing import Callable, Iterator, List, Optional, Sequence

from adanet_amd.experimental.ensemble_model import MeanEnsemble
from adanet_amd.experimental.keras_model import CompiledModel
from adanet_amd.experimental.storages import (InMemoryStorage,
                                              ModelContainer, Storage)
from adanet_amd.experimental.work_units import TrainerWorkUnit, WorkUnit


class Phase(abc.ABC):
    """Reference phases/phase.py: a phase consumes its predecessor and
    yields WorkUnits; DatasetProvider/ModelProvider mixins collapse to the
    get_train/eval_dataset + get_models accessors."""

    #: a phase's work units are mutually independent (ThreadedScheduler may
    #: run them concurrently); RepeatPhase overrides.
    parallel_ok = True

    def __init__(self):
        self._previous: Optional[Phase] = None

    def set_previous(self, previous: "Phase"):
        self._previous = previous
        return self

    @abc.abstractmethod
    def work_units(self) -> Iterator[WorkUnit]:
        ...

    # dataset provider chain
    def get_train_dataset(self):
        return self._previous.get_train_dataset() if self._previous else None

    def get_eval_dataset(self):
        return self._previous.get_eval_dataset() if self._previous else None

    # model provider chain
    def get_models(self) -> List:
        return self._previous.get_models() if self._previous else []

    def get_best_models(self, num_models: int = 1) -> List:
        return self._previous.get_best_models(num_models) if (
            self._previous) else []


class InputPhase(Phase):
    """Provides the datasets (reference input_phase.py:25)."""

    def __init__(self, train_dataset, eval_dataset):
        super().__init__()
        self._train = train_dataset
        self._eval = eval_dataset

    def work_units(self):
        return iter(())

    def get_train_dataset(self):
        return self._train

    def get_eval_dataset(self):
        return self._eval


class TrainerPhase(Phase):
    """Trains a fixed list of models (reference keras_trainer_phase.py:28).

    ``devices``: optional device list round-robined over the models (e.g.
    ["cuda:0", ..., "cuda:7"]) so a ThreadedScheduler trains candidates on
    the node's GPUs concurrently."""

    def __init__(self, models: Sequence, epochs: int = 1,
                 steps_per_epoch: Optional[int] = None,
                 eval_steps: Optional[int] = None,
                 storage: Optional[Storage] = None,
                 devices: Optional[Sequence] = None):
        super().__init__()
        self._models = list(models)
        self._epochs = epochs
        self._steps_per_epoch = steps_per_epoch
        self._eval_steps = eval_steps
        self._storage = storage or InMemoryStorage()
        self._devices = list(devices) if devices else None

    def work_units(self):
        import torch
        for i, model in enumerate(self._models):
            if not isinstance(model, CompiledModel):
                device = (torch.device(self._devices[i % len(self._devices)])
                          if self._devices else None)
                model = CompiledModel(model, device=device)
            yield TrainerWorkUnit(model, self.get_train_dataset(),
                                  self.get_eval_dataset(), self._storage,
                                  epochs=self._epochs,
                                  steps_per_epoch=self._steps_per_epoch,
                                  eval_steps=self._eval_steps)

    def get_models(self):
        return self._storage.get_models()

    def get_best_models(self, num_models: int = 1):
        return self._storage.get_best_models(num_models)


class TunerPhase(TrainerPhase):
    """Random-search tuner: samples `num_trials` models from a builder
    callable over an hparam space (stand-in for the reference's
    keras-tuner wrapper, keras_tuner_phase.py:29)."""

    def __init__(self, build_fn: Callable[[dict], object],
                 hparam_space: dict, num_trials: int = 3, seed: int = 0,
                 **kwargs):
        rng = random.Random(seed)
        models = []
        for _ in range(num_trials):
            hp = {k: rng.choice(list(v)) for k, v in hparam_space.items()}
            models.append(build_fn(hp))
        super().__init__(models, **kwargs)


# --- ensemble strategies local to ModelFlow (reference
# autoensemble_phase.py:73-113) ---
class EnsembleStrategy(abc.ABC):

    @abc.abstractmethod
    def __call__(self, candidates: List) -> Iterator[List]:
        ...


class GrowStrategy(EnsembleStrategy):
    """Only grows the previous best ensemble (autoensemble_phase.py:73)."""

    def __call__(self, candidates):
        return iter([[c] for c in candidates])


class AllStrategy(EnsembleStrategy):

    def __call__(self, candidates):
        return iter([candidates])


class RandomKStrategy(EnsembleStrategy):
    """k random candidates (reference autoensemble_phase.py:89-113)."""

    def __init__(self, k: int, seed: Optional[int] = None):
        self._k = k
        self._seed = seed

    def __call__(self, candidates):
        rng = random.Random(self._seed)
        k = min(self._k, len(candidates))
        return iter([rng.sample(list(candidates), k)])


class AutoEnsemblePhase(Phase):
    """Ensembles the best models so far (reference
    autoensemble_phase.py:114-190): for each candidate group from each
    strategy, build ensemble(previous_best + group), evaluate, store."""

    def __init__(self, ensemblers: Sequence[Callable],
                 ensemble_strategies: Sequence[EnsembleStrategy],
                 num_candidates: int = 3,
                 storage: Optional[Storage] = None,
                 eval_steps: Optional[int] = None):
        super().__init__()
        self._ensemblers = list(ensemblers)
        self._strategies = list(ensemble_strategies)
        self._num_candidates = num_candidates
        self._storage = storage or InMemoryStorage()
        self._eval_steps = eval_steps

    def work_units(self):
        candidates = self.get_best_models(self._num_candidates)
        previous_best = self._storage.get_best_models(1)
        for strategy in self._strategies:
            for group in strategy(candidates):
                for ensembler in self._ensemblers:
                    members = list(group)
                    if previous_best:
                        prev = previous_best[0]
                        prev_members = getattr(prev, "submodels", [prev])
                        members = list(prev_members) + members
                    ensemble = ensembler(members)
                    model = CompiledModel(ensemble)
                    yield TrainerWorkUnit(
                        model, self.get_train_dataset(),
                        self.get_eval_dataset(), self._storage, epochs=1,
                        steps_per_epoch=0, eval_steps=self._eval_steps)

    def get_models(self):
        return self._storage.get_models()

    def get_best_models(self, num_models: int = 1):
        # Prefer own storage once populated; else upstream.
        own = self._storage.get_best_models(num_models)
        if own:
            return own
        return super().get_best_models(num_models)


class RepeatPhase(Phase):
    """Repeats an inner phase pipeline n times (reference
    repeat_phase.py:26): phase factories are re-invoked per repetition."""

    parallel_ok = False  # repetitions chain on each other

    def __init__(self, phase_factories: Sequence[Callable[[], Phase]],
                 repetitions: int):
        super().__init__()
        self._factories = list(phase_factories)
        self._repetitions = repetitions
        self._last: Optional[Phase] = None

    def work_units(self):
        for _ in range(self._repetitions):
            prev = self._last or self._previous
            for factory in self._factories:
                phase = factory()
                phase.set_previous(prev)
                for wu in phase.work_units():
                    yield wu
                prev = phase
            self._last = prev

    def get_models(self):
        return self._last.get_models() if self._last else super().get_models()

    def get_best_models(self, num_models: int = 1):
        if self._last:
            return self._last.get_best_models(num_models)
        return super().get_best_models(num_models)
