"""Experimental "ModelFlow" AutoML pipeline (reference: adanet/experimental/)."""

from adanet_amd.experimental.controllers import Controller
from adanet_amd.experimental.controllers import SequentialController
from adanet_amd.experimental.ensemble_model import (EnsembleModel,
                                                    MeanEnsemble,
                                                    WeightedEnsemble)
from adanet_amd.experimental.keras_model import CompiledModel
from adanet_amd.experimental.model_search import ModelSearch
from adanet_amd.experimental.phases import (AllStrategy, AutoEnsemblePhase,
                                            GrowStrategy, InputPhase, Phase,
                                            RandomKStrategy, RepeatPhase,
                                            TrainerPhase, TunerPhase)
from adanet_amd.experimental.schedulers import (InProcessScheduler,
                                                MultiGpuScheduler,
                                                Scheduler,
                                                ThreadedScheduler)
from adanet_amd.experimental.storages import (InMemoryStorage,
                                              ModelContainer, Storage)
from adanet_amd.experimental.work_units import TrainerWorkUnit, WorkUnit

__all__ = [
    "Controller", "SequentialController", "EnsembleModel", "MeanEnsemble",
    "WeightedEnsemble", "CompiledModel", "ModelSearch", "AllStrategy",
    "AutoEnsemblePhase", "GrowStrategy", "InputPhase", "Phase",
    "RandomKStrategy", "RepeatPhase", "TrainerPhase", "TunerPhase",
    "InProcessScheduler", "MultiGpuScheduler", "Scheduler", "ThreadedScheduler", "InMemoryStorage", "ModelContainer",
    "Storage", "TrainerWorkUnit", "WorkUnit",
]
