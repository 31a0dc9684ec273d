"""Work units: the scheduler's unit of execution.

Reference: adanet/experimental/work_units/{work_unit.py,
keras_trainer_work_unit.py:41-55, keras_tuner_work_unit.py}.
"""

from __future__ import annotations

import abc

from adanet_amd.experimental.storages import ModelContainer, Storage


class WorkUnit(abc.ABC):

    @abc.abstractmethod
    def execute(self):
        ...


class TrainerWorkUnit(WorkUnit):
    """Fits a CompiledModel and pushes (eval loss, model) into storage
    (reference keras_trainer_work_unit.py:41-55)."""

    def __init__(self, model, train_dataset, eval_dataset, storage: Storage,
                 epochs: int = 1, steps_per_epoch=None, eval_steps=None):
        self._model = model
        self._train = train_dataset
        self._eval = eval_dataset
        self._storage = storage
        self._epochs = epochs
        self._steps_per_epoch = steps_per_epoch
        self._eval_steps = eval_steps
        #: routing hint for MultiGpuScheduler (CompiledModel.device)
        self.device = getattr(model, "device", None)

    def execute(self):
        self._model.fit(self._train, epochs=self._epochs,
                        steps_per_epoch=self._steps_per_epoch)
        metrics = self._model.evaluate(self._eval, steps=self._eval_steps)
        self._storage.save_model(
            ModelContainer(metrics["loss"], self._model, metrics))
