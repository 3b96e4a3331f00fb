"""Training session driver wiring evaluation + checkpointing callbacks into
the trainer (capability parity with reference src/modalities/gym.py:35-121)."""

from typing import Optional

from modalities_amd.loss_functions import Loss
from modalities_amd.training.evaluator import Evaluator
from modalities_amd.training.progress import TrainingProgress
from modalities_amd.training.trainer import Trainer


class Gym:
    def __init__(self, trainer: Trainer, evaluator: Evaluator, loss_fun: Loss,
                 num_ranks: int = 1):
        self.trainer = trainer
        self.evaluator = evaluator
        self.loss_fun = loss_fun
        self.num_ranks = num_ranks

    def run(self, model, optimizer, scheduler, train_data_loader,
            evaluation_data_loaders: Optional[list] = None,
            checkpoint_saving=None, app_state=None):
        evaluation_data_loaders = evaluation_data_loaders or []

        def evaluation_callback(num_train_steps_done: int) -> None:
            if evaluation_data_loaders:
                self.evaluator.evaluate(model, evaluation_data_loaders, self.loss_fun,
                                        num_train_steps_done)

        def checkpointing_callback(training_progress: TrainingProgress) -> None:
            if checkpoint_saving is not None and app_state is not None:
                checkpoint_saving.save_checkpoint_and_free_memory(
                    training_progress, app_state)

        self.trainer.train(model=model, train_loader=train_data_loader,
                           optimizer=optimizer, scheduler=scheduler,
                           loss_fun=self.loss_fun,
                           evaluation_callback=evaluation_callback,
                           checkpointing_callback=checkpointing_callback)
