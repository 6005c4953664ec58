"""KerasExperiment descriptor (reference
``tf_yarn/tensorflow/keras_experiment.py:5-12``): same six fields."""

from typing import Any, Callable, Dict, NamedTuple, Optional


class KerasExperiment(NamedTuple):
    model: Any                      # KerasModel (or nn.Module wrapped)
    model_dir: str
    train_params: Dict              # epochs, batch_size, callbacks, ...
    input_data_fn: Optional[Callable]
    target_data_fn: Optional[Callable]
    validation_data_fn: Optional[Callable]
