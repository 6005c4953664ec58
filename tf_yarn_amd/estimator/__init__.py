from tf_yarn_amd.estimator.client import run_on_yarn
from tf_yarn_amd.estimator.estimator import (DNNClassifier, Estimator,
                                             EvalSpec, LinearClassifier,
                                             RunConfig, TrainSpec,
                                             train_and_evaluate)
from tf_yarn_amd.estimator.experiment import Experiment
from tf_yarn_amd.estimator.keras import (KerasModel, ModelCheckpoint,
                                         load_model)
from tf_yarn_amd.estimator.keras_experiment import KerasExperiment

__all__ = ["run_on_yarn", "Experiment", "KerasExperiment",
           "Estimator", "DNNClassifier", "LinearClassifier",
           "TrainSpec", "EvalSpec",
           "RunConfig", "train_and_evaluate", "KerasModel",
           "ModelCheckpoint", "load_model"]
