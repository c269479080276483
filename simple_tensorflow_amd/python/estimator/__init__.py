from simple_tensorflow_amd.python.estimator.estimator import (  # noqa
    Estimator, EstimatorSpec, ModeKeys, RunConfig)
