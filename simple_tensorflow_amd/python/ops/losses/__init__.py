from simple_tensorflow_amd.python.ops.losses.losses import *  # noqa
from simple_tensorflow_amd.python.ops.losses import losses  # noqa
