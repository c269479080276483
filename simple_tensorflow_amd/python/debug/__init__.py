from simple_tensorflow_amd.python.debug.debug_wrapper import (  # noqa
    DebugDumpDir, DebugTensorDatum, DumpingDebugWrapperSession,
    has_inf_or_nan, watch_graph)
