from simple_tensorflow_amd.python.saved_model.saved_model import (  # noqa
    SavedModelBuilder, loader, tag_constants, signature_constants,
    build_tensor_info, predict_signature_def, maybe_saved_model_directory)
