import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: tests that need a real MI355X GPU (run via gpurun)')


def pytest_collection_modifyitems(config, items):
    try:
        import simple_tensorflow_amd as tf
        has_gpu = tf.Session(config={'device_count': {'GPU': 1}}).num_gpus() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason='no GPU available')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip)
