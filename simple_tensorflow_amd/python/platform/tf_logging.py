"""tf.logging (reference python/platform/tf_logging.py)."""
import logging as _logging
import sys as _sys

DEBUG = _logging.DEBUG
INFO = _logging.INFO
WARN = _logging.WARN
ERROR = _logging.ERROR
FATAL = _logging.FATAL

_logger = _logging.getLogger('simple_tensorflow_amd')
if not _logger.handlers:
    _h = _logging.StreamHandler(_sys.stderr)
    _h.setFormatter(_logging.Formatter(
        '%(levelname).1s %(asctime)s %(message)s'))
    _logger.addHandler(_h)
    _logger.setLevel(_logging.INFO)

debug = _logger.debug
info = _logger.info
warn = _logger.warning
warning = _logger.warning
error = _logger.error
fatal = _logger.critical
log = _logger.log


def set_verbosity(level):
    _logger.setLevel(level)


def get_verbosity():
    return _logger.level
