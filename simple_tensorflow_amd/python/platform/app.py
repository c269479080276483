"""tf.app (reference python/platform/app.py): run(main) after flag parse."""
import sys as _sys

from simple_tensorflow_amd.python.platform import flags as _flags


def run(main=None, argv=None):
    f = _flags.FLAGS
    args = argv[1:] if argv else _sys.argv[1:]
    unparsed = f._parse_flags(args)
    main = main or _sys.modules['__main__'].main
    _sys.exit(main([_sys.argv[0]] + unparsed))
