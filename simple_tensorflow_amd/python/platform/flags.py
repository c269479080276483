"""tf.app.flags (reference python/platform/flags.py argparse wrapper)."""
import argparse as _argparse


class _FlagValues(object):
    def __init__(self):
        self.__dict__['__parser'] = _argparse.ArgumentParser()
        self.__dict__['__parsed'] = False

    def _parse_flags(self, args=None):
        result, unparsed = self.__dict__['__parser'].parse_known_args(args)
        for name, val in vars(result).items():
            self.__dict__[name] = val
        self.__dict__['__parsed'] = True
        return unparsed

    def __getattr__(self, name):
        if not self.__dict__['__parsed']:
            self._parse_flags([])
        if name not in self.__dict__:
            raise AttributeError(name)
        return self.__dict__[name]


FLAGS = _FlagValues()


def _define(flag_type, name, default, docstring):
    parser = FLAGS.__dict__['__parser']
    if flag_type is bool:
        parser.add_argument('--' + name, default=default, help=docstring,
                            type=lambda v: str(v).lower() in
                            ('true', 't', '1'), nargs='?', const=True)
        parser.add_argument('--no' + name, dest=name, action='store_false')
    else:
        parser.add_argument('--' + name, default=default, help=docstring,
                            type=flag_type)
    FLAGS.__dict__['__parsed'] = False


def DEFINE_string(name, default, docstring):
    _define(str, name, default, docstring)


def DEFINE_integer(name, default, docstring):
    _define(int, name, default, docstring)


def DEFINE_float(name, default, docstring):
    _define(float, name, default, docstring)


def DEFINE_boolean(name, default, docstring):
    _define(bool, name, default, docstring)


DEFINE_bool = DEFINE_boolean
