"""tf.gfile (reference python/platform/gfile.py over local filesystem)."""
import glob as _glob
import os as _os
import shutil as _shutil

GFile = open
Open = open
FastGFile = open


def Exists(path):
    return _os.path.exists(path)


def IsDirectory(path):
    return _os.path.isdir(path)


def Glob(pattern):
    return _glob.glob(pattern)


def MkDir(path):
    _os.mkdir(path)


def MakeDirs(path):
    _os.makedirs(path, exist_ok=True)


def Remove(path):
    _os.remove(path)


def DeleteRecursively(path):
    _shutil.rmtree(path)


def Rename(src, dst, overwrite=False):
    if overwrite and _os.path.exists(dst):
        _os.remove(dst)
    _os.rename(src, dst)


def Copy(src, dst, overwrite=False):
    if not overwrite and _os.path.exists(dst):
        raise OSError('destination exists: %s' % dst)
    _shutil.copyfile(src, dst)


def ListDirectory(path):
    return _os.listdir(path)


def Walk(top):
    return _os.walk(top)


def Stat(path):
    return _os.stat(path)
