"""Structure flatten/pack (reference python/util/nest.py)."""


def is_sequence(x):
    return isinstance(x, (list, tuple)) and not isinstance(x, str)


def flatten(structure):
    if not is_sequence(structure):
        return [structure]
    out = []
    for item in structure:
        out.extend(flatten(item))
    return out


def _pack(structure, flat, idx):
    if not is_sequence(structure):
        return flat[idx], idx + 1
    items = []
    for s in structure:
        v, idx = _pack(s, flat, idx)
        items.append(v)
    if isinstance(structure, tuple) and hasattr(structure, '_fields'):
        return type(structure)(*items), idx
    return type(structure)(items), idx


def pack_sequence_as(structure, flat_sequence):
    packed, idx = _pack(structure, list(flat_sequence), 0)
    if idx != len(flat_sequence):
        raise ValueError('Structure/flat length mismatch')
    return packed


def assert_same_structure(a, b):
    fa, fb = flatten(a), flatten(b)
    if len(fa) != len(fb):
        raise ValueError('Different structure lengths')


def map_structure(fn, structure):
    return pack_sequence_as(structure, [fn(x) for x in flatten(structure)])
