"""tf.summary ops (analog of reference python/summary/summary.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def scalar(name, tensor, collections=None):
    g = ops.get_default_graph()
    t = apply_op('ScalarSummary', ops.constant(name),
                 convert_to_tensor(tensor), name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t


def histogram(name, values, collections=None):
    g = ops.get_default_graph()
    t = apply_op('HistogramSummary', ops.constant(name),
                 convert_to_tensor(values), name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t


def merge(inputs, collections=None, name=None):
    return apply_op('MergeSummary', list(inputs), name=name)


def merge_all(key=ops.GraphKeys.SUMMARIES):
    summaries = ops.get_default_graph().get_collection(key)
    if not summaries:
        return None
    return merge(summaries)


def _summary_proto(tag, field_bytes, field_no):
    from simple_tensorflow_amd.python.framework.pbwire import f_bytes
    value = f_bytes(1, tag) + f_bytes(field_no, field_bytes)
    return f_bytes(1, value)  # Summary.value = 1


def image(name, tensor, max_outputs=3, collections=None):
    """Image summary: PNG-encodes up to max_outputs images of a
    [batch, h, w, c] tensor into a Summary proto (reference
    core/kernels/summary_image_op.cc via the python png codec)."""
    import numpy as np
    from simple_tensorflow_amd.python.framework.pbwire import (f_bytes,
                                                               f_varint)
    from simple_tensorflow_amd.python.lib.io import png_codec
    from simple_tensorflow_amd.python.ops import script_ops

    def _encode(imgs):
        out = b''
        n = min(len(imgs), max_outputs)
        for i in range(n):
            img = imgs[i]
            if img.dtype != np.uint8:
                lo, hi = float(img.min()), float(img.max())
                scale = 255.0 / (hi - lo) if hi > lo else 1.0
                img = ((img - lo) * scale).astype(np.uint8)
            png = png_codec.encode_png(img)
            im = (f_varint(1, img.shape[0]) + f_varint(2, img.shape[1]) +
                  f_varint(3, img.shape[2]) + f_bytes(4, png))
            tag = name if n == 1 else '%s/image/%d' % (name, i)
            out += _summary_proto(tag, im, 4)  # Value.image = 4
        return out

    g = ops.get_default_graph()
    t = script_ops.py_func(_encode, [convert_to_tensor(tensor)],
                           dtypes.string, name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t


def audio(name, tensor, sample_rate, max_outputs=3, collections=None):
    """Audio summary: WAV-encodes [batch, frames] or [batch, frames, ch]
    float waveforms (reference summary_audio_op.cc)."""
    import numpy as np
    import struct
    from simple_tensorflow_amd.python.framework.pbwire import (f_bytes,
                                                               f_float,
                                                               f_varint)
    from simple_tensorflow_amd.python.ops import script_ops

    def _wav(x, rate):
        if x.ndim == 1:
            x = x[:, None]
        frames, ch = x.shape
        pcm = np.clip(x * 32767.0, -32768, 32767).astype('<i2').tobytes()
        hdr = (b'RIFF' + struct.pack('<I', 36 + len(pcm)) + b'WAVEfmt ' +
               struct.pack('<IHHIIHH', 16, 1, ch, int(rate),
                           int(rate) * ch * 2, ch * 2, 16) +
               b'data' + struct.pack('<I', len(pcm)))
        return hdr + pcm

    def _encode(batch):
        out = b''
        n = min(len(batch), max_outputs)
        for i in range(n):
            wav = _wav(np.asarray(batch[i]), sample_rate)
            au = (f_float(1, float(sample_rate)) +
                  f_varint(2, 1 if batch[i].ndim == 1
                           else batch[i].shape[-1]) +
                  f_varint(3, batch[i].shape[0]) + f_bytes(4, wav) +
                  f_bytes(5, 'audio/wav'))
            tag = name if n == 1 else '%s/audio/%d' % (name, i)
            out += _summary_proto(tag, au, 6)  # Value.audio = 6
        return out

    g = ops.get_default_graph()
    t = script_ops.py_func(_encode, [convert_to_tensor(tensor)],
                           dtypes.string, name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t
