"""image/audio summaries + PNG codec (reference summary_image_op.cc /
core/lib/png analogs)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import pbreader
from simple_tensorflow_amd.python.lib.io import png_codec


def setup_function(_):
    tf.reset_default_graph()


def test_png_round_trip():
    rng = np.random.RandomState(0)
    img = rng.randint(0, 256, (5, 7, 3), dtype=np.uint8)
    data = png_codec.encode_png(img)
    back = png_codec.decode_png(data)
    np.testing.assert_array_equal(back, img)


def test_image_summary_proto():
    imgs = np.random.RandomState(1).rand(2, 4, 4, 3).astype(np.float32)
    summ = tf.summary.image('imgs', tf.constant(imgs), max_outputs=2)
    with tf.Session() as s:
        blob = s.run(summ)
    tags, pngs = [], []
    for f, w, v in pbreader._fields(blob):
        assert f == 1  # Summary.value
        for f2, _, v2 in pbreader._fields(v):
            if f2 == 1:
                tags.append(v2.decode())
            elif f2 == 4:  # image
                for f3, _, v3 in pbreader._fields(v2):
                    if f3 == 4:
                        pngs.append(bytes(v3))
    assert len(tags) == 2 and len(pngs) == 2
    decoded = png_codec.decode_png(pngs[0])
    assert decoded.shape == (4, 4, 3)


def test_encode_decode_png_ops():
    img = np.random.RandomState(2).randint(0, 255, (3, 3, 3),
                                           dtype=np.uint8)
    enc = tf.image.encode_png(tf.constant(img.astype(np.float32)))
    with tf.Session() as s:
        blob = s.run(enc)
    np.testing.assert_array_equal(png_codec.decode_png(blob), img)


def test_audio_summary():
    wave = np.sin(np.linspace(0, 20, 800)).astype(np.float32)[None, :]
    summ = tf.summary.audio('tone', tf.constant(wave), sample_rate=8000)
    with tf.Session() as s:
        blob = s.run(summ)
    assert b'RIFF' in blob and b'WAVE' in blob
