"""JPEG codec (python/lib/io/jpeg_codec.py; reference core/lib/jpeg
jpeg_mem.cc analog) + tf.image.encode_jpeg/decode_jpeg ops."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.lib.io import jpeg_codec


def setup_function(_):
    tf.reset_default_graph()


def _gradient_image(h=40, w=56):
    yy, xx = np.mgrid[0:h, 0:w]
    return np.stack([(yy * 4) % 256, (xx * 3) % 256,
                     ((yy + xx) * 2) % 256], -1).astype(np.uint8)


def test_roundtrip_rgb():
    img = _gradient_image()
    blob = jpeg_codec.encode_jpeg(img, quality=90)
    assert blob[:2] == b'\xff\xd8' and blob[-2:] == b'\xff\xd9'
    dec = jpeg_codec.decode_jpeg(blob)
    assert dec.shape == img.shape
    assert np.abs(dec.astype(int) - img.astype(int)).mean() < 3.0


def test_roundtrip_grayscale():
    yy, xx = np.mgrid[0:24, 0:32]
    g = ((yy * 7 + xx * 3) % 256).astype(np.uint8)[:, :, None]
    dec = jpeg_codec.decode_jpeg(jpeg_codec.encode_jpeg(g, 90))
    assert dec.shape == (24, 32, 1)
    assert np.abs(dec.astype(int) - g.astype(int)).mean() < 3.0


def test_quality_tradeoff():
    img = _gradient_image()
    hi = jpeg_codec.encode_jpeg(img, quality=95)
    lo = jpeg_codec.encode_jpeg(img, quality=20)
    assert len(lo) < len(hi)
    dec_lo = jpeg_codec.decode_jpeg(lo)
    assert np.abs(dec_lo.astype(int) - img.astype(int)).mean() < 30.0


def test_non_multiple_of_8():
    img = _gradient_image(h=13, w=21)
    dec = jpeg_codec.decode_jpeg(jpeg_codec.encode_jpeg(img, 90))
    assert dec.shape == (13, 21, 3)


def test_rejects_garbage():
    with pytest.raises(ValueError):
        jpeg_codec.decode_jpeg(b'not a jpeg at all')


def test_tf_ops_roundtrip():
    img = _gradient_image()
    enc = tf.image.encode_jpeg(tf.constant(img), quality=90)
    dec = tf.image.decode_jpeg(enc)
    auto = tf.image.decode_image(enc)
    with tf.Session() as s:
        dv, av = s.run([dec, auto])
    assert np.abs(dv.astype(int) - img.astype(int)).mean() < 3.0
    np.testing.assert_array_equal(dv, av)


def test_decode_channels_conversion():
    img = _gradient_image()
    enc = tf.image.encode_jpeg(tf.constant(img))
    one = tf.image.decode_jpeg(enc, channels=1)
    with tf.Session() as s:
        g = s.run(one)
    assert g.shape == (40, 56, 1)
