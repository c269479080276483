"""GIF codec (lib/io/gif_codec.py; reference core/lib/gif/gif_io.cc
DecodeGif) — encoder→decoder round trip, multi-frame shape, interlace,
transparency/disposal compositing, decode_image dispatch, garbage
rejection."""
import struct

import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.lib.io import gif_codec


def _grad_frame(h, w, shift=0):
    yy, xx = np.mgrid[0:h, 0:w]
    return np.stack([(yy * 12 + shift) % 256, (xx * 8) % 256,
                     ((yy + xx) * 5) % 256], -1).astype(np.uint8)


def test_round_trip_multiframe():
    f0 = _grad_frame(20, 30)
    f1 = np.roll(f0, 5, axis=1)
    frames = np.stack([f0, f1])
    dec = gif_codec.decode_gif(gif_codec.encode_gif(frames))
    assert dec.shape == (2, 20, 30, 3)
    err = np.abs(dec.astype(int) - frames.astype(int)).mean()
    assert err < 30  # 6x7x6 uniform palette quantization bound


def test_round_trip_random_grows_code_width():
    # random pixels force the LZW table past 512/1024 entries so the
    # variable code width path is exercised on both sides
    rng = np.random.RandomState(7)
    img = rng.randint(0, 256, (64, 97, 3), np.uint8)
    dec = gif_codec.decode_gif(gif_codec.encode_gif(img))
    assert dec.shape == (1, 64, 97, 3)
    assert np.abs(dec[0].astype(int) - img.astype(int)).mean() < 35


def test_interlaced_frame():
    # re-pack an encoded frame's rows in interlace order and set the flag
    img = _grad_frame(16, 8)
    blob = bytearray(gif_codec.encode_gif(img))
    # image descriptor starts after 13-byte header + 768-byte GCT
    desc = 13 + 768
    assert blob[desc] == 0x2C
    # decode the existing index stream, reorder rows, re-encode
    h, w = 16, 8
    idx = np.frombuffer(
        gif_codec._lzw_decode(8, _collect_subblocks(bytes(blob), desc + 11)),
        np.uint8)[:h * w].reshape(h, w)
    order = [y for start, step in ((0, 8), (4, 8), (2, 4), (1, 2))
             for y in range(start, h, step)]
    inter = idx[order].reshape(-1)
    out = bytearray(blob[:desc + 10])
    out[desc + 9] |= 0x40  # interlace flag lives in the descriptor flags
    out += bytes([8]) + gif_codec._lzw_encode(8, inter.tobytes())
    out += b'\x00\x3b'
    dec = gif_codec.decode_gif(bytes(out))
    ref = gif_codec.decode_gif(bytes(blob))
    np.testing.assert_array_equal(dec, ref)


def _collect_subblocks(data, off):
    lzw = bytearray()
    while True:
        sz = data[off]
        off += 1
        if sz == 0:
            return bytes(lzw)
        lzw += data[off:off + sz]
        off += sz


def test_transparency_composites_over_previous_frame():
    """Frame 2 marks palette index 0 transparent; those pixels must show
    frame 1 underneath (disposal 1 = leave in place)."""
    # hand-build a 2-frame gif with a 4-color table
    pal = bytes([255, 0, 0, 0, 255, 0, 0, 0, 255, 9, 9, 9]) + b'\x00' * 756
    head = b'GIF89a' + struct.pack('<HHBBB', 2, 2, 0x80 | 7, 3, 0) + pal
    f1 = bytes([1, 1, 1, 1])          # all green
    f2 = bytes([0, 2, 0, 2])          # red/blue, index 0 transparent
    gce = b'\x21\xf9\x04' + bytes([(1 << 2) | 1]) + b'\x00\x00\x00\x00'
    body = b''
    for i, (ix, g) in enumerate(((f1, b''), (f2, gce))):
        body += g + b'\x2c' + struct.pack('<HHHHB', 0, 0, 2, 2, 0)
        body += bytes([2]) + gif_codec._lzw_encode(2, ix) + b'\x00'
    dec = gif_codec.decode_gif(head + body + b'\x3b')
    assert dec.shape == (2, 2, 2, 3)
    np.testing.assert_array_equal(dec[0, 0, 0], [0, 255, 0])
    # transparent slots keep green, opaque slots become blue
    np.testing.assert_array_equal(dec[1, 0, 0], [0, 255, 0])
    np.testing.assert_array_equal(dec[1, 0, 1], [0, 0, 255])


def test_bad_signature_rejected():
    with pytest.raises(ValueError):
        gif_codec.decode_gif(b'NOTAGIF' + b'\x00' * 40)


def test_tf_op_round_trip():
    tf.reset_default_graph()
    frames = np.stack([_grad_frame(12, 17), _grad_frame(12, 17, 40)])
    blob = tf.image.encode_gif(tf.constant(frames))
    dec = tf.image.decode_gif(blob)
    with tf.Session() as s:
        out = s.run(dec)
    assert out.shape == (2, 12, 17, 3)
    assert np.abs(out.astype(int) - frames.astype(int)).mean() < 30


def test_decode_image_dispatches_gif():
    tf.reset_default_graph()
    img = _grad_frame(9, 11)
    blob = gif_codec.encode_gif(img)
    assert blob[:4] == b'GIF8'
    out_t = tf.image.decode_image(tf.constant(blob))
    with tf.Session() as s:
        out = s.run(out_t)
    assert out.shape == (1, 9, 11, 3)
