"""tf.image resize / flip / crop / standardization (reference
image_ops_impl.py + resize kernels analogs)."""
import colorsys

import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def test_resize_bilinear_upscale():
    img = np.arange(4, dtype=np.float32).reshape(1, 2, 2, 1)
    out = tf.image.resize_bilinear(tf.constant(img), [4, 4])
    with tf.Session() as s:
        v = s.run(out)
    assert v.shape == (1, 4, 4, 1)
    assert v[0, 0, 0, 0] == 0.0
    assert abs(v[0, 0, 1, 0] - 0.5) < 1e-6  # halfway between 0 and 1


def test_resize_nearest():
    img = np.array([[1.0, 2.0], [3.0, 4.0]],
                   dtype=np.float32).reshape(1, 2, 2, 1)
    out = tf.image.resize_nearest_neighbor(tf.constant(img), [4, 4])
    with tf.Session() as s:
        v = s.run(out)[0, :, :, 0]
    np.testing.assert_allclose(v[:2, :2], [[1, 1], [1, 1]])  # floor mapping
    np.testing.assert_allclose(v[2:, 2:], [[4, 4], [4, 4]])


def test_resize_bilinear_gradient():
    img = tf.placeholder(tf.float32, [1, 2, 2, 1])
    out = tf.image.resize_bilinear(img, [4, 4])
    loss = tf.reduce_sum(out)
    g = tf.gradients(loss, [img])[0]
    with tf.Session() as s:
        gv = s.run(g, {img: np.ones((1, 2, 2, 1), dtype=np.float32)})
    # gradient mass conserves: sum == out size
    assert abs(gv.sum() - 16.0) < 1e-4


def test_flips_and_crop():
    img = np.arange(12, dtype=np.float32).reshape(2, 2, 3)
    lr = tf.image.flip_left_right(tf.constant(img))
    ud = tf.image.flip_up_down(tf.constant(img))
    crop = tf.image.crop_to_bounding_box(tf.constant(img), 0, 1, 2, 1)
    with tf.Session() as s:
        vlr, vud, vc = s.run([lr, ud, crop])
    np.testing.assert_allclose(vlr, img[:, ::-1, :])
    np.testing.assert_allclose(vud, img[::-1, :, :])
    np.testing.assert_allclose(vc, img[:, 1:2, :])


def test_per_image_standardization():
    img = np.random.RandomState(0).rand(4, 4, 3).astype(np.float32)
    out = tf.image.per_image_standardization(tf.constant(img))
    with tf.Session() as s:
        v = s.run(out)
    assert abs(v.mean()) < 1e-5
    assert abs(v.std() - 1.0) < 1e-4


def _run(t):
    with tf.Session() as s:
        return s.run(t)


def test_hsv_roundtrip_and_reference():
    img = np.random.rand(5, 7, 3).astype(np.float32)
    hsv = _run(tf.image.rgb_to_hsv(tf.constant(img)))
    back = _run(tf.image.hsv_to_rgb(tf.constant(hsv)))
    np.testing.assert_allclose(back, img, atol=1e-5)
    for i in range(5):
        want = colorsys.rgb_to_hsv(*img[i, 0])
        np.testing.assert_allclose(hsv[i, 0], want, atol=1e-5)


def test_adjust_brightness_contrast():
    img = np.random.rand(2, 4, 4, 3).astype(np.float32)
    b = _run(tf.image.adjust_brightness(tf.constant(img), 0.1))
    np.testing.assert_allclose(b, img + 0.1, rtol=1e-6)
    c = _run(tf.image.adjust_contrast(tf.constant(img), 2.0))
    mean = img.mean(axis=(1, 2), keepdims=True)
    np.testing.assert_allclose(c, (img - mean) * 2.0 + mean, atol=1e-5)


def test_adjust_saturation_hue():
    img = np.random.rand(3, 3, 3).astype(np.float32)
    desat = _run(tf.image.adjust_saturation(tf.constant(img), 0.0))
    # zero saturation -> grayscale (r == g == b)
    np.testing.assert_allclose(desat[..., 0], desat[..., 1], atol=1e-5)
    np.testing.assert_allclose(desat[..., 1], desat[..., 2], atol=1e-5)
    rot = _run(tf.image.adjust_hue(tf.constant(img), 1.0))  # full wrap
    np.testing.assert_allclose(rot, img, atol=1e-4)


def test_grayscale():
    img = np.random.rand(4, 4, 3).astype(np.float32)
    g = _run(tf.image.rgb_to_grayscale(tf.constant(img)))
    assert g.shape == (4, 4, 1)
    want = img @ np.array([0.2989, 0.587, 0.114], np.float32)
    np.testing.assert_allclose(g[..., 0], want, atol=1e-5)
    rgb = _run(tf.image.grayscale_to_rgb(tf.constant(g)))
    assert rgb.shape == (4, 4, 3)


def test_non_max_suppression():
    boxes = np.array([[0, 0, 1, 1], [0, 0.05, 1, 1.05], [0, 2, 1, 3],
                      [0, 2.02, 1, 3.02]], np.float32)
    scores = np.array([0.9, 0.85, 0.7, 0.95], np.float32)
    sel = _run(tf.image.non_max_suppression(boxes, scores, 4, 0.5))
    assert sel.tolist() == [3, 0]


def test_sample_distorted_bounding_box():
    begin, size, bb = _run(list(tf.image.sample_distorted_bounding_box(
        np.array([100, 200, 3], np.int32),
        np.zeros((1, 1, 4), np.float32), seed=11)))
    assert begin.shape == (3,) and size.shape == (3,)
    assert 0 <= begin[0] and begin[0] + size[0] <= 100
    assert 0 <= begin[1] and begin[1] + size[1] <= 200
    assert size[2] == -1 and begin[2] == 0
    assert bb.shape == (1, 1, 4)


def test_pad_and_crop_or_pad():
    img = np.random.rand(4, 6, 3).astype(np.float32)
    padded = _run(tf.image.pad_to_bounding_box(tf.constant(img), 2, 1, 10,
                                               9))
    assert padded.shape == (10, 9, 3)
    np.testing.assert_allclose(padded[2:6, 1:7], img)
    assert padded[0].sum() == 0
    fitted = _run(tf.image.resize_image_with_crop_or_pad(tf.constant(img),
                                                         2, 8))
    assert fitted.shape == (2, 8, 3)


def test_total_variation():
    img = np.random.rand(5, 5, 3).astype(np.float32)
    tv = _run(tf.image.total_variation(tf.constant(img)))
    want = np.abs(np.diff(img, axis=0)).sum() + \
        np.abs(np.diff(img, axis=1)).sum()
    np.testing.assert_allclose(tv, want, rtol=1e-4)
