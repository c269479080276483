"""tf.image resize / flip / crop / standardization (reference
image_ops_impl.py + resize kernels analogs)."""
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
