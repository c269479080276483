"""Inception-v3 (BASELINE.json config 4): branch-heavy NHWC bf16 graph —
stresses concat/split and the dataflow scheduler. Standard v3 topology
(stem, 3x block35, block17 group with 1x7/7x1 factorized convs, block8
group), f32 master weights + BatchNormMi."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.models.resnet import batch_norm, _conv_var
from simple_tensorflow_amd.python.ops import variables


def conv_bn(x, name, r, s, cin, cout, stride=1, padding='SAME'):
    w = _conv_var(name + '_w', [r, s, cin, cout], r * s * cin)
    y = tf.nn.conv2d(x, tf.cast(w.ref(), tf.bfloat16),
                     [1, stride, stride, 1], padding)
    return batch_norm(y, cout, name + '_bn', relu=True)


def inception_v3_loss(images, labels, num_classes=1000):
    # stem: 299x299x3 -> 35x35x192
    x = conv_bn(images, 'stem1', 3, 3, 3, 32, 2, 'VALID')      # 149
    x = conv_bn(x, 'stem2', 3, 3, 32, 32, 1, 'VALID')          # 147
    x = conv_bn(x, 'stem3', 3, 3, 32, 64, 1, 'SAME')           # 147
    x = tf.nn.max_pool(x, [1, 3, 3, 1], [1, 2, 2, 1], 'VALID')  # 73
    x = conv_bn(x, 'stem4', 1, 1, 64, 80, 1, 'VALID')
    x = conv_bn(x, 'stem5', 3, 3, 80, 192, 1, 'VALID')          # 71
    x = tf.nn.max_pool(x, [1, 3, 3, 1], [1, 2, 2, 1], 'VALID')  # 35

    def block35(x, cin, name, pool_ch):
        b1 = conv_bn(x, name + '_b1', 1, 1, cin, 64)
        b2 = conv_bn(x, name + '_b2a', 1, 1, cin, 48)
        b2 = conv_bn(b2, name + '_b2b', 5, 5, 48, 64)
        b3 = conv_bn(x, name + '_b3a', 1, 1, cin, 64)
        b3 = conv_bn(b3, name + '_b3b', 3, 3, 64, 96)
        b3 = conv_bn(b3, name + '_b3c', 3, 3, 96, 96)
        b4 = tf.nn.avg_pool(x, [1, 3, 3, 1], [1, 1, 1, 1], 'SAME')
        b4 = conv_bn(b4, name + '_b4', 1, 1, cin, pool_ch)
        return tf.concat([b1, b2, b3, b4], 3)

    x = block35(x, 192, 'mixed0', 32)    # 256
    x = block35(x, 256, 'mixed1', 64)    # 288
    x = block35(x, 288, 'mixed2', 64)    # 288

    # reduction to 17x17
    b1 = conv_bn(x, 'red1_b1', 3, 3, 288, 384, 2, 'VALID')
    b2 = conv_bn(x, 'red1_b2a', 1, 1, 288, 64)
    b2 = conv_bn(b2, 'red1_b2b', 3, 3, 64, 96)
    b2 = conv_bn(b2, 'red1_b2c', 3, 3, 96, 96, 2, 'VALID')
    b3 = tf.nn.max_pool(x, [1, 3, 3, 1], [1, 2, 2, 1], 'VALID')
    x = tf.concat([b1, b2, b3], 3)       # 17x17x768

    def block17(x, name, ch7):
        b1 = conv_bn(x, name + '_b1', 1, 1, 768, 192)
        b2 = conv_bn(x, name + '_b2a', 1, 1, 768, ch7)
        b2 = conv_bn(b2, name + '_b2b', 1, 7, ch7, ch7)
        b2 = conv_bn(b2, name + '_b2c', 7, 1, ch7, 192)
        b3 = conv_bn(x, name + '_b3a', 1, 1, 768, ch7)
        b3 = conv_bn(b3, name + '_b3b', 7, 1, ch7, ch7)
        b3 = conv_bn(b3, name + '_b3c', 1, 7, ch7, ch7)
        b3 = conv_bn(b3, name + '_b3d', 7, 1, ch7, ch7)
        b3 = conv_bn(b3, name + '_b3e', 1, 7, ch7, 192)
        b4 = tf.nn.avg_pool(x, [1, 3, 3, 1], [1, 1, 1, 1], 'SAME')
        b4 = conv_bn(b4, name + '_b4', 1, 1, 768, 192)
        return tf.concat([b1, b2, b3, b4], 3)   # 768

    x = block17(x, 'mixed4', 128)
    x = block17(x, 'mixed5', 160)
    x = block17(x, 'mixed6', 160)
    x = block17(x, 'mixed7', 192)

    # reduction to 8x8
    b1 = conv_bn(x, 'red2_b1a', 1, 1, 768, 192)
    b1 = conv_bn(b1, 'red2_b1b', 3, 3, 192, 320, 2, 'VALID')
    b2 = conv_bn(x, 'red2_b2a', 1, 1, 768, 192)
    b2 = conv_bn(b2, 'red2_b2b', 1, 7, 192, 192)
    b2 = conv_bn(b2, 'red2_b2c', 7, 1, 192, 192)
    b2 = conv_bn(b2, 'red2_b2d', 3, 3, 192, 192, 2, 'VALID')
    b3 = tf.nn.max_pool(x, [1, 3, 3, 1], [1, 2, 2, 1], 'VALID')
    x = tf.concat([b1, b2, b3], 3)       # 8x8x1280

    def block8(x, cin, name):
        b1 = conv_bn(x, name + '_b1', 1, 1, cin, 320)
        b2 = conv_bn(x, name + '_b2a', 1, 1, cin, 384)
        b2a = conv_bn(b2, name + '_b2b', 1, 3, 384, 384)
        b2b = conv_bn(b2, name + '_b2c', 3, 1, 384, 384)
        b3 = conv_bn(x, name + '_b3a', 1, 1, cin, 448)
        b3 = conv_bn(b3, name + '_b3b', 3, 3, 448, 384)
        b3a = conv_bn(b3, name + '_b3c', 1, 3, 384, 384)
        b3b = conv_bn(b3, name + '_b3d', 3, 1, 384, 384)
        b4 = tf.nn.avg_pool(x, [1, 3, 3, 1], [1, 1, 1, 1], 'SAME')
        b4 = conv_bn(b4, name + '_b4', 1, 1, cin, 192)
        return tf.concat([b1, b2a, b2b, b3a, b3b, b4], 3)  # 2048

    x = block8(x, 1280, 'mixed9')
    x = block8(x, 2048, 'mixed10')

    x = tf.nn.avg_pool(x, [1, 8, 8, 1], [1, 1, 1, 1], 'VALID')
    n = x._shape[0]
    x = tf.reshape(x, [n, 2048])
    fc_w = variables.Variable(
        tf.truncated_normal([2048, num_classes], stddev=0.01), name='fc_w')
    fc_b = variables.Variable(tf.zeros([num_classes]), name='fc_b')
    logits = tf.cast(tf.matmul(x, tf.cast(fc_w.ref(), tf.bfloat16)),
                     tf.float32) + fc_b.ref()
    loss_vec = tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits)
    return tf.reduce_mean(loss_vec)


def build_train_graph(batch=128, lr=0.045, seed=1234):
    images = tf.random_uniform([batch, 299, 299, 3], dtype=tf.bfloat16,
                               seed=seed)
    rng = np.random.RandomState(seed)
    labels = tf.constant(rng.randint(0, 1000, batch).astype(np.int64))
    loss = inception_v3_loss(images, labels)
    opt = tf.train.MomentumOptimizer(lr, 0.9)
    train_op = opt.minimize(loss)
    return loss, train_op
