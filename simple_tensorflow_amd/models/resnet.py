"""ResNet-50 v1 (bottleneck) for the MI355X bf16 training benchmark
(BASELINE.json configs 2-3): NHWC, bf16 activations/conv compute with f32
master weights (cast to bf16 per step), f32 batch-norm statistics via the
MI355X-native fused BatchNormMi op, f32 loss.

Reference parity note: the reference repo (TF 1.0) carries no model zoo; this
mirrors the tf_cnn_benchmarks-era ResNet-50 v1 graph structure the published
TF 1.0 numbers used.
"""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework.ops import apply_op


def _conv_var(name, shape, fan_in):
    init = tf.truncated_normal(shape, stddev=float(np.sqrt(2.0 / fan_in)),
                               dtype=tf.float32)
    from simple_tensorflow_amd.python.ops import variables
    return variables.Variable(init, name=name)


def batch_norm(x, c, name, relu=True):
    from simple_tensorflow_amd.python.ops import variables, array_ops
    scale = variables.Variable(array_ops.ones([c], tf.float32),
                               name=name + '_scale')
    offset = variables.Variable(array_ops.zeros([c], tf.float32),
                                name=name + '_offset')
    # BN+ReLU fused in one kernel pass (fwd relu in BnNormKernel, bwd mask
    # in BnGrad*) — saves a full-tensor elementwise pass each way.
    y, _, _, _ = apply_op('BatchNormMi', x, scale.ref(), offset.ref(),
                          epsilon=1e-4, fuse_relu=relu, name=name)
    y.set_shape(x._shape)
    return y


def conv2d_bf16(x, name, r, s, cin, cout, stride, relu_bn=True, bn_relu=True):
    w = _conv_var(name + '_w', [r, s, cin, cout], r * s * cin)
    w16 = tf.cast(w.ref(), tf.bfloat16)
    y = tf.nn.conv2d(x, w16, strides=[1, stride, stride, 1], padding='SAME')
    if relu_bn:
        y = batch_norm(y, cout, name + '_bn', relu=bn_relu)
    return y


def bottleneck(x, name, cin, cmid, cout, stride):
    from simple_tensorflow_amd.python.ops import variables, array_ops
    shortcut = x
    if cin != cout or stride != 1:
        shortcut = conv2d_bf16(x, name + '_proj', 1, 1, cin, cout, stride,
                               relu_bn=True, bn_relu=False)
    y = conv2d_bf16(x, name + '_a', 1, 1, cin, cmid, stride)
    y = conv2d_bf16(y, name + '_b', 3, 3, cmid, cmid, 1)
    # tail: conv -> fused BN+residual-add+ReLU (one elementwise pass)
    y = conv2d_bf16(y, name + '_c', 1, 1, cmid, cout, 1, relu_bn=False)
    scale = variables.Variable(array_ops.ones([cout], tf.float32),
                               name=name + '_c_bn_scale')
    offset = variables.Variable(array_ops.zeros([cout], tf.float32),
                                name=name + '_c_bn_offset')
    out, _, _, _ = apply_op('BatchNormAddReluMi', y, scale.ref(),
                            offset.ref(), shortcut, epsilon=1e-4,
                            name=name + '_c_bnar')
    out.set_shape(y._shape)
    return out


def resnet50_loss(images_bf16, labels_i64, num_classes=1000):
    """Builds the ResNet-50 v1 forward graph + sparse-xent loss.

    images_bf16: [N, 224, 224, 3] bf16; labels: [N] int64.
    Returns scalar f32 loss.
    """
    from simple_tensorflow_amd.python.ops import variables, array_ops
    x = conv2d_bf16(images_bf16, 'conv1', 7, 7, 3, 64, 2)
    x = tf.nn.max_pool(x, [1, 3, 3, 1], [1, 2, 2, 1], 'SAME')

    stages = [(64, 256, 3, 1), (128, 512, 4, 2), (256, 1024, 6, 2),
              (512, 2048, 3, 2)]
    cin = 64
    for si, (cmid, cout, blocks, stride) in enumerate(stages):
        for b in range(blocks):
            x = bottleneck(x, 'res%d_%d' % (si + 2, b), cin, cmid, cout,
                           stride if b == 0 else 1)
            cin = cout

    # global average pool 7x7 -> [N, 1, 1, 2048] -> [N, 2048]
    x = tf.nn.avg_pool(x, [1, 7, 7, 1], [1, 1, 1, 1], 'VALID')
    n = x._shape[0]
    x = tf.reshape(x, [n, 2048])

    fc_w = variables.Variable(
        tf.truncated_normal([2048, num_classes], stddev=0.01,
                            dtype=tf.float32), name='fc_w')
    fc_b = variables.Variable(array_ops.zeros([num_classes], tf.float32),
                              name='fc_b')
    logits16 = tf.matmul(x, tf.cast(fc_w.ref(), tf.bfloat16))
    logits = tf.cast(logits16, tf.float32) + fc_b.ref()
    loss_vec = tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels_i64, logits=logits)
    return tf.reduce_mean(loss_vec)


def synthetic_inputs(batch, image_size=224, num_classes=1000, seed=1234):
    """Synthetic 224x224x3 data resident in HBM: a random bf16 image batch
    (regenerated on-device once — Const-cached) and fixed random labels."""
    images = tf.random_uniform([batch, image_size, image_size, 3],
                               dtype=tf.bfloat16, seed=seed)
    rng = np.random.RandomState(seed)
    labels = tf.constant(rng.randint(0, num_classes, batch).astype(np.int64),
                         dtype=tf.int64)
    return images, labels


def build_train_graph(batch, lr=0.1, momentum=0.9):
    images, labels = synthetic_inputs(batch)
    loss = resnet50_loss(images, labels)
    opt = tf.train.MomentumOptimizer(lr, momentum)
    train_op = opt.minimize(loss)
    return loss, train_op
