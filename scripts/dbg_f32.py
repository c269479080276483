"""Piecewise GPU-vs-numpy check of the MNIST f32 training step."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

import simple_tensorflow_amd as tf

rng = np.random.RandomState(0)
labels = rng.randint(0, 10, 512)
x_np = rng.randn(512, 784).astype(np.float32) * 0.5
for c in range(10):
    x_np[labels == c, c * 78:(c + 1) * 78] += 1.5
onehot = np.eye(10, dtype=np.float32)[labels]
w_np = rng.randn(784, 10).astype(np.float32) * 0.01
b_np = rng.randn(10).astype(np.float32) * 0.01

x = tf.constant(x_np)
y_ = tf.constant(onehot)
w = tf.Variable(tf.constant(w_np))
b = tf.Variable(tf.constant(b_np))
logits = tf.matmul(x, w) + b
loss = tf.reduce_mean(
    tf.nn.softmax_cross_entropy_with_logits(labels=y_, logits=logits))
gw, gb = tf.gradients(loss, [w.ref(), b.ref()])

# numpy reference
lg = x_np @ w_np + b_np
e = np.exp(lg - lg.max(1, keepdims=True))
p = e / e.sum(1, keepdims=True)
loss_np = -(onehot * np.log(p)).sum(1).mean()
dlg = (p - onehot) / 512.0
gw_np = x_np.T @ dlg
gb_np = dlg.sum(0)

with tf.Session() as s:
    s.run(tf.global_variables_initializer())
    lv, gwv, gbv, lgv = s.run([loss, gw, gb, logits])
    print('logits diff', np.abs(lgv - lg).max())
    print('loss', lv, 'np', loss_np, 'diff', abs(lv - loss_np))
    print('gw diff', np.abs(gwv - gw_np).max(), 'norm', np.abs(gw_np).max())
    print('gb diff', np.abs(gbv - gb_np).max(), 'norm', np.abs(gb_np).max())

    # raw transposed matmul check
    d = tf.constant(dlg)
    mm = tf.matmul(x, d, transpose_a=True)
    mmv = s.run(mm)
    print('mm_ta diff', np.abs(mmv - gw_np).max())

    # run 5 GD steps, print loss trajectory
    train = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
    for i in range(5):
        s.run(train)
        print('step', i, 'loss', s.run(loss))

# exact test_mnist graph: placeholders + feed_dict, zeros init
tf.reset_default_graph()
xp = tf.placeholder(tf.float32, [None, 784])
yp = tf.placeholder(tf.float32, [None, 10])
w2 = tf.Variable(tf.zeros([784, 10]))
b2 = tf.Variable(tf.zeros([10]))
logits2 = tf.matmul(xp, w2) + b2
loss2 = tf.reduce_mean(
    tf.nn.softmax_cross_entropy_with_logits(labels=yp, logits=logits2))
train2 = tf.train.GradientDescentOptimizer(0.5).minimize(loss2)
with tf.Session() as s:
    s.run(tf.global_variables_initializer())
    feed = {xp: x_np, yp: onehot}
    print('feed loss0', s.run(loss2, feed_dict=feed))
    s.run(train2, feed_dict=feed)
    xg = xp + 0.0
    xent = loss2.op.inputs[0]
    lgv, wv, bv, xv, xe = s.run([logits2, w2.ref(), b2.ref(), xg, xent],
                                feed_dict=feed)
    print('x roundtrip diff', np.abs(xv - x_np).max())
    lg_ref = x_np @ wv + bv
    print('logits diff vs (x@w_fetched)', np.abs(lgv - lg_ref).max())
    ee = np.exp(lg_ref - lg_ref.max(1, keepdims=True))
    pp = ee / ee.sum(1, keepdims=True)
    xent_ref = -(onehot * np.log(pp)).sum(1)
    print('xent diff', np.abs(xe - xent_ref).max(), 'mean ref',
          xent_ref.mean(), 'mean got', xe.mean())
    for i in range(4):
        s.run(train2, feed_dict=feed)
        print('feed step', i + 1, 'loss', s.run(loss2, feed_dict=feed),
              'wmax', np.abs(s.run(w2.ref())).max())
