import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import variables


def build():
    tf.reset_default_graph()
    rng = np.random.RandomState(13)
    x = tf.constant(rng.randn(256, 64).astype(np.float32))
    labels = tf.constant(rng.randint(0, 10, 256).astype(np.int64))
    w1 = variables.Variable(tf.truncated_normal([64, 128], stddev=0.1, seed=1))
    w2 = variables.Variable(tf.truncated_normal([128, 10], stddev=0.1, seed=2))
    h = tf.nn.relu(tf.matmul(x, w1.ref()))
    logits = tf.matmul(h, w2.ref())
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits))
    opt = tf.train.MomentumOptimizer(0.05, 0.9)
    train = opt.minimize(loss)
    mom2 = opt.get_slot(w2, 'momentum')
    probes = [loss, tf.reduce_sum(w1.ref() * w1.ref()),
              tf.reduce_sum(w2.ref() * w2.ref()),
              tf.reduce_sum(mom2.ref() * mom2.ref())]
    return train, probes


def run(tag, steps=8):
    train, probes = build()
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        for i in range(steps):
            s.run(train)
            vals = s.run(probes)
            print('%s step %2d  loss %10.4f  |w1|2 %10.4f  |w2|2 %10.4f'
                  '  |m2|2 %12.6f' % ((tag, i) + tuple(float(v) for v in vals)),
                  flush=True)


run('cap')

def run_nofetch(tag):
    train, probes = build()
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        for i in range(30):
            s.run(train)
        print(tag, 'final %.4f' % float(s.run(probes[0])), flush=True)

run_nofetch('nofetch')
