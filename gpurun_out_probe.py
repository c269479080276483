import sys, os
sys.path.insert(0, '/root/repo')
import numpy as np
import simple_tensorflow_amd as tf
np.random.seed(0)
X = np.random.randn(64, 8).astype(np.float32)
Y = (X @ np.random.randn(8, 1).astype(np.float32))
for name, opt in [('Adagrad', tf.train.AdagradOptimizer(0.5)),
                  ('RMSProp', tf.train.RMSPropOptimizer(0.01)),
                  ('Adadelta', tf.train.AdadeltaOptimizer(1.0)),
                  ('Ftrl', tf.train.FtrlOptimizer(0.5))]:
    tf.reset_default_graph()
    w = tf.Variable(np.zeros((8, 1), np.float32))
    loss = tf.reduce_mean((tf.matmul(tf.constant(X), w.ref()) - tf.constant(Y)) ** 2.0)
    train = opt.minimize(loss)
    s = tf.Session()
    assert s.num_gpus() > 0
    s.run(tf.global_variables_initializer())
    l0 = s.run(loss)
    for _ in range(60):
        s.run(train)
    l1 = s.run(loss)
    status = 'OK' if l1 < l0 * 0.5 else 'BROKEN'
    print('%s: %.4f -> %.4f %s' % (name, l0, l1, status))
