"""tf.estimator.Estimator train/evaluate/predict (reference
python/estimator/estimator.py:47 analog)."""
import numpy as np

import simple_tensorflow_amd as tf


def _model_fn(features, labels, mode, params):
    from simple_tensorflow_amd.python.ops import variables
    if isinstance(features, dict):
        features = features['x']
    w = tf.get_variable('w', [1, 1],
                        initializer=tf.constant_initializer(0.0))
    b = tf.get_variable('b', [1],
                        initializer=tf.constant_initializer(0.0))
    pred = tf.matmul(features, w) + b
    if mode == tf.estimator.ModeKeys.PREDICT:
        return tf.estimator.EstimatorSpec(mode, predictions=pred)
    loss = tf.reduce_mean(tf.square(pred - labels))
    if mode == tf.estimator.ModeKeys.EVAL:
        return tf.estimator.EstimatorSpec(mode, loss=loss)
    gstep = tf.train.get_global_step()
    train_op = tf.train.GradientDescentOptimizer(0.1).minimize(
        loss, global_step=gstep)
    return tf.estimator.EstimatorSpec(mode, loss=loss, train_op=train_op)


def _input_fn():
    x = np.array([[1.0], [2.0], [3.0], [4.0]], dtype=np.float32)
    y = 2.0 * x + 1.0
    return tf.constant(x), tf.constant(y)


def test_estimator_train_eval_predict(tmp_path):
    est = tf.estimator.Estimator(_model_fn, model_dir=str(tmp_path))
    est.train(_input_fn, steps=200)
    metrics = est.evaluate(_input_fn, steps=1)
    assert metrics['loss'] < 0.05
    preds = list(est.predict(lambda: tf.constant(
        np.array([[5.0]], dtype=np.float32))))
    assert abs(float(preds[0][0]) - 11.0) < 0.8


def test_estimator_export_savedmodel(tmp_path):
    import numpy as np
    from simple_tensorflow_amd.python import saved_model as sm
    est = tf.estimator.Estimator(_model_fn, model_dir=str(tmp_path / 'm'))
    est.train(_input_fn, steps=100)
    recv_fn = tf.estimator.estimator.build_raw_serving_input_receiver_fn(
        {'x': (tf.float32, [None, 1])})
    export_dir = est.export_savedmodel(str(tmp_path / 'exp'), recv_fn)
    tf.reset_default_graph()
    with tf.Session() as s:
        info = sm.loader.load(s, [sm.tag_constants.SERVING], export_dir)
        sig = info['signatures']['serving_default']
        g = tf.get_default_graph()
        out = s.run(g.get_tensor_by_name(sig['outputs']['output']['name']),
                    {g.get_tensor_by_name(sig['inputs']['x']['name']):
                     np.array([[2.0]], dtype=np.float32)})
    assert abs(float(out[0][0]) - 5.0) < 1.0  # ~2*2+1
