"""tf.estimator.Estimator — model_fn-driven train/evaluate/predict over
MonitoredTrainingSession (reference python/estimator/estimator.py:47,
model_fn.py EstimatorSpec, run_config.py RunConfig)."""
import os

from simple_tensorflow_amd.python.framework import ops


class ModeKeys(object):
    TRAIN = 'train'
    EVAL = 'eval'
    PREDICT = 'infer'


class EstimatorSpec(object):
    def __init__(self, mode, predictions=None, loss=None, train_op=None,
                 eval_metric_ops=None, training_hooks=None,
                 evaluation_hooks=None, export_outputs=None,
                 scaffold=None):
        self.mode = mode
        self.predictions = predictions
        self.loss = loss
        self.train_op = train_op
        self.eval_metric_ops = eval_metric_ops or {}
        self.training_hooks = list(training_hooks or [])
        self.evaluation_hooks = list(evaluation_hooks or [])
        self.export_outputs = export_outputs
        self.scaffold = scaffold
        if mode == ModeKeys.TRAIN and train_op is None:
            raise ValueError('train_op required in TRAIN mode')


class RunConfig(object):
    def __init__(self, model_dir=None, save_summary_steps=100,
                 save_checkpoints_steps=None, save_checkpoints_secs=600,
                 keep_checkpoint_max=5, log_step_count_steps=100,
                 tf_random_seed=None, session_config=None):
        self.model_dir = model_dir
        self.save_summary_steps = save_summary_steps
        self.save_checkpoints_steps = save_checkpoints_steps
        self.save_checkpoints_secs = save_checkpoints_secs
        self.keep_checkpoint_max = keep_checkpoint_max
        self.log_step_count_steps = log_step_count_steps
        self.tf_random_seed = tf_random_seed
        self.session_config = session_config


class Estimator(object):
    def __init__(self, model_fn, model_dir=None, config=None, params=None):
        self._model_fn = model_fn
        self._config = config or RunConfig()
        self._model_dir = model_dir or self._config.model_dir or \
            os.path.join(os.getcwd(), 'estimator_model')
        self._params = params or {}

    @property
    def model_dir(self):
        return self._model_dir

    @property
    def config(self):
        return self._config

    @property
    def params(self):
        return self._params

    def _call_model_fn(self, features, labels, mode):
        import inspect
        kwargs = {}
        sig = inspect.signature(self._model_fn).parameters
        if 'params' in sig:
            kwargs['params'] = self._params
        if 'config' in sig:
            kwargs['config'] = self._config
        if 'mode' in sig:
            kwargs['mode'] = mode
        if 'labels' in sig:
            return self._model_fn(features, labels, **kwargs)
        return self._model_fn(features, **kwargs)

    def train(self, input_fn, steps=None, max_steps=None, hooks=None):
        import simple_tensorflow_amd as tf
        g = ops.Graph()
        with g.as_default():
            gstep = tf.train.get_or_create_global_step()
            features, labels = input_fn()
            spec = self._call_model_fn(features, labels, ModeKeys.TRAIN)
            all_hooks = list(hooks or []) + spec.training_hooks
            if max_steps is not None:
                all_hooks.append(
                    tf.train.StopAtStepHook(last_step=max_steps))
            elif steps is not None:
                all_hooks.append(tf.train.StopAtStepHook(num_steps=steps))
            with tf.train.MonitoredTrainingSession(
                    checkpoint_dir=self._model_dir, hooks=all_hooks,
                    save_checkpoint_secs=600) as sess:
                while not sess.should_stop():
                    sess.run(spec.train_op)
        return self

    def _restore_session(self, g):
        import simple_tensorflow_amd as tf
        sess = tf.Session(graph=g)
        with g.as_default():
            sess.run(tf.global_variables_initializer())
            ckpt = tf.train.latest_checkpoint(self._model_dir)
            if ckpt:
                tf.train.Saver().restore(sess, ckpt)
        return sess

    def evaluate(self, input_fn, steps=1, hooks=None):
        import simple_tensorflow_amd as tf
        g = ops.Graph()
        with g.as_default():
            tf.train.get_or_create_global_step()
            features, labels = input_fn()
            spec = self._call_model_fn(features, labels, ModeKeys.EVAL)
            sess = self._restore_session(g)
            totals = {k: 0.0 for k in spec.eval_metric_ops}
            totals['loss'] = 0.0
            for _ in range(steps):
                vals = sess.run(
                    dict(list(spec.eval_metric_ops.items()) +
                         [('loss', spec.loss)]))
                for k, v in vals.items():
                    totals[k] += float(v if not isinstance(v, tuple) else
                                       v[0])
            sess.close()
            return {k: v / steps for k, v in totals.items()}

    def predict(self, input_fn, hooks=None):
        import simple_tensorflow_amd as tf
        g = ops.Graph()
        with g.as_default():
            tf.train.get_or_create_global_step()
            features = input_fn()
            if isinstance(features, tuple):
                features = features[0]
            spec = self._call_model_fn(features, None, ModeKeys.PREDICT)
            sess = self._restore_session(g)
            preds = sess.run(spec.predictions)
            sess.close()
            if isinstance(preds, dict):
                n = len(next(iter(preds.values())))
                for i in range(n):
                    yield {k: v[i] for k, v in preds.items()}
            else:
                for row in preds:
                    yield row


def _default_serving_input_fn_placeholder(features_spec):
    import simple_tensorflow_amd as tf
    return {k: tf.placeholder(dtype, shape, name=k)
            for k, (dtype, shape) in features_spec.items()}


class ServingInputReceiver(object):
    def __init__(self, features, receiver_tensors):
        self.features = features
        self.receiver_tensors = receiver_tensors


def build_raw_serving_input_receiver_fn(features):
    """features: dict name -> placeholder-like Tensor spec (dtype, shape)."""
    def fn():
        import simple_tensorflow_amd as tf
        phs = {k: tf.placeholder(v[0], v[1], name=k)
               for k, v in features.items()}
        return ServingInputReceiver(phs, phs)
    return fn


def _export_savedmodel(self, export_dir_base, serving_input_receiver_fn):
    """Estimator.export_savedmodel (reference estimator export.py): builds
    the PREDICT graph, restores the latest checkpoint, writes a SavedModel
    with a serving_default signature."""
    import os as _os
    import time as _time
    import simple_tensorflow_amd as tf
    from simple_tensorflow_amd.python import saved_model as sm
    g = ops.Graph()
    with g.as_default():
        tf.train.get_or_create_global_step()
        receiver = serving_input_receiver_fn()
        spec = self._call_model_fn(receiver.features, None,
                                   ModeKeys.PREDICT)
        sess = self._restore_session(g)
        export_dir = _os.path.join(export_dir_base,
                                   str(int(_time.time())))
        b = sm.SavedModelBuilder(export_dir)
        preds = spec.predictions
        if not isinstance(preds, dict):
            preds = {'output': preds}
        sig = sm.predict_signature_def(inputs=receiver.receiver_tensors,
                                       outputs=preds)
        b.add_meta_graph_and_variables(
            sess, [sm.tag_constants.SERVING],
            signature_def_map={
                sm.signature_constants.DEFAULT_SERVING_SIGNATURE_DEF_KEY:
                    sig})
        b.save()
        sess.close()
    return export_dir


Estimator.export_savedmodel = _export_savedmodel
Estimator.export_saved_model = _export_savedmodel
