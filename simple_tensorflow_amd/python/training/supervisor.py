"""tf.train.Supervisor — legacy training harness predating MonitoredSession
(reference python/training/supervisor.py:1074; same lifecycle: prepare or
wait for session, launch queue runners, periodic checkpoint/summary, clean
stop)."""
import os
import time

from simple_tensorflow_amd.python.framework import ops


class Supervisor(object):
    def __init__(self, graph=None, is_chief=True, init_op=None, logdir=None,
                 summary_op=None, saver=None, global_step=None,
                 save_model_secs=600, save_summaries_secs=120,
                 checkpoint_basename='model.ckpt', session_manager=None,
                 recovery_wait_secs=30):
        import simple_tensorflow_amd as tf
        self._graph = graph or ops.get_default_graph()
        self._is_chief = is_chief
        self._logdir = logdir
        self._save_model_secs = save_model_secs
        self._checkpoint_path = os.path.join(logdir, checkpoint_basename) \
            if logdir else None
        with self._graph.as_default():
            from simple_tensorflow_amd.python.ops import variables
            if init_op is None and variables.global_variables():
                init_op = tf.global_variables_initializer()
            self._init_op = init_op
            if saver is None and variables.global_variables():
                saver = tf.train.Saver()
            self._saver = saver
            self._global_step = global_step if global_step is not None \
                else tf.train.get_global_step()
        from simple_tensorflow_amd.python.training import coordinator
        self._coord = coordinator.Coordinator()
        self._last_save = 0.0
        self._sess = None

    @property
    def coord(self):
        return self._coord

    @property
    def saver(self):
        return self._saver

    @property
    def global_step(self):
        return self._global_step

    def managed_session(self, master='', config=None,
                        start_standard_services=True):
        import contextlib

        @contextlib.contextmanager
        def _ctx():
            sess = self.prepare_or_wait_for_session(master, config)
            try:
                yield sess
            except Exception:
                self._coord.request_stop()
                raise
            finally:
                self.stop()
            return

        return _ctx()

    def prepare_or_wait_for_session(self, master='', config=None):
        import simple_tensorflow_amd as tf
        with self._graph.as_default():
            sess = tf.Session(master, graph=self._graph, config=config)
            ckpt = tf.train.latest_checkpoint(self._logdir) \
                if self._logdir else None
            if ckpt and self._saver:
                self._saver.restore(sess, ckpt)
            elif self._init_op is not None:
                sess.run(self._init_op)
            from simple_tensorflow_amd.python.training import coordinator
            coordinator.start_queue_runners(sess, self._coord)
        self._sess = sess
        self._last_save = time.time()
        return sess

    def should_stop(self):
        return self._coord.should_stop()

    def request_stop(self, ex=None):
        self._coord.request_stop(ex)

    def stop(self, threads=None):
        self._coord.request_stop()
        try:
            self._coord.join(threads)
        except Exception:
            pass
        if self._is_chief and self._sess is not None and self._saver and \
                self._checkpoint_path:
            self._saver.save(self._sess, self._checkpoint_path,
                             global_step=self._global_step)

    def maybe_save(self, sess=None):
        """Periodic checkpoint (called from the training loop)."""
        if not (self._is_chief and self._saver and self._checkpoint_path):
            return
        now = time.time()
        if now - self._last_save >= self._save_model_secs:
            self._saver.save(sess or self._sess, self._checkpoint_path,
                             global_step=self._global_step)
            self._last_save = now

    def summary_computed(self, sess, summary, global_step=None):
        pass  # summaries flow through tf.summary.FileWriter directly
