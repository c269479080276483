"""MonitoredSession stack: Scaffold, SessionManager, session creators,
recoverable/coordinated/hooked sessions, SessionRunHook API and the basic
hooks (analogs of reference python/training/monitored_session.py:554,
session_manager.py, basic_session_run_hooks.py)."""
import os
import time

import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import errors, ops
from simple_tensorflow_amd.python.training import coordinator as coord_lib
from simple_tensorflow_amd.python.training import saver as saver_lib


# ---------------------------------------------------------------------------
# Hook API
# ---------------------------------------------------------------------------
class SessionRunContext(object):
    def __init__(self, original_args, session):
        self.original_args = original_args
        self.session = session
        self._stop = False

    def request_stop(self):
        self._stop = True

    @property
    def stop_requested(self):
        return self._stop


class SessionRunArgs(object):
    def __init__(self, fetches, feed_dict=None):
        self.fetches = fetches
        self.feed_dict = feed_dict


class SessionRunValues(object):
    def __init__(self, results):
        self.results = results


class SessionRunHook(object):
    def begin(self):
        pass

    def after_create_session(self, session, coord):
        pass

    def before_run(self, run_context):
        return None

    def after_run(self, run_context, run_values):
        pass

    def end(self, session):
        pass


# ---------------------------------------------------------------------------
# Scaffold + SessionManager
# ---------------------------------------------------------------------------
class Scaffold(object):
    def __init__(self, init_op=None, ready_op=None, local_init_op=None,
                 saver=None, summary_op=None, init_fn=None):
        self.init_op = init_op
        self.local_init_op = local_init_op
        self.saver = saver
        self.summary_op = summary_op
        self.init_fn = init_fn
        self._finalized = False

    def finalize(self):
        if self._finalized:
            return self
        if self.init_op is None:
            self.init_op = tf.global_variables_initializer()
        if self.saver is None and tf.global_variables():
            self.saver = saver_lib.Saver()
        self._finalized = True
        return self


class SessionManager(object):
    """prepare/recover/wait (reference session_manager.py:192,283,347)."""

    def __init__(self, local_init_op=None, ready_op=None, graph=None,
                 recovery_wait_secs=1):
        self._local_init_op = local_init_op
        self._graph = graph
        self._recovery_wait_secs = recovery_wait_secs

    def prepare_session(self, master='', init_op=None, saver=None,
                        checkpoint_dir=None, checkpoint_filename_with_path=None,
                        wait_for_checkpoint=False, max_wait_secs=7200,
                        config=None, init_feed_dict=None, init_fn=None):
        sess, restored = self.recover_session(
            master, saver, checkpoint_dir=checkpoint_dir,
            checkpoint_filename_with_path=checkpoint_filename_with_path,
            config=config)
        if not restored:
            if init_op is None and init_fn is None:
                raise RuntimeError('no init_op/init_fn and no checkpoint')
            if init_op is not None:
                sess.run(init_op, feed_dict=init_feed_dict)
            if init_fn is not None:
                init_fn(sess)
        if self._local_init_op is not None:
            sess.run(self._local_init_op)
        return sess

    def recover_session(self, master='', saver=None, checkpoint_dir=None,
                        checkpoint_filename_with_path=None, config=None):
        from simple_tensorflow_amd.python.client import session as sess_lib
        sess = sess_lib.Session(master, graph=self._graph, config=config)
        ckpt_path = checkpoint_filename_with_path
        if ckpt_path is None and checkpoint_dir:
            ckpt_path = saver_lib.latest_checkpoint(checkpoint_dir)
        if saver is None or ckpt_path is None or \
                not saver_lib.checkpoint_exists(ckpt_path):
            return sess, False
        saver.restore(sess, ckpt_path)
        return sess, True

    def wait_for_session(self, master='', config=None, max_wait_secs=None):
        from simple_tensorflow_amd.python.client import session as sess_lib
        return sess_lib.Session(master, graph=self._graph, config=config)


# ---------------------------------------------------------------------------
# Session creators + monitored sessions
# ---------------------------------------------------------------------------
class ChiefSessionCreator(object):
    def __init__(self, scaffold=None, master='', config=None,
                 checkpoint_dir=None, checkpoint_filename_with_path=None):
        self._scaffold = (scaffold or Scaffold()).finalize()
        self._master = master
        self._config = config
        self._checkpoint_dir = checkpoint_dir
        self._checkpoint_path = checkpoint_filename_with_path

    def create_session(self):
        sm = SessionManager()
        return sm.prepare_session(
            self._master, init_op=self._scaffold.init_op,
            saver=self._scaffold.saver, checkpoint_dir=self._checkpoint_dir,
            checkpoint_filename_with_path=self._checkpoint_path,
            config=self._config, init_fn=self._scaffold.init_fn)

    @property
    def scaffold(self):
        return self._scaffold


class WorkerSessionCreator(object):
    def __init__(self, scaffold=None, master='', config=None):
        self._scaffold = (scaffold or Scaffold()).finalize()
        self._master = master
        self._config = config

    def create_session(self):
        return SessionManager().wait_for_session(self._master, self._config)


class _RecoverableSession(object):
    """Retries `run` on AbortedError/UnavailableError by recreating the
    underlying session through the creator (reference
    monitored_session.py:778 _RecoverableSession): a preempted worker or a
    restarted parameter server surfaces as Aborted/Unavailable; training
    resumes from the last checkpoint the creator restores."""

    def __init__(self, creator, on_recreate=None, max_retries=None):
        self._creator = creator
        self._on_recreate = on_recreate
        self._max_retries = max_retries
        self._sess = self._create()

    def _create(self):
        while True:
            try:
                return self._creator.create_session()
            except (errors.AbortedError, errors.UnavailableError):
                time.sleep(1.0)

    def run(self, *a, **kw):
        retries = 0
        while True:
            try:
                return self._sess.run(*a, **kw)
            except (errors.AbortedError, errors.UnavailableError) as ex:
                retries += 1
                if self._max_retries is not None and retries > self._max_retries:
                    raise
                tf.logging.warning(
                    'Session run aborted (%s); recreating session' % ex)
                try:
                    self._sess.close()
                except Exception:  # noqa: BLE001
                    pass
                self._sess = self._create()
                if self._on_recreate:
                    self._on_recreate(self._sess)

    @property
    def raw_session(self):
        return self._sess

    def __getattr__(self, name):
        return getattr(self._sess, name)


class _CoordinatedSession(object):
    """Runs queue-runner threads under a coordinator (reference
    monitored_session.py:829)."""

    def __init__(self, sess, coord):
        self._sess = sess
        self._coord = coord

    def run(self, *a, **kw):
        try:
            return self._sess.run(*a, **kw)
        except Exception as ex:  # noqa: BLE001
            self._coord.request_stop(ex)
            raise

    def __getattr__(self, name):
        return getattr(self._sess, name)


class MonitoredSession(object):
    def __init__(self, session_creator=None, hooks=None,
                 stop_grace_period_secs=120, recoverable=True):
        self._hooks = list(hooks or [])
        self._creator = session_creator or ChiefSessionCreator()
        for h in self._hooks:
            h.begin()
        self._coord = coord_lib.Coordinator()

        def _on_recreate(sess):
            # A recreated session needs its queue runners restarted and
            # hooks re-pointed at the live session.
            self._raw_sess = sess
            coord_lib.start_queue_runners(sess, coord=self._coord)
            for h in self._hooks:
                h.after_create_session(sess, self._coord)

        if recoverable:
            self._sess = _RecoverableSession(self._creator,
                                             on_recreate=_on_recreate)
            self._raw_sess = self._sess.raw_session
        else:
            self._sess = self._creator.create_session()
            self._raw_sess = self._sess
        coord_lib.start_queue_runners(self._raw_sess, coord=self._coord)
        for h in self._hooks:
            h.after_create_session(self._raw_sess, self._coord)
        self._csess = _CoordinatedSession(self._sess, self._coord)
        self._should_stop = False

    def run(self, fetches, feed_dict=None, options=None, run_metadata=None):
        args = SessionRunArgs(fetches, feed_dict)
        ctx = SessionRunContext(args, self._raw_sess)
        hook_fetches = [fetches]
        hook_feeds = dict(feed_dict or {})
        per_hook = []
        for h in self._hooks:
            r = h.before_run(ctx)
            if r is not None and r.fetches is not None:
                per_hook.append(len(hook_fetches))
                hook_fetches.append(r.fetches)
                if r.feed_dict:
                    hook_feeds.update(r.feed_dict)
            else:
                per_hook.append(None)
        results = self._csess.run(hook_fetches, feed_dict=hook_feeds or None)
        for h, idx in zip(self._hooks, per_hook):
            vals = SessionRunValues(results[idx] if idx is not None else None)
            h.after_run(ctx, vals)
        if ctx.stop_requested:
            self._should_stop = True
        return results[0]

    def should_stop(self):
        return self._should_stop or self._coord.should_stop()

    def close(self):
        try:
            for h in self._hooks:
                h.end(self._raw_sess)
        finally:
            self._coord.request_stop()
            try:
                self._coord.join(stop_grace_period_secs=5)
            except Exception:  # noqa: BLE001
                pass

    @property
    def graph(self):
        return self._raw_sess.graph

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False


def MonitoredTrainingSession(master='', is_chief=True, checkpoint_dir=None,
                             scaffold=None, hooks=None, chief_only_hooks=None,
                             save_checkpoint_secs=600,
                             save_summaries_steps=100, config=None,
                             stop_grace_period_secs=120):
    scaffold = (scaffold or Scaffold()).finalize()
    all_hooks = list(hooks or [])
    if is_chief:
        creator = ChiefSessionCreator(scaffold=scaffold, master=master,
                                      config=config,
                                      checkpoint_dir=checkpoint_dir)
        all_hooks.extend(chief_only_hooks or [])
        if checkpoint_dir and save_checkpoint_secs:
            all_hooks.append(CheckpointSaverHook(
                checkpoint_dir, save_secs=save_checkpoint_secs,
                saver=scaffold.saver))
    else:
        creator = WorkerSessionCreator(scaffold=scaffold, master=master,
                                       config=config)
    return MonitoredSession(session_creator=creator, hooks=all_hooks)


# ---------------------------------------------------------------------------
# basic hooks (reference basic_session_run_hooks.py)
# ---------------------------------------------------------------------------
class StopAtStepHook(SessionRunHook):
    def __init__(self, num_steps=None, last_step=None):
        self._num_steps = num_steps
        self._last_step = last_step
        self._step = 0

    def before_run(self, run_context):
        return None

    def after_run(self, run_context, run_values):
        self._step += 1
        if self._num_steps is not None and self._step >= self._num_steps:
            run_context.request_stop()
        if self._last_step is not None and self._step >= self._last_step:
            run_context.request_stop()


class CheckpointSaverHook(SessionRunHook):
    def __init__(self, checkpoint_dir, save_secs=None, save_steps=None,
                 saver=None, checkpoint_basename='model.ckpt'):
        self._dir = checkpoint_dir
        self._save_secs = save_secs
        self._save_steps = save_steps
        self._saver = saver
        self._path = os.path.join(checkpoint_dir, checkpoint_basename)
        self._last_save = time.time()
        self._step = 0

    def after_create_session(self, session, coord):
        os.makedirs(self._dir, exist_ok=True)
        if self._saver is None:
            self._saver = saver_lib.Saver()
        self._saver.save(session, self._path, global_step=0)

    def after_run(self, run_context, run_values):
        self._step += 1
        due = False
        if self._save_steps and self._step % self._save_steps == 0:
            due = True
        if self._save_secs and time.time() - self._last_save > self._save_secs:
            due = True
        if due:
            self._saver.save(run_context.session, self._path,
                             global_step=self._step)
            self._last_save = time.time()

    def end(self, session):
        if self._saver is not None:
            self._saver.save(session, self._path, global_step=self._step)


class LoggingTensorHook(SessionRunHook):
    def __init__(self, tensors, every_n_iter=100, formatter=None):
        self._tensors = tensors
        self._every_n = every_n_iter
        self._step = 0
        self._formatter = formatter

    def before_run(self, run_context):
        if self._step % self._every_n == 0:
            return SessionRunArgs(self._tensors)
        return None

    def after_run(self, run_context, run_values):
        if self._step % self._every_n == 0 and run_values.results is not None:
            if self._formatter:
                print(self._formatter(run_values.results))
            else:
                print('step %d: %r' % (self._step, run_values.results))
        self._step += 1


class StepCounterHook(SessionRunHook):
    def __init__(self, every_n_steps=100, output_dir=None, summary_writer=None):
        self._every_n = every_n_steps
        self._step = 0
        self._t0 = time.time()

    def after_run(self, run_context, run_values):
        self._step += 1
        if self._step % self._every_n == 0:
            dt = time.time() - self._t0
            print('%.3f steps/sec' % (self._every_n / dt))
            self._t0 = time.time()


class NanTensorHook(SessionRunHook):
    def __init__(self, loss_tensor, fail_on_nan_loss=True):
        self._loss = loss_tensor
        self._fail = fail_on_nan_loss

    def before_run(self, run_context):
        return SessionRunArgs(self._loss)

    def after_run(self, run_context, run_values):
        if run_values.results is not None and \
                not np.isfinite(run_values.results):
            if self._fail:
                raise errors.InvalidArgumentError('NaN loss', 3)
            run_context.request_stop()


class SummarySaverHook(SessionRunHook):
    def __init__(self, save_steps=100, output_dir=None, summary_writer=None,
                 summary_op=None):
        self._save_steps = save_steps
        self._summary_op = summary_op
        self._writer = summary_writer
        self._output_dir = output_dir
        self._step = 0

    def begin(self):
        if self._writer is None and self._output_dir:
            from simple_tensorflow_amd.python.summary import writer
            self._writer = writer.FileWriter(self._output_dir)

    def before_run(self, run_context):
        if self._summary_op is not None and \
                self._step % self._save_steps == 0:
            return SessionRunArgs(self._summary_op)
        return None

    def after_run(self, run_context, run_values):
        if run_values.results is not None and self._writer is not None:
            self._writer.add_summary(run_values.results, self._step)
        self._step += 1
