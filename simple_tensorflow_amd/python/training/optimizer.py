"""Optimizer base + the standard TF-1.0 optimizers.

Analog of reference python/training/optimizer.py (minimize:277,
compute_gradients:327, apply_gradients:395) + gradient_descent.py,
momentum.py, adam.py, rmsprop.py, adagrad.py, adadelta.py. Slot variables are
created next to the trainable variables (slot_creator.py analog inlined).
"""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor
from simple_tensorflow_amd.python.ops import (array_ops, control_flow_ops,
                                              gradients_impl, math_ops,
                                              state_ops, variables)


class Optimizer(object):
    GATE_NONE = 0
    GATE_OP = 1
    GATE_GRAPH = 2

    def __init__(self, use_locking=False, name=None):
        self._name = name or type(self).__name__
        self._use_locking = use_locking
        self._slots = {}  # slot_name -> {var_ref_name: Variable}

    def minimize(self, loss, global_step=None, var_list=None,
                 gate_gradients=GATE_OP, aggregation_method=None,
                 colocate_gradients_with_ops=False, name=None,
                 grad_loss=None):
        grads_and_vars = self.compute_gradients(loss, var_list=var_list,
                                                grad_loss=grad_loss)
        return self.apply_gradients(grads_and_vars, global_step=global_step,
                                    name=name)

    def compute_gradients(self, loss, var_list=None, gate_gradients=GATE_OP,
                          aggregation_method=None,
                          colocate_gradients_with_ops=False, grad_loss=None):
        if var_list is None:
            var_list = variables.trainable_variables()
        grads = gradients_impl.gradients(loss, var_list, grad_ys=grad_loss)
        return list(zip(grads, var_list))

    def apply_gradients(self, grads_and_vars, global_step=None, name=None):
        g = ops.get_default_graph()
        with g.name_scope(name or self._name):
            self._create_slots([v for _, v in grads_and_vars])
            self._prepare()
            update_ops = []
            for grad, var in grads_and_vars:
                if grad is None:
                    continue
                update_ops.append(self._apply_dense(grad, var))
            if global_step is not None:
                with ops.control_dependencies(
                        [u.op if isinstance(u, ops.Tensor) else u
                         for u in update_ops]):
                    apply_updates = state_ops.assign_add(
                        global_step._as_graph_element()
                        if hasattr(global_step, '_as_graph_element')
                        else global_step, 1).op
            else:
                apply_updates = control_flow_ops.group(*update_ops)
            return apply_updates

    # ---- slots ----
    def _zeros_slot(self, var, slot_name, op_name):
        slots = self._slots.setdefault(slot_name, {})
        key = var.name if hasattr(var, 'name') else str(id(var))
        if key not in slots:
            shape = var.get_shape().as_list() if hasattr(var, 'get_shape') \
                else list(var._shape)
            slots[key] = variables.Variable(
                array_ops.zeros(shape, var.dtype), trainable=False,
                name=op_name + '/' + slot_name)
        return slots[key]

    def get_slot(self, var, name):
        key = var.name if hasattr(var, 'name') else str(id(var))
        return self._slots.get(name, {}).get(key)

    def get_slot_names(self):
        return sorted(self._slots)

    def _create_slots(self, var_list):
        pass

    def _prepare(self):
        pass

    def _apply_dense(self, grad, var):
        raise NotImplementedError

    def _var_ref(self, var):
        return var._as_graph_element() if hasattr(var, '_as_graph_element') \
            else var


class GradientDescentOptimizer(Optimizer):
    def __init__(self, learning_rate, use_locking=False,
                 name='GradientDescent'):
        super().__init__(use_locking, name)
        self._learning_rate = learning_rate

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        return apply_op('ApplyGradientDescent', ref,
                        convert_to_tensor(self._learning_rate,
                                          dtype=ref.dtype),
                        grad, use_locking=self._use_locking)


class MomentumOptimizer(Optimizer):
    def __init__(self, learning_rate, momentum, use_locking=False,
                 name='Momentum', use_nesterov=False):
        super().__init__(use_locking, name)
        self._learning_rate = learning_rate
        self._momentum = momentum
        self._use_nesterov = use_nesterov

    def _create_slots(self, var_list):
        for v in var_list:
            self._zeros_slot(v, 'momentum', self._name)

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        mom = self.get_slot(var, 'momentum')
        return apply_op('ApplyMomentum', ref, self._var_ref(mom),
                        convert_to_tensor(self._learning_rate, dtype=ref.dtype),
                        grad,
                        convert_to_tensor(self._momentum, dtype=ref.dtype),
                        use_locking=self._use_locking,
                        use_nesterov=self._use_nesterov)


class AdamOptimizer(Optimizer):
    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, use_locking=False, name='Adam'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._beta1 = beta1
        self._beta2 = beta2
        self._epsilon = epsilon
        self._beta1_power = None
        self._beta2_power = None

    def _create_slots(self, var_list):
        if self._beta1_power is None:
            self._beta1_power = variables.Variable(self._beta1,
                                                   trainable=False,
                                                   name='beta1_power')
            self._beta2_power = variables.Variable(self._beta2,
                                                   trainable=False,
                                                   name='beta2_power')
        for v in var_list:
            self._zeros_slot(v, 'm', self._name)
            self._zeros_slot(v, 'v', self._name)

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        m = self.get_slot(var, 'm')
        v = self.get_slot(var, 'v')
        return apply_op(
            'ApplyAdam', ref, self._var_ref(m), self._var_ref(v),
            self._beta1_power.value(), self._beta2_power.value(),
            convert_to_tensor(self._lr, dtype=ref.dtype),
            convert_to_tensor(self._beta1, dtype=ref.dtype),
            convert_to_tensor(self._beta2, dtype=ref.dtype),
            convert_to_tensor(self._epsilon, dtype=ref.dtype), grad,
            use_locking=self._use_locking)

    def apply_gradients(self, grads_and_vars, global_step=None, name=None):
        update = super().apply_gradients(grads_and_vars, global_step, name)
        with ops.control_dependencies([update]):
            b1 = state_ops.assign(
                self._beta1_power._as_graph_element(),
                self._beta1_power.value() * self._beta1)
            b2 = state_ops.assign(
                self._beta2_power._as_graph_element(),
                self._beta2_power.value() * self._beta2)
        return control_flow_ops.group(update, b1.op, b2.op)


class RMSPropOptimizer(Optimizer):
    def __init__(self, learning_rate, decay=0.9, momentum=0.0, epsilon=1e-10,
                 use_locking=False, name='RMSProp'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._decay = decay
        self._momentum = momentum
        self._epsilon = epsilon

    def _create_slots(self, var_list):
        for v in var_list:
            self._zeros_slot(v, 'rms', self._name)
            self._zeros_slot(v, 'momentum', self._name)

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        ms = self.get_slot(var, 'rms')
        mom = self.get_slot(var, 'momentum')
        return apply_op('ApplyRMSProp', ref, self._var_ref(ms),
                        self._var_ref(mom),
                        convert_to_tensor(self._lr, dtype=ref.dtype),
                        convert_to_tensor(self._decay, dtype=ref.dtype),
                        convert_to_tensor(self._momentum, dtype=ref.dtype),
                        convert_to_tensor(self._epsilon, dtype=ref.dtype),
                        grad, use_locking=self._use_locking)


class AdagradOptimizer(Optimizer):
    def __init__(self, learning_rate, initial_accumulator_value=0.1,
                 use_locking=False, name='Adagrad'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._init_acc = initial_accumulator_value

    def _create_slots(self, var_list):
        for v in var_list:
            slots = self._slots.setdefault('accumulator', {})
            key = v.name
            if key not in slots:
                shape = v.get_shape().as_list()
                slots[key] = variables.Variable(
                    array_ops.ones(shape, v.dtype) * self._init_acc,
                    trainable=False, name=self._name + '/accumulator')

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        acc = self.get_slot(var, 'accumulator')
        return apply_op('ApplyAdagrad', ref, self._var_ref(acc),
                        convert_to_tensor(self._lr, dtype=ref.dtype), grad,
                        use_locking=self._use_locking)


class FtrlOptimizer(Optimizer):
    """FTRL-proximal (reference ftrl.py + ApplyFtrl kernel)."""

    def __init__(self, learning_rate, learning_rate_power=-0.5,
                 initial_accumulator_value=0.1,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, use_locking=False,
                 name='Ftrl'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._lr_power = learning_rate_power
        self._init_acc = initial_accumulator_value
        self._l1 = l1_regularization_strength
        self._l2 = l2_regularization_strength

    def _create_slots(self, var_list):
        for v in var_list:
            slots = self._slots.setdefault('accum', {})
            if v.name not in slots:
                shape = v.get_shape().as_list()
                slots[v.name] = variables.Variable(
                    array_ops.ones(shape, v.dtype) * self._init_acc,
                    trainable=False, name=self._name + '/accum')
            self._zeros_slot(v, 'linear', self._name)

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        accum = self.get_slot(var, 'accum')
        linear = self.get_slot(var, 'linear')
        return apply_op(
            'ApplyFtrl', ref, self._var_ref(accum), self._var_ref(linear),
            grad, convert_to_tensor(self._lr, dtype=ref.dtype),
            convert_to_tensor(self._l1, dtype=ref.dtype),
            convert_to_tensor(self._l2, dtype=ref.dtype),
            convert_to_tensor(self._lr_power, dtype=ref.dtype),
            use_locking=self._use_locking)


class ProximalGradientDescentOptimizer(Optimizer):
    def __init__(self, learning_rate, l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, use_locking=False,
                 name='ProximalGradientDescent'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._l1 = l1_regularization_strength
        self._l2 = l2_regularization_strength

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        return apply_op(
            'ApplyProximalGradientDescent', ref,
            convert_to_tensor(self._lr, dtype=ref.dtype),
            convert_to_tensor(self._l1, dtype=ref.dtype),
            convert_to_tensor(self._l2, dtype=ref.dtype), grad,
            use_locking=self._use_locking)


class AdadeltaOptimizer(Optimizer):
    def __init__(self, learning_rate=0.001, rho=0.95, epsilon=1e-8,
                 use_locking=False, name='Adadelta'):
        super().__init__(use_locking, name)
        self._lr = learning_rate
        self._rho = rho
        self._epsilon = epsilon

    def _create_slots(self, var_list):
        for v in var_list:
            self._zeros_slot(v, 'accum', self._name)
            self._zeros_slot(v, 'accum_update', self._name)

    def _apply_dense(self, grad, var):
        ref = self._var_ref(var)
        a = self.get_slot(var, 'accum')
        au = self.get_slot(var, 'accum_update')
        return apply_op('ApplyAdadelta', ref, self._var_ref(a),
                        self._var_ref(au),
                        convert_to_tensor(self._lr, dtype=ref.dtype),
                        convert_to_tensor(self._rho, dtype=ref.dtype),
                        convert_to_tensor(self._epsilon, dtype=ref.dtype),
                        grad, use_locking=self._use_locking)
