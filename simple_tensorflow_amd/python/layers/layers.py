"""tf.layers: Layer base + Dense/Conv2D/BatchNormalization/Pooling/Dropout/
Flatten (analog of reference python/layers/{base,core,convolutional,
normalization,pooling}.py)."""
import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op
from simple_tensorflow_amd.python.ops import (array_ops, init_ops, math_ops,
                                              nn_ops, variable_scope,
                                              variables)


class Layer(object):
    """Variable-tracking layer base (reference python/layers/base.py)."""

    def __init__(self, trainable=True, name=None, **kw):
        self.trainable = trainable
        self.name = name or type(self).__name__.lower()
        self.built = False
        self._trainable_weights = []
        self._non_trainable_weights = []

    def add_variable(self, name, shape, dtype=dtypes.float32,
                     initializer=None, trainable=True):
        with variable_scope.variable_scope(self.name):
            v = variable_scope.get_variable(name, shape, dtype=dtype,
                                            initializer=initializer,
                                            trainable=trainable and
                                            self.trainable)
        (self._trainable_weights if trainable
         else self._non_trainable_weights).append(v)
        return v

    add_weight = add_variable

    @property
    def trainable_weights(self):
        return list(self._trainable_weights)

    @property
    def weights(self):
        return self._trainable_weights + self._non_trainable_weights

    def build(self, input_shape):
        self.built = True

    def call(self, inputs, **kw):
        raise NotImplementedError

    def __call__(self, inputs, **kw):
        if not self.built:
            self.build(inputs.shape if hasattr(inputs, 'shape') else None)
            self.built = True
        return self.call(inputs, **kw)

    apply = __call__


class Dense(Layer):
    def __init__(self, units, activation=None, use_bias=True,
                 kernel_initializer=None, bias_initializer=None, name=None,
                 **kw):
        super().__init__(name=name or 'dense', **kw)
        self.units = units
        self.activation = activation
        self.use_bias = use_bias
        self.kernel_initializer = kernel_initializer or \
            init_ops.glorot_uniform_initializer()
        self.bias_initializer = bias_initializer or \
            init_ops.zeros_initializer()
        self.kernel = None
        self.bias = None

    def build(self, input_shape):
        in_dim = input_shape[-1]
        self.kernel = self.add_variable('kernel', [in_dim, self.units],
                                        initializer=self.kernel_initializer)
        if self.use_bias:
            self.bias = self.add_variable('bias', [self.units],
                                          initializer=self.bias_initializer)
        super().build(input_shape)

    def call(self, inputs):
        y = math_ops.matmul(inputs, self.kernel.ref())
        if self.use_bias:
            y = nn_ops.bias_add(y, self.bias.ref())
        if self.activation is not None:
            y = self.activation(y)
        return y


class Conv2D(Layer):
    def __init__(self, filters, kernel_size, strides=(1, 1), padding='valid',
                 activation=None, use_bias=True, kernel_initializer=None,
                 bias_initializer=None, data_format='channels_last',
                 name=None, **kw):
        super().__init__(name=name or 'conv2d', **kw)
        self.filters = filters
        if isinstance(kernel_size, int):
            kernel_size = (kernel_size, kernel_size)
        if isinstance(strides, int):
            strides = (strides, strides)
        self.kernel_size = kernel_size
        self.strides = strides
        self.padding = padding.upper()
        self.activation = activation
        self.use_bias = use_bias
        self.kernel_initializer = kernel_initializer or \
            init_ops.glorot_uniform_initializer()
        self.bias_initializer = bias_initializer or \
            init_ops.zeros_initializer()

    def build(self, input_shape):
        cin = input_shape[-1]
        self.kernel = self.add_variable(
            'kernel', list(self.kernel_size) + [cin, self.filters],
            initializer=self.kernel_initializer)
        if self.use_bias:
            self.bias = self.add_variable('bias', [self.filters],
                                          initializer=self.bias_initializer)
        super().build(input_shape)

    def call(self, inputs):
        k = self.kernel.ref()
        if inputs.dtype == dtypes.bfloat16:
            k = math_ops.cast(k, dtypes.bfloat16)
        y = nn_ops.conv2d(inputs, k,
                          [1, self.strides[0], self.strides[1], 1],
                          self.padding)
        if self.use_bias:
            b = self.bias.ref()
            if y.dtype == dtypes.bfloat16:
                b = math_ops.cast(b, dtypes.bfloat16)
            y = nn_ops.bias_add(y, b)
        if self.activation is not None:
            y = self.activation(y)
        return y


class BatchNormalization(Layer):
    def __init__(self, axis=-1, momentum=0.99, epsilon=1e-3, center=True,
                 scale=True, fused=True, name=None, **kw):
        super().__init__(name=name or 'batch_normalization', **kw)
        self.momentum = momentum
        self.epsilon = epsilon
        self.center = center
        self.scale = scale

    def build(self, input_shape):
        c = input_shape[-1]
        self.gamma = self.add_variable('gamma', [c],
                                       initializer=init_ops.ones_initializer())
        self.beta = self.add_variable('beta', [c],
                                      initializer=init_ops.zeros_initializer())
        self.moving_mean = self.add_variable(
            'moving_mean', [c], initializer=init_ops.zeros_initializer(),
            trainable=False)
        self.moving_variance = self.add_variable(
            'moving_variance', [c], initializer=init_ops.ones_initializer(),
            trainable=False)
        super().build(input_shape)

    def call(self, inputs, training=False):
        from simple_tensorflow_amd.python.ops import state_ops
        g = ops.get_default_graph()
        if training:
            if inputs.dtype == dtypes.bfloat16:
                y, mean, var, _ = apply_op('BatchNormMi', inputs,
                                           self.gamma.ref(), self.beta.ref(),
                                           epsilon=self.epsilon)
                y.set_shape(inputs._shape)
            else:
                y, mean, var = nn_ops.fused_batch_norm(
                    inputs, self.gamma.ref(), self.beta.ref(),
                    epsilon=self.epsilon, is_training=True)
            upd_m = state_ops.assign_sub(
                self.moving_mean._as_graph_element(),
                (self.moving_mean.value() - mean) * (1.0 - self.momentum))
            upd_v = state_ops.assign_sub(
                self.moving_variance._as_graph_element(),
                (self.moving_variance.value() - var) * (1.0 - self.momentum))
            g.add_to_collection(ops.GraphKeys.UPDATE_OPS, upd_m.op)
            g.add_to_collection(ops.GraphKeys.UPDATE_OPS, upd_v.op)
            return y
        # inference: composed normalization on the moving stats
        x = inputs
        if x.dtype == dtypes.bfloat16:
            x = math_ops.cast(x, dtypes.float32)
        y = nn_ops.batch_normalization(
            x, self.moving_mean.value(), self.moving_variance.value(),
            self.beta.value(), self.gamma.value(), self.epsilon)
        if inputs.dtype == dtypes.bfloat16:
            y = math_ops.cast(y, dtypes.bfloat16)
        return y


class MaxPooling2D(Layer):
    def __init__(self, pool_size, strides, padding='valid', name=None, **kw):
        super().__init__(name=name or 'max_pooling2d', **kw)
        if isinstance(pool_size, int):
            pool_size = (pool_size, pool_size)
        if isinstance(strides, int):
            strides = (strides, strides)
        self.pool_size = pool_size
        self.strides = strides
        self.padding = padding.upper()

    def call(self, inputs):
        return nn_ops.max_pool(inputs,
                               [1, self.pool_size[0], self.pool_size[1], 1],
                               [1, self.strides[0], self.strides[1], 1],
                               self.padding)


class AveragePooling2D(MaxPooling2D):
    def call(self, inputs):
        return nn_ops.avg_pool(inputs,
                               [1, self.pool_size[0], self.pool_size[1], 1],
                               [1, self.strides[0], self.strides[1], 1],
                               self.padding)


class Dropout(Layer):
    def __init__(self, rate=0.5, seed=None, name=None, **kw):
        super().__init__(name=name or 'dropout', **kw)
        self.rate = rate
        self.seed = seed

    def call(self, inputs, training=False):
        if not training:
            return inputs
        return nn_ops.dropout(inputs, 1.0 - self.rate, seed=self.seed)


class Flatten(Layer):
    def call(self, inputs):
        dims = inputs._shape
        n = 1
        for d in dims[1:]:
            n *= d
        return array_ops.reshape(inputs, [-1, n])


# functional wrappers (tf.layers.dense style)
def dense(inputs, units, activation=None, use_bias=True, name=None, **kw):
    return Dense(units, activation, use_bias, name=name, **kw)(inputs)


def conv2d(inputs, filters, kernel_size, strides=(1, 1), padding='valid',
           activation=None, use_bias=True, name=None, **kw):
    return Conv2D(filters, kernel_size, strides, padding, activation,
                  use_bias, name=name, **kw)(inputs)


def batch_normalization(inputs, axis=-1, momentum=0.99, epsilon=1e-3,
                        training=False, name=None, **kw):
    return BatchNormalization(axis, momentum, epsilon, name=name,
                              **kw)(inputs, training=training)


def max_pooling2d(inputs, pool_size, strides, padding='valid', name=None):
    return MaxPooling2D(pool_size, strides, padding, name=name)(inputs)


def average_pooling2d(inputs, pool_size, strides, padding='valid', name=None):
    return AveragePooling2D(pool_size, strides, padding, name=name)(inputs)


def dropout(inputs, rate=0.5, training=False, seed=None, name=None):
    return Dropout(rate, seed, name=name)(inputs, training=training)


def flatten(inputs, name=None):
    return Flatten(name=name)(inputs)
