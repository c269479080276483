"""PTB LSTM language model (BASELINE.json config 5): seq_len 35, hidden 1500,
2 layers, vocab 10k — bf16 compute with f32 master weights, statically
unrolled (the TF-1.0 tutorial execution style), words/sec metric."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import rnn_cell_impl, variables


def build_ptb_graph(batch=20, seq_len=35, hidden=1500, vocab=10000,
                    layers=2, lr=1.0, seed=1234):
    rng = np.random.RandomState(seed)
    data = rng.randint(0, vocab, (batch, seq_len + 1))
    inputs_np = data[:, :-1].astype(np.int64)
    targets_np = data[:, 1:].astype(np.int64)

    inputs = tf.constant(inputs_np)
    targets = tf.constant(targets_np)

    embedding = variables.Variable(
        tf.random_uniform([vocab, hidden], -0.05, 0.05), name='embedding')
    emb16 = tf.cast(embedding.ref(), tf.bfloat16)
    emb = tf.gather(emb16, inputs)            # [batch, seq, hidden]
    xs = tf.unstack(emb, num=seq_len, axis=1)  # seq * [batch, hidden]

    cell = rnn_cell_impl.MultiRNNCell(
        [rnn_cell_impl.LSTMBlockCell(hidden) for _ in range(layers)])
    outputs, _ = rnn_cell_impl.static_rnn(cell, xs, dtype=tf.bfloat16)

    output = tf.concat(outputs, 0)            # [seq*batch, hidden]
    softmax_w = variables.Variable(
        tf.random_uniform([hidden, vocab], -0.05, 0.05), name='softmax_w')
    softmax_b = variables.Variable(tf.zeros([vocab]), name='softmax_b')
    logits16 = tf.matmul(output, tf.cast(softmax_w.ref(), tf.bfloat16))
    logits = tf.cast(logits16, tf.float32) + softmax_b.ref()

    # targets are time-major after the concat: [seq, batch] flattened
    tgt = tf.reshape(tf.transpose(targets, [1, 0]), [seq_len * batch])
    loss_vec = tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=tgt, logits=logits)
    loss = tf.reduce_mean(loss_vec)

    opt = tf.train.GradientDescentOptimizer(lr)
    gvs = opt.compute_gradients(loss)
    clipped, _ = tf.clip_by_global_norm([g for g, _ in gvs], 5.0)
    train_op = opt.apply_gradients(
        [(c, v) for c, (_, v) in zip(clipped, gvs)])
    return loss, train_op
