#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 v1 bf16 training, images/sec (whole job),
batch 256/GPU, synthetic 224x224x3 data, random-init weights
(BASELINE.json metric).

Single GPU:   python bench.py --steps 20 --warmup 5
Multi-GPU:    torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
              (one process per GPU over RCCL; rank/env from torchrun)
"""
import argparse
import json
import os
import sys
import time

# Pin this rank's GPU BEFORE any HIP initialization.
_local_rank = int(os.environ.get('LOCAL_RANK', 0))
_world = int(os.environ.get('WORLD_SIZE', 1))
if _world > 1:
    os.environ.setdefault('HIP_VISIBLE_DEVICES', str(_local_rank))
    # RCCL collectives inside hipGraph capture are not yet validated on
    # multi-GPU nodes (replay of the communicator's internal sequencing);
    # run the distributed step eagerly until round 2 proves it out.
    os.environ.setdefault('STF_NO_HIPGRAPH', '1')

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=20)
    ap.add_argument('--warmup', type=int, default=5)
    ap.add_argument('--batch', type=int, default=0)  # 0 = model default
    ap.add_argument('--lr', type=float, default=0.1)
    ap.add_argument('--model', default='resnet50',
                    choices=['resnet50', 'inception3', 'ptb_lstm'])
    ap.add_argument('--seq-len', type=int, default=35)
    args = ap.parse_args()
    if args.batch == 0:
        args.batch = {'resnet50': 256, 'inception3': 128,
                      'ptb_lstm': 20}[args.model]

    import simple_tensorflow_amd as tf
    from simple_tensorflow_amd.models import resnet

    world = _world
    rank = int(os.environ.get('RANK', 0))

    comm = None
    if world > 1:
        from simple_tensorflow_amd.parallel import dist
        comm = dist.init(world, rank)

    loss_t, train_op = _build(args, world)

    sess = tf.Session()
    sess.run(tf.global_variables_initializer())
    if comm is not None:
        comm.broadcast_variables(sess)

    # warmup
    for _ in range(args.warmup):
        sess.run(train_op)
    _sync(sess, loss_t)
    if comm is not None:
        comm.barrier()

    t0 = time.time()
    for _ in range(args.steps):
        sess.run(train_op)
    _sync(sess, loss_t)
    if comm is not None:
        comm.barrier()
    t1 = time.time()

    elapsed = t1 - t0
    if comm is not None:
        elapsed = comm.max_scalar(elapsed)

    ms_per_step = elapsed / args.steps * 1000.0
    if args.model == 'ptb_lstm':
        value = args.batch * args.seq_len * world * args.steps / elapsed
        metric, unit = 'words/sec', 'words/sec'
        model_name, seq_len, vsb = 'ptb_lstm_h1500', args.seq_len, None
    else:
        value = args.batch * world * args.steps / elapsed
        metric, unit = 'images/sec', 'images/sec'
        seq_len = None
        model_name = 'resnet50_v1' if args.model == 'resnet50' \
            else 'inception_v3'
        vsb = round(value / 219.0, 3) if args.model == 'resnet50' else None
    if rank == 0:
        out = {
            'metric': metric,
            'value': round(value, 2),
            'unit': unit,
            'n_gpus': world,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(ms_per_step, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': vsb,
            'dtype': 'bf16',
            'data': 'synthetic',
            'config': {
                'model': model_name,
                'global_batch': args.batch * world,
                'seq_len': seq_len,
                'parallelism': 'dp%d' % world,
            },
        }
        print(json.dumps(out))


def _build(args, world):
    import simple_tensorflow_amd as tf
    if args.model == 'inception3':
        from simple_tensorflow_amd.models import inception
        loss, train_op = inception.build_train_graph(batch=args.batch,
                                                     lr=args.lr)
        if world > 1:
            raise SystemExit('inception bench is single-GPU in round 1')
        return loss, train_op
    if args.model == 'ptb_lstm':
        from simple_tensorflow_amd.models import ptb_lstm
        loss, train_op = ptb_lstm.build_ptb_graph(
            batch=args.batch, seq_len=args.seq_len, hidden=1500,
            vocab=10000, lr=args.lr)
        if world > 1:
            raise SystemExit('use the resnet50 model for multi-GPU scaling')
        return loss, train_op
    from simple_tensorflow_amd.models import resnet
    images, labels = resnet.synthetic_inputs(args.batch)
    loss = resnet.resnet50_loss(images, labels)
    opt = tf.train.MomentumOptimizer(args.lr, 0.9)
    if world > 1:
        from simple_tensorflow_amd.parallel import dist
        opt = dist.DistributedOptimizer(opt, world)
    train_op = opt.minimize(loss)
    return loss, train_op


def _sync(sess, loss_t):
    sess.sync()


if __name__ == '__main__':
    main()
