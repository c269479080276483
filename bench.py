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

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=20)
    ap.add_argument('--warmup', type=int, default=5)
    ap.add_argument('--batch', type=int, default=256)
    ap.add_argument('--lr', type=float, default=0.1)
    args = ap.parse_args()

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
    imgs_per_sec = args.batch * world * args.steps / elapsed
    if rank == 0:
        out = {
            'metric': 'images/sec',
            'value': round(imgs_per_sec, 2),
            'unit': 'images/sec',
            'n_gpus': world,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(ms_per_step, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': round(imgs_per_sec / 219.0, 3),
            'dtype': 'bf16',
            'data': 'synthetic',
            'config': {
                'model': 'resnet50_v1',
                'global_batch': args.batch * world,
                'seq_len': None,
                'parallelism': 'dp%d' % world,
            },
        }
        print(json.dumps(out))


def _build(args, world):
    import simple_tensorflow_amd as tf
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
