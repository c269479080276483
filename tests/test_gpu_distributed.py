"""RCCL-backed data-parallel tests on a real MI355X.

The driver box has one GPU, so the real-transport validation runs RCCL with
world=1 (full GPU plumbing: comm stream, pack/unpack kernels, event fencing,
RcclCommSync) and attempts world=2 with both ranks on the same device (RCCL
communicator init + enqueue-order shakeout per VERDICT.md next-steps #1).
The 8-GPU scaling run itself belongs to the driver (bench.py --gpus N).
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

_HELPER = os.path.join(os.path.dirname(__file__), 'helpers', 'dist_worker.py')


def _torchrun(nproc, port, extra_env=None, timeout=240):
    env = dict(os.environ)
    env['MASTER_ADDR'] = '127.0.0.1'
    env['MASTER_PORT'] = str(port)
    env['STF_FORCE_RCCL'] = '1'
    env['STF_NO_HIPGRAPH'] = '1'
    if extra_env:
        env.update(extra_env)
    return subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', str(nproc), '--master-addr', '127.0.0.1',
         '--master-port', str(port), _HELPER],
        capture_output=True, text=True, timeout=timeout, env=env)


def test_rccl_single_rank_bucket():
    """world=1 over a real RCCL communicator: exercises ncclCommInitRank,
    the comm-stream bucket pipeline and the sync join end-to-end on GPU."""
    r = _torchrun(1, 29741)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert 'DIST_OK' in r.stdout


def test_rccl_two_ranks_one_gpu():
    """2 ranks sharing one device. If this RCCL build refuses duplicate
    devices in a communicator, the worker reports it and we skip — the
    capability is then validated only on a real multi-GPU node."""
    r = _torchrun(2, 29743, timeout=300)
    out = r.stdout + r.stderr
    if 'RCCL_DUP_UNSUPPORTED' in out:
        pytest.skip('this RCCL build rejects two ranks on one device')
    assert r.returncode == 0, out[-3000:]
    assert 'DIST_OK' in r.stdout
