"""CLI-contract tests: the reference's main.py / export.py / evaluate.py
entry points run end to end as subprocesses (reference main.py:9-17,
export.py:99-130, evaluate.py:245-290). Tiny synthetic configs on CPU."""

import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(args, cwd):
    env = dict(os.environ)
    env['PYTHONPATH'] = REPO
    return subprocess.run(
        [sys.executable] + args, cwd=cwd, env=env,
        capture_output=True, text=True, timeout=900)


TRAIN_FLAGS = [
    '--train-flag', '--synthetic', '--synthetic-size', '4',
    '--gpu-no', '-1', '--batch-size', '2', '--end-epoch', '1',
    '--num-workers', '0', '--imsize', '64',
    '--num-stack', '1', '--hourglass-inch', '16', '--print-interval', '1',
    '--random-seed', '7',
]


def test_main_train_then_eval(tmp_path):
    save = str(tmp_path / 'run')
    r = run_cli([os.path.join(REPO, 'main.py')] + TRAIN_FLAGS +
                ['--save-path', save], cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    ckpt = os.path.join(save, 'check_point_1.pth')
    assert os.path.isfile(ckpt), os.listdir(save)
    assert os.path.isfile(os.path.join(save, 'argument.pickle'))
    assert os.path.isfile(os.path.join(save, 'argument.txt'))
    # checkpoint keeps the reference dict format
    d = torch.load(ckpt, map_location='cpu', weights_only=False)
    for k in ('epoch', 'state_dict', 'optimizer', 'scheduler', 'scaler',
              'loss_log'):
        assert k in d, k

    # eval mode restores the architecture flags from argument.pickle
    r2 = run_cli([os.path.join(REPO, 'main.py'),
                  '--synthetic', '--synthetic-size', '2', '--gpu-no', '-1',
                  '--num-workers', '0', '--imsize', '64',
                  '--save-path', save, '--model-load', ckpt],
                 cwd=str(tmp_path))
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert os.path.isfile(os.path.join(save, 'prediction_results.pickle'))
    assert os.path.isdir(os.path.join(save, 'txt')), os.listdir(save)


def test_export_cli(tmp_path):
    save = str(tmp_path)
    r = run_cli([os.path.join(REPO, 'export.py'),
                 '--num-stack', '1', '--hourglass-inch', '16',
                 '--imsize', '64', '--save-path', save, '--gpu-no', '-1'],
                cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.isfile(os.path.join(save, 'jit_traced_model_cpu.pth'))
    m = torch.jit.load(os.path.join(save, 'jit_traced_model_cpu.pth'))
    with torch.no_grad():
        out = m(torch.randn(1, 3, 64, 64))
    assert len(out) == 3


def test_main_train_torchrun_2rank(tmp_path):
    """The exact multi-rank CLI path the 8-GPU run uses: torchrun launches
    main.py --train-flag, one worker per rank (gloo on CPU), bucketed
    all-reduce, rank-0 checkpoint."""
    env = dict(os.environ)
    env['PYTHONPATH'] = REPO
    env.setdefault('MASTER_ADDR', '127.0.0.1')
    r = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29641', 'main.py'] + TRAIN_FLAGS +
        ['--save-path', str(tmp_path) + '/'],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    # rank-0 checkpoint with the reference dict format
    ckpt_path = tmp_path / 'check_point_1.pth'
    assert ckpt_path.exists(), r.stdout[-1500:]
    ckpt = torch.load(ckpt_path, map_location='cpu', weights_only=False)
    for k in ('epoch', 'state_dict', 'optimizer', 'scheduler', 'scaler',
              'loss_log'):
        assert k in ckpt, k
