"""Checkpoint save/load in the reference ``check_point_%d.pth`` format.

Dict keys (mandated by BASELINE.json; reference train.py:76-82):
``{'epoch', 'state_dict', 'optimizer', 'scheduler', 'scaler', 'loss_log'}``
— state_dict is the bare module's (DDP wrapper unwrapped), saved by rank 0
each epoch. The arch-hyperparameter sidecar ``argument.pickle`` written at
train time (config.py) makes checkpoints self-describing for eval.

Unlike the reference (which saves scaler state but never restores it —
train.py:195-199), ``load_checkpoint`` restores everything it finds.
"""

import os

import torch


def _unwrap(network):
    return network.module if hasattr(network, 'module') else network


def save_checkpoint(path_or_dir, epoch, network, optimizer=None,
                    scheduler=None, scaler=None, loss_calculator=None):
    """Write check_point_{epoch}.pth (epoch is 1-based in the filename)."""
    if os.path.isdir(path_or_dir):
        path = os.path.join(path_or_dir, 'check_point_%d.pth' % epoch)
    else:
        path = path_or_dir
    loss_log = None
    if loss_calculator is not None:
        loss_log = loss_calculator.get_loss_log() \
            if hasattr(loss_calculator, 'get_loss_log') else loss_calculator.log
    torch.save({
        'epoch': epoch,
        'state_dict': _unwrap(network).state_dict(),
        'optimizer': optimizer.state_dict() if optimizer is not None else None,
        'scheduler': scheduler.state_dict() if scheduler is not None else None,
        'scaler': scaler.state_dict() if scaler is not None else None,
        'loss_log': loss_log,
    }, path)
    return path


def load_checkpoint(path, network, optimizer=None, scheduler=None,
                    scaler=None, loss_calculator=None, map_location='cpu'):
    """Load a checkpoint into the given components; returns the dict."""
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    _unwrap(network).load_state_dict(ckpt['state_dict'])
    if optimizer is not None and ckpt.get('optimizer') is not None:
        optimizer.load_state_dict(ckpt['optimizer'])
    if scheduler is not None and ckpt.get('scheduler') is not None:
        scheduler.load_state_dict(ckpt['scheduler'])
    if scaler is not None and ckpt.get('scaler') is not None:
        scaler.load_state_dict(ckpt['scaler'])
    if loss_calculator is not None and ckpt.get('loss_log') is not None:
        if hasattr(loss_calculator, 'load_loss_log'):
            loss_calculator.load_loss_log(ckpt['loss_log'])
        else:
            loss_calculator.log = ckpt['loss_log']
    return ckpt
