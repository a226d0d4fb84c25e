"""Optimizer/scheduler factory (reference /root/reference/optim.py:3-12).

Adam at ``--lr`` plus MultiStepLR(milestones, gamma). ``--optim`` selects the
algorithm; Adam is the reference default and SGD/AdamW are accepted for
experiments (same signature either way).
"""

import torch.optim as optim


def get_optimizer(network, lr, lr_milestone, lr_gamma, algo='Adam',
                  capturable=False):
    """capturable=True keeps the fused-Adam step state on device so the
    step can be recorded into a hipGraph (engine/graphed.py); only valid
    for CUDA parameters."""
    algo = (algo or 'Adam').lower()
    params = network.parameters()
    if algo == 'adam':
        # fused Adam: one kernel per dtype group instead of ~200 small
        # elementwise launches per step (measured ~1 ms/step)
        try:
            optimizer = optim.Adam(params, lr=lr, fused=True,
                                   capturable=capturable)
        except (RuntimeError, TypeError, ValueError):
            optimizer = optim.Adam(params, lr=lr, foreach=True)
    elif algo == 'adamw':
        optimizer = optim.AdamW(params, lr=lr)
    elif algo == 'sgd':
        optimizer = optim.SGD(params, lr=lr, momentum=0.9)
    else:
        raise ValueError(f'unknown optimizer {algo!r}')

    scheduler = None
    if lr_milestone is not None:
        scheduler = optim.lr_scheduler.MultiStepLR(
            optimizer=optimizer, milestones=lr_milestone, gamma=lr_gamma)
    return optimizer, scheduler
