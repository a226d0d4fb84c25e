"""bf16 mixed precision for MI355X (CDNA4 MFMA), replacing torch.cuda.amp.

The reference uses fp16 autocast + GradScaler dynamic loss scaling
(/root/reference/train.py:63,96-97,128-132). On CDNA4 the right dtype is
bf16: same exponent range as fp32, runs on the bf16 MFMA pipe (~2.5 PF
dense), and needs NO loss scaling. So:

- ``autocast(enabled)``: context manager that (a) flips a flag the HIP op
  layer reads (custom kernels take bf16 inputs, accumulate fp32) and
  (b) enables torch.autocast(device_type='cuda', dtype=bf16) so any residual
  torch ops follow the same policy.
- ``GradScaler``: API-compatible no-op (scale/step/unscale_/update/
  state_dict/load_state_dict) so the train loop and the checkpoint format
  keep the reference shape (checkpoints store a 'scaler' entry,
  train.py:76-82 — and unlike the reference, we restore it on resume).
"""

import contextlib

import torch

_autocast_depth = 0


def is_autocast_enabled():
    return _autocast_depth > 0


@contextlib.contextmanager
def autocast(enabled=True):
    global _autocast_depth
    if not enabled:
        # torch semantics: a nested disabled region suspends autocast
        prev = _autocast_depth
        _autocast_depth = 0
        try:
            if torch.cuda.is_available():
                with torch.autocast(device_type='cuda', enabled=False):
                    yield
            else:
                yield
        finally:
            _autocast_depth = prev
        return
    _autocast_depth += 1
    try:
        if torch.cuda.is_available():
            with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
                yield
        else:
            # CPU: keep fp32 (the tiny-config CPU path is a correctness
            # harness, not a perf path).
            yield
    finally:
        _autocast_depth -= 1


class GradScaler:
    """bf16 needs no loss scaling; keep the torch.cuda.amp.GradScaler API."""

    def __init__(self, enabled=True):
        self._enabled = enabled

    def scale(self, loss):
        return loss

    def unscale_(self, optimizer):
        pass

    def step(self, optimizer):
        optimizer.step()

    def update(self):
        pass

    def state_dict(self):
        return {'enabled': self._enabled, 'kind': 'bf16-noop'}

    def load_state_dict(self, state):
        self._enabled = bool(state.get('enabled', True))

    def is_enabled(self):
        return self._enabled


# ---------------------------- fp8 inference mode ----------------------------
# BASELINE config 5: fp8 e4m3 MFMA inference. Weights are per-channel scaled
# into e4m3 range (scale folded into the conv epilogue); activations are
# converted bf16 -> fp8 inside the conv staging (saturating). Inference only.

_fp8_depth = 0


def fp8_enabled():
    return _fp8_depth > 0


@contextlib.contextmanager
def fp8_autocast(enabled=True):
    global _fp8_depth
    if not enabled:
        yield
        return
    _fp8_depth += 1
    try:
        yield
    finally:
        _fp8_depth -= 1
