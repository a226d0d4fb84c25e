"""hipGraph-captured training iteration for the REAL training loop.

The reference's hot loop (/root/reference/train.py:86-162) launches every
iteration eagerly; on MI355X the hand-written step is ~150 kernel launches
and at 512x512/batch-16 the launch+python overhead is a measurable slice of
the 17 ms step. ``GraphedTrainStep`` captures forward + fused loss +
backward (+ bucketed RCCL all-reduce when world>1) + fused-Adam step into
ONE hipGraph per distinct input-shape/lr combination and replays it, with
the DataLoader batch copied into static device buffers each iteration.

Semantics preserved:
- loss history: the static per-stack loss tensors are cloned after each
  replay and appended to the LossCalculator log (same entries as eager).
- BN running stats / num_batches_tracked: their in-place device updates are
  part of the captured graph and re-execute on every replay.
- MultiStepLR: the learning rate is baked into the captured fused-Adam
  launch, so the graph key includes lr — a milestone change triggers a
  fresh capture.
- multiscale / last-batch shapes: one graph per shape key, up to
  ``MAX_SHAPES``; unseen keys beyond that run eagerly.

Not graphed (falls back to eager stepping): gradient accumulation
(``--sub-divisions`` > 1), CPU runs, capture failures of any kind.

Caveat: each capture performs ``WARMUP_STEPS`` real optimizer steps on the
pending batch (stream-capture needs warmed allocator/comm state), so that
batch contributes a couple of extra updates — same data, real gradients.
"""

import torch

from .. import amp
from ..parallel.ddp import BucketedDataParallel


class _GraphEntry:
    __slots__ = ('graph', 'statics', 'losses', 'per_stack', 'hm_logits')


class GraphedTrainStep:
    MAX_SHAPES = 8
    WARMUP_STEPS = 2

    def __init__(self, network, loss_calculator, optimizer, num_cls,
                 normalized_coord, amp_on):
        self.network = network
        self.calc = loss_calculator
        self.opt = optimizer
        self.num_cls = num_cls
        self.normalized = normalized_coord
        self.amp_on = amp_on
        self.enabled = torch.cuda.is_available()
        self.graphs = {}
        self.is_bucketed = isinstance(network, BucketedDataParallel)
        self._w = None  # loss-weight vector, created OUTSIDE capture

    # ------------------------------------------------------------------

    def _lr(self):
        return tuple(g['lr'] for g in self.opt.param_groups)

    def _eager_body(self, image, hm, off, wh, mask):
        """One full training iteration on the given (static) tensors."""
        from ..ops import hip
        with amp.autocast(enabled=self.amp_on):
            out = self.network(image)
        losses = hip.centernet_losses_logits(
            out, hm, off, wh, mask, self.calc.focal_alpha,
            self.calc.focal_beta, self.normalized)
        if self._w is None or self._w.device != losses.device:
            # built during warmup (pre-capture) — an H2D tensor
            # materialization inside stream capture would be illegal
            self._w = torch.tensor(
                [self.calc.hm_weight, self.calc.offset_weight,
                 self.calc.size_weight], device=losses.device,
                dtype=losses.dtype)
        per_stack = losses @ self._w
        total = per_stack.sum()
        total.backward()
        if self.is_bucketed:
            self.network.finish_backward()
        self.opt.step()
        self.opt.zero_grad(set_to_none=True)
        return losses, per_stack, out[:, -1, :self.num_cls]

    def _capture(self, batch):
        statics = tuple(t.clone() for t in batch)
        # warmup steps are real weight updates on this batch but are NOT
        # logged (the body computes losses directly, bypassing the
        # LossCalculator history)
        for _ in range(self.WARMUP_STEPS):
            self._eager_body(*statics)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        # thread_local capture mode: the DataLoader pin-memory thread (and
        # the RCCL watchdog) make HIP calls concurrently with the capture;
        # the default 'global' mode invalidates the capture on ANY
        # thread's unsafe call (measured: hipErrorStreamCaptureInvalidated
        # the moment a loader is attached)
        with torch.cuda.graph(g, capture_error_mode='thread_local'):
            losses, per_stack, hm_logits = self._eager_body(*statics)
        # replays never run python: keep the inference-fold caches honest
        from ..ops import hip
        hip.bump_train_stamp()
        e = _GraphEntry()
        e.graph = g
        e.statics = statics
        e.losses = losses
        e.per_stack = per_stack
        e.hm_logits = hm_logits
        return e

    # ------------------------------------------------------------------

    def step(self, image, hm, off, wh, mask):
        """Run one training iteration through the graph; returns the
        last-stack heatmap logits (for the PNG log) or None when this
        shape runs eagerly (caller falls back)."""
        if not self.enabled:
            return None
        key = (tuple(image.shape), tuple(hm.shape), self._lr())
        entry = self.graphs.get(key)
        if entry is None:
            if len(self.graphs) >= self.MAX_SHAPES:
                return None
            try:
                entry = self._capture((image, hm, off, wh, mask))
            except Exception as exc:
                print('rthd: train-graph capture failed (%s); '
                      'eager stepping' % exc)
                self.enabled = False
                # a failed capture can leave the stream poisoned; drain
                # it so the eager fallback starts clean
                try:
                    torch.cuda.synchronize()
                except Exception:
                    pass
                self.opt.zero_grad(set_to_none=True)
                return None
            self.graphs[key] = entry
            # stream capture RECORDS without executing: the static loss
            # tensors hold garbage until the first replay, and the batch
            # has not had its (post-warmup) update — fall through to the
            # replay below (statics already hold this batch)
        for dst, src in zip(entry.statics, (image, hm, off, wh, mask)):
            dst.copy_(src, non_blocking=True)
        entry.graph.replay()
        from ..ops import hip
        hip.bump_train_stamp()
        self._log_replay(entry)
        return entry.hm_logits

    def _log_replay(self, entry):
        det = entry.losses.detach().clone()
        ps = entry.per_stack.detach().clone()
        for i in range(det.shape[0]):
            self.calc._pending.append((det[i, 0], det[i, 1], det[i, 2],
                                       ps[i]))
