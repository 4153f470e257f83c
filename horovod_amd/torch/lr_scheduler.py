"""Learning-rate warmup for large-batch data-parallel training.

Reference: the Keras LearningRateWarmupCallback /
LearningRateScheduleCallback (horovod/_keras/callbacks.py:108-193), which
implement the Goyal et al. gradual-warmup recipe for lr scaled by hvd.size().
Torch-native equivalent: a scheduler that ramps lr from lr/size to lr over
`warmup_epochs`, then defers to an optional wrapped scheduler.
"""
from torch.optim.lr_scheduler import _LRScheduler

from horovod_amd.torch.mpi_ops import size


class WarmupScheduler(_LRScheduler):
    """Linear warmup from base_lr/size() to base_lr over warmup_steps, then
    the wrapped scheduler (if any) takes over."""

    def __init__(self, optimizer, warmup_steps, after_scheduler=None,
                 start_scale=None, last_epoch=-1):
        self.warmup_steps = max(1, warmup_steps)
        self.after_scheduler = after_scheduler
        self.start_scale = (1.0 / max(size(), 1)
                            if start_scale is None else start_scale)
        self._finished = False
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch >= self.warmup_steps:
            if self.after_scheduler is not None:
                return self.after_scheduler.get_last_lr()
            return self.base_lrs
        frac = self.last_epoch / self.warmup_steps
        scale = self.start_scale + (1.0 - self.start_scale) * frac
        return [lr * scale for lr in self.base_lrs]

    def step(self, epoch=None):
        if self.last_epoch + 1 >= self.warmup_steps and \
                self.after_scheduler is not None:
            if not self._finished:
                self._finished = True
            self.after_scheduler.step(
                None if epoch is None else epoch - self.warmup_steps)
        super().step(epoch)
