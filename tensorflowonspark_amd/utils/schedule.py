"""Host-side LR schedules (parity: the reference's piecewise schedule,
``resnet_cifar_dist.py:35-66`` — warmup then step decays at epoch boundaries)."""


class PiecewiseLR:
    """Linear warmup to base_lr, then multiply at each boundary epoch.

    >>> sched = PiecewiseLR(base_lr=0.1, warmup_epochs=5,
    ...                     boundaries=[30, 60, 80], decays=[0.1, 0.01, 0.001])
    >>> opt.lr = sched(epoch_float)
    """

    def __init__(self, base_lr, warmup_epochs=0, boundaries=(), decays=()):
        assert len(boundaries) == len(decays)
        self.base_lr = base_lr
        self.warmup_epochs = warmup_epochs
        self.boundaries = list(boundaries)
        self.decays = list(decays)

    def __call__(self, epoch):
        if self.warmup_epochs and epoch < self.warmup_epochs:
            return self.base_lr * (epoch + 1e-9) / self.warmup_epochs
        lr = self.base_lr
        for b, d in zip(self.boundaries, self.decays):
            if epoch >= b:
                lr = self.base_lr * d
        return lr


class CosineLR:
    def __init__(self, base_lr, total_epochs, warmup_epochs=0, min_lr=0.0):
        self.base_lr = base_lr
        self.total = total_epochs
        self.warmup = warmup_epochs
        self.min_lr = min_lr

    def __call__(self, epoch):
        import math
        if self.warmup and epoch < self.warmup:
            return self.base_lr * (epoch + 1e-9) / self.warmup
        t = (epoch - self.warmup) / max(1e-9, self.total - self.warmup)
        return self.min_lr + (self.base_lr - self.min_lr) * 0.5 * (
            1 + math.cos(math.pi * min(t, 1.0)))
