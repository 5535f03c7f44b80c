"""Observability helpers (SURVEY.md §5 'tracing / profiling' plan):
per-step wall-clock + throughput meter, and the reference's `imshow`
helper (dead code in the reference, kept for surface parity —
/root/reference/cifar_example.py:10-14)."""

from __future__ import annotations

import time


class SpeedMeter:
    """Images/sec meter over a sliding window of steps."""

    def __init__(self, print_every: int = 100, label: str = "train"):
        self.print_every = print_every
        self.label = label
        self._n = 0
        self._images = 0
        self._t0 = time.perf_counter()

    def step(self, batch_size: int, sync=None):
        self._n += 1
        self._images += batch_size
        if self._n % self.print_every == 0:
            if sync is not None:
                sync()
            dt = time.perf_counter() - self._t0
            print(f"[{self.label}] {self._images / dt:.1f} images/sec "
                  f"({dt / self.print_every * 1e3:.2f} ms/step)")
            self._images = 0
            self._t0 = time.perf_counter()


def imshow(img):
    """Display a (normalized) CHW tensor — reference parity helper
    (unused by the training path, like the reference's)."""
    import matplotlib.pyplot as plt
    import numpy as np

    img = img / 2 + 0.5
    npimg = img.numpy()
    plt.imshow(np.transpose(npimg, (1, 2, 0)))
    plt.show()
