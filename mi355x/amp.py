"""Mixed-precision policy for the mi355x kernels.

Compute dtype is bf16 by default (MFMA bf16 peak ~2.5 PF dense on MI355X).
An `autocast(fp16)` context switches the 16-bit kernel dtype to fp16 and —
because fp16 has a narrow exponent — pairs with `GradScaler` loss scaling
(BASELINE config 5: "ResNet-18 AMP fp16 + SyncBatchNorm").

The scaler works on the FLAT fp32 gradient buffer maintained by
mi355x.parallel.flat (one inf/nan scan kernel over the whole buffer,
after the reducer's all-reduce — SURVEY.md §7 hard-part 4).
"""

from __future__ import annotations

import contextlib

import torch

_COMPUTE_DTYPE = torch.bfloat16


def get_compute_dtype() -> torch.dtype:
    return _COMPUTE_DTYPE


def set_compute_dtype(dtype: torch.dtype) -> None:
    global _COMPUTE_DTYPE
    assert dtype in (torch.bfloat16, torch.float16)
    _COMPUTE_DTYPE = dtype


@contextlib.contextmanager
def autocast(dtype: torch.dtype = torch.float16):
    global _COMPUTE_DTYPE
    prev = _COMPUTE_DTYPE
    set_compute_dtype(dtype)
    try:
        yield
    finally:
        _COMPUTE_DTYPE = prev


class GradScaler:
    """Loss scaling for fp16 training over a flat fp32 grad buffer.

    scale() multiplies the loss; after backward + gradient all-reduce the
    caller passes the flat grad buffer to step_ok(), which scans it for
    inf/nan ONCE; if clean, the optimizer consumes grad_scale=1/scale.
    Dynamic scaling: halve on overflow, double every `growth_interval`
    clean steps.
    """

    def __init__(self, init_scale=2.0 ** 14, growth_factor=2.0,
                 backoff_factor=0.5, growth_interval=1000):
        self.scale_value = float(init_scale)
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._clean_steps = 0

    def scale(self, loss: torch.Tensor) -> torch.Tensor:
        return loss * self.scale_value

    def inv_scale(self) -> float:
        return 1.0 / self.scale_value

    def step_ok(self, flat_grad: torch.Tensor) -> bool:
        """True if flat_grad is finite; updates the dynamic scale.

        Callers must capture the scale BEFORE calling (this may grow it):
            used = scaler.scale_value; (loss*used).backward(); ...
            if scaler.step_ok(flat): opt.grad_scale = base/used; opt.step()
        """
        finite = bool(torch.isfinite(flat_grad).all().item())
        if finite:
            self._clean_steps += 1
            if self._clean_steps >= self.growth_interval:
                self.scale_value *= self.growth_factor
                self._clean_steps = 0
        else:
            self.scale_value *= self.backoff_factor
            self._clean_steps = 0
        return finite
