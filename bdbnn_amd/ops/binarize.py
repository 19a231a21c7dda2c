"""1-bit quantizers: sign forward, STE/EDE/polynomial backward.

Semantic contracts (these define what the HIP kernels must reproduce
bit-for-bit at the +-1 level):

* ``binsign(x) = +1 if x >= 0 else -1``.  Zero maps to +1 (a bit-packed
  representation has no third state; PyTorch's ``torch.sign(0) == 0``
  is NOT the contract here).
* Clip-STE backward (IR-Net/XNOR lineage):  dL/dx = g * 1(|x| <= 1).
* EDE backward (IR-Net; the (t, k) schedule is injected per epoch by the
  training engine, ref:train.py:409-415 + ref:utils/utils.py:8-14):
  dL/dx = g * k * t * (1 - tanh^2(t*x)).
* Polynomial backward (ReActNet's BinaryActivation):
  dL/dx = g * (2 + 2x) for -1 <= x < 0, g * (2 - 2x) for 0 <= x < 1,
  else 0.
* Weight binarization (ReActNet HardBinaryConv contract, reconstructed
  from call sites — ref:train.py:30-32):
  Wb = alpha_k * binsign(W) with per-output-channel
  alpha_k = mean(|W[k]|) DETACHED; backward dL/dW = g * 1(|W| <= 1)
  (straight-through of clamp(W, -1, 1); no gradient through alpha).
"""

import torch
import torch.nn as nn


def binsign(x: torch.Tensor) -> torch.Tensor:
    """+-1 sign with binsign(0) = +1 (bit-packable; kernel contract)."""
    return torch.where(x >= 0, torch.ones_like(x), -torch.ones_like(x))


class SignSTE(torch.autograd.Function):
    """sign forward / clipped straight-through backward."""

    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return binsign(x)

    @staticmethod
    def backward(ctx, g):
        (x,) = ctx.saved_tensors
        return g * (x.abs() <= 1).to(g.dtype)


class SignEDE(torch.autograd.Function):
    """sign forward / IR-Net error-decay-estimator backward k*t*(1-tanh^2(t*x))."""

    @staticmethod
    def forward(ctx, x, t, k):
        ctx.save_for_backward(x)
        ctx.t = float(t)
        ctx.k = float(k)
        return binsign(x)

    @staticmethod
    def backward(ctx, g):
        (x,) = ctx.saved_tensors
        th = torch.tanh(ctx.t * x)
        return g * ctx.k * ctx.t * (1.0 - th * th), None, None


class SignApprox(torch.autograd.Function):
    """sign forward / ReActNet piecewise-polynomial backward."""

    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return binsign(x)

    @staticmethod
    def backward(ctx, g):
        (x,) = ctx.saved_tensors
        neg = (x >= -1) & (x < 0)
        pos = (x >= 0) & (x < 1)
        grad = torch.zeros_like(x)
        grad = torch.where(neg, 2.0 + 2.0 * x, grad)
        grad = torch.where(pos, 2.0 - 2.0 * x, grad)
        return g * grad


def weight_scale(w: torch.Tensor) -> torch.Tensor:
    """Per-output-channel alpha_k = mean(|W[k]|), shape (K, 1, 1, 1), detached."""
    return w.abs().mean(dim=(1, 2, 3), keepdim=True).detach()


class _BinarizeWeight(torch.autograd.Function):
    @staticmethod
    def forward(ctx, w):
        ctx.save_for_backward(w)
        return weight_scale(w) * binsign(w)

    @staticmethod
    def backward(ctx, g):
        (w,) = ctx.saved_tensors
        return g * (w.abs() <= 1).to(g.dtype)


def binarize_weight(w: torch.Tensor) -> torch.Tensor:
    """alpha * sign(W) with clip-STE backward (alpha detached)."""
    return _BinarizeWeight.apply(w)


class LearnableBias(nn.Module):
    """ReActNet per-channel learnable shift (applied before/after activations)."""

    def __init__(self, channels: int):
        super().__init__()
        self.bias = nn.Parameter(torch.zeros(1, channels, 1, 1))

    def forward(self, x):
        return x + self.bias.expand_as(x)


class BinaryActivation(nn.Module):
    """sign activation.

    mode 'approx': ReActNet polynomial backward (default, react models);
    mode 'ste':    clip straight-through;
    mode 'ede':    IR-Net EDE — requires ``self.t`` / ``self.k`` attributes,
                   injected per-epoch by the engine (ref:train.py:409-415).
    If ``t``/``k`` are present on the module they take precedence (the
    reference injects them into every conv; we mirror that hook).
    """

    def __init__(self, mode: str = "approx"):
        super().__init__()
        assert mode in ("approx", "ste", "ede")
        self.mode = mode
        self.t = None
        self.k = None

    def forward(self, x):
        if self.t is not None and self.k is not None:
            return SignEDE.apply(x, float(self.t), float(self.k))
        if self.mode == "approx":
            return SignApprox.apply(x)
        return SignSTE.apply(x)
