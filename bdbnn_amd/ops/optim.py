"""Optimizers (ref:train.py:316-336) + fused multi-tensor step kernels (K9).

Reference recipe reproduced exactly:
* default: SGD(lr, momentum, weight_decay) + CosineAnnealingLR(T_max=epochs)
* imagenet: Adam with TWO param groups — conv-like params (ndim==4 or
  'conv' in name) carry weight_decay, everything else wd=0 — plus a
  linear-decay LambdaLR(1 - epoch/epochs).  (Note the reference applies
  wd to the CONV group only, ref:train.py:326-334.)

GPU fast path: one fused HIP kernel per step over all parameters
(csrc/optim.hip) instead of per-tensor torch ops.
"""

import torch

from .. import _C


def _match_layout(t, like):
    """Dense tensor t re-laid-out to match `like`'s strides (channels_last
    conv params after .to(memory_format=channels_last))."""
    if t.stride() == like.stride():
        return t
    if like.dim() == 4 and like.is_contiguous(
            memory_format=torch.channels_last):
        return t.contiguous(memory_format=torch.channels_last)
    return t.contiguous()


class FusedSGD(torch.optim.Optimizer):
    """SGD with momentum + weight decay; multi-tensor fused step on GPU."""

    def __init__(self, params, lr, momentum=0.0, weight_decay=0.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            mom = group["momentum"]
            wd = group["weight_decay"]
            ps, gs, bufs = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p)
                ps.append(p)
                gs.append(p.grad)
                bufs.append(state["momentum_buffer"])
            if not ps:
                continue
            if ps[0].is_cuda and _C.has_native():
                gs = [_match_layout(g.float(), p) for p, g in zip(ps, gs)]
                nat = _C.native_required()
                for i in range(0, len(ps), 64):  # kernel-arg tensor cap
                    nat.fused_sgd(ps[i:i + 64], gs[i:i + 64],
                                  bufs[i:i + 64], lr, mom, wd)
            else:
                for p, g, b in zip(ps, gs, bufs):
                    if wd != 0:
                        g = g.add(p, alpha=wd)
                    b.mul_(mom).add_(g)
                    p.add_(b, alpha=-lr)
        return loss


class FusedAdam(torch.optim.Optimizer):
    """Adam (no amsgrad, bias-corrected, torch semantics) fused on GPU."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            ps, gs, m1s, m2s = [], [], [], []
            step_t = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                    state["step"] = 0
                state["step"] += 1
                step_t = state["step"]
                ps.append(p)
                gs.append(p.grad)
                m1s.append(state["exp_avg"])
                m2s.append(state["exp_avg_sq"])
            if not ps:
                continue
            bc1 = 1 - beta1 ** step_t
            bc2 = 1 - beta2 ** step_t
            if ps[0].is_cuda and _C.has_native():
                gs = [_match_layout(g.float(), p) for p, g in zip(ps, gs)]
                nat = _C.native_required()
                for i in range(0, len(ps), 64):  # kernel-arg tensor cap
                    nat.fused_adam(ps[i:i + 64], gs[i:i + 64],
                                   m1s[i:i + 64], m2s[i:i + 64],
                                   lr, beta1, beta2, eps, wd, bc1, bc2)
            else:
                for p, g, m, v in zip(ps, gs, m1s, m2s):
                    if wd != 0:
                        g = g.add(p, alpha=wd)
                    m.mul_(beta1).add_(g, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
                    denom = (v / bc2).sqrt_().add_(eps)
                    p.addcdiv_(m / bc1, denom, value=-lr)
        return loss


def build_optimizer(args, model):
    """Reference-recipe optimizer + scheduler (ref:train.py:316-336)."""
    if getattr(args, "dataset", "cifar10") == "imagenet":
        conv_params, other_params = [], []
        for name, p in model.named_parameters():
            if not p.requires_grad:
                continue
            if p.ndim == 4 or "conv" in name:
                conv_params.append(p)
            else:
                other_params.append(p)
        optimizer = FusedAdam(
            [{"params": other_params, "weight_decay": 0.0},
             {"params": conv_params, "weight_decay": args.weight_decay}],
            lr=args.lr)
        scheduler = torch.optim.lr_scheduler.LambdaLR(
            optimizer, lambda epoch: 1 - epoch / args.epochs)
    else:
        optimizer = FusedSGD(model.parameters(), lr=args.lr,
                             momentum=args.momentum,
                             weight_decay=args.weight_decay)
        scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=args.epochs, eta_min=0)
    return optimizer, scheduler
