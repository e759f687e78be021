"""Flat parameter/gradient storage + fused Adam.

All parameters of a module are re-materialized as views into ONE contiguous
fp32 buffer, and their .grad fields as views into a flat grad buffer.  This
makes:
  * the optimizer a single grid-stride HIP kernel (riqn_adam_step),
  * the Ape-X weight broadcast a zero-copy RCCL call on the flat buffer,
  * the target-net hard update one device memcpy (K14),
all of it hipGraph-capturable (device-side Adam step counter).
"""

from __future__ import annotations

from typing import List

import torch
from torch import nn

from . import get_extension


def flatten_module(module: nn.Module) -> torch.Tensor:
    """Re-point every parameter of `module` into one flat fp32 buffer.

    Returns the flat buffer; parameter tensors keep identity (same Parameter
    objects) but their storage becomes a view of the buffer.
    """
    params: List[nn.Parameter] = [p for p in module.parameters()]
    total = sum(p.numel() for p in params)
    flat = torch.empty(total, dtype=params[0].dtype,
                       device=params[0].device if params else "cpu")
    off = 0
    for p in params:
        k = p.numel()
        flat[off : off + k].copy_(p.detach().reshape(-1))
        with torch.no_grad():
            p.set_(flat[off : off + k].view_as(p))
        off += k
    return flat


def attach_flat_grads(module: nn.Module) -> torch.Tensor:
    """Give every parameter a .grad view into one flat buffer and return it."""
    params = [p for p in module.parameters()]
    total = sum(p.numel() for p in params)
    flat_grad = torch.zeros(total, dtype=params[0].dtype, device=params[0].device)
    off = 0
    for p in params:
        k = p.numel()
        p.grad = flat_grad[off : off + k].view_as(p)
        off += k
    return flat_grad


class FusedAdam:
    """Single-kernel Adam over a flat param/grad pair (GPU) with a device
    step counter so it replays correctly inside a hip graph."""

    def __init__(self, flat_param: torch.Tensor, flat_grad: torch.Tensor,
                 lr: float, eps: float, betas=(0.9, 0.999)):
        assert flat_param.is_cuda, "FusedAdam is the GPU path"
        self.ext = get_extension()
        self.param = flat_param
        self.grad = flat_grad
        self.m = torch.zeros_like(flat_param)
        self.v = torch.zeros_like(flat_param)
        self.step_t = torch.zeros(1, dtype=torch.int32, device=flat_param.device)
        self.lr = lr
        self.eps = eps
        self.betas = betas

    def zero_grad(self) -> None:
        self.grad.zero_()

    def step(self) -> None:
        self.ext.adam_step(self.param, self.grad, self.m, self.v, self.lr,
                           self.betas[0], self.betas[1], self.eps, self.step_t)

    def state_dict(self) -> dict:
        return {"m": self.m, "v": self.v, "step": self.step_t}

    def load_state_dict(self, sd: dict) -> None:
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.step_t.copy_(sd["step"])


class FusedAdamMixed:
    """fp32 master + bf16 module weights/grads: one kernel reads the bf16
    grads, updates the fp32 master and writes the bf16 params (K13 + the
    master->weight cast fused)."""

    def __init__(self, flat_param_bf16: torch.Tensor, flat_grad_bf16: torch.Tensor,
                 lr: float, eps: float, betas=(0.9, 0.999)):
        assert flat_param_bf16.dtype == torch.bfloat16
        self.ext = get_extension()
        self.param = flat_param_bf16
        self.grad = flat_grad_bf16
        self.master = flat_param_bf16.float()
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)
        self.step_t = torch.zeros(1, dtype=torch.int32, device=flat_param_bf16.device)
        self.lr = lr
        self.eps = eps
        self.betas = betas

    def zero_grad(self) -> None:
        self.grad.zero_()

    def step(self) -> None:
        self.ext.adam_step_mixed(self.master, self.grad, self.param, self.m,
                                 self.v, self.lr, self.betas[0], self.betas[1],
                                 self.eps, self.step_t)

    def state_dict(self) -> dict:
        return {"m": self.m, "v": self.v, "step": self.step_t,
                "master": self.master}

    def load_state_dict(self, sd: dict) -> None:
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.step_t.copy_(sd["step"])
        self.master.copy_(sd["master"])
        self.param.copy_(self.master.to(torch.bfloat16))


def master_state_dict(module: nn.Module, master: torch.Tensor) -> dict:
    """Reference-format fp32 state_dict from the fp32 master flat buffer
    (params) + the module's buffers (cast to fp32)."""
    sd = {}
    off = 0
    pmap = {}
    for name, p in module.named_parameters():
        k = p.numel()
        pmap[name] = master[off : off + k].view(p.shape).clone()
        off += k
    for name, t in module.state_dict().items():
        sd[name] = pmap[name] if name in pmap else t.float().clone()
    return sd
