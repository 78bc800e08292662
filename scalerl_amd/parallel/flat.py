"""Flat-parameter rebinding.

Rebinds every parameter of an nn.Module as a view into ONE contiguous fp32
buffer, and every ``.grad`` as a view into a matching flat grad buffer.
This is the load-bearing memory layout of the learner:

- optimizer step   = one fused HIP kernel over the flat pair;
- grad all-reduce  = one RCCL call on the flat grad (xGMI: fewer, larger
  collectives — 288 GB HBM means no reason to fragment);
- weight publish   = one flat memcpy (D2H into the actors' shared CPU flat
  buffer, which their model's params alias).

The buffer can be placed in torch shared memory (``share=True``) so forked
actor processes alias the same storage — the MI355X replacement for the
reference's per-tensor ``share_memory_()`` (impala_atari.py:53-64).
"""

from __future__ import annotations

from typing import Dict, Iterator, List, Tuple

import torch
import torch.nn as nn


class FlatParams:
    def __init__(self, module: nn.Module, device=None, share: bool = False):
        self.module = module
        params: List[Tuple[str, nn.Parameter]] = list(module.named_parameters())
        assert all(p.dtype == torch.float32 for _, p in params), \
            "flat-param learner keeps master weights in fp32"
        self.numel = sum(p.numel() for _, p in params)
        device = device or next(module.parameters()).device
        self.flat = torch.empty(self.numel, dtype=torch.float32, device=device)
        if share:
            self.flat.share_memory_()
        self.flat_grad = torch.zeros_like(self.flat)
        self._slices: Dict[str, Tuple[int, int, torch.Size]] = {}
        off = 0
        with torch.no_grad():
            for name, p in params:
                n = p.numel()
                self.flat[off:off + n].copy_(p.detach().reshape(-1))
                p.data = self.flat[off:off + n].view(p.shape)
                p.grad = self.flat_grad[off:off + n].view(p.shape)
                self._slices[name] = (off, n, p.shape)
                off += n

    def zero_grad(self) -> None:
        self.flat_grad.zero_()

    def rebind_grads(self) -> None:
        """Re-attach .grad views (autograd may replace .grad if set to None)."""
        for name, p in self.module.named_parameters():
            off, n, shape = self._slices[name]
            if p.grad is None or p.grad.data_ptr() != self.flat_grad[off:off + n].data_ptr():
                p.grad = self.flat_grad[off:off + n].view(shape)

    @torch.no_grad()
    def copy_into(self, dst_flat: torch.Tensor, non_blocking: bool = False) -> None:
        dst_flat.copy_(self.flat, non_blocking=non_blocking)

    @torch.no_grad()
    def load_from(self, src_flat: torch.Tensor, non_blocking: bool = False) -> None:
        self.flat.copy_(src_flat, non_blocking=non_blocking)

    def grad_views(self) -> Iterator[torch.Tensor]:
        for name, p in self.module.named_parameters():
            yield p.grad
