"""SGD — stateless by default (reference: shallowspeed/optimizer.py:4-13),
with optional momentum and weight decay (beyond-reference extension).

CPU: elementwise torch updates per parameter.
GPU: ONE fused multi-tensor HIP kernel launch updates every parameter
of the stage: f32 master -= lr * (grad + wd*master [+ momentum]), and
re-emits the bf16 compute copy AND the transposed bf16 copy in the same
pass (the transposed copy is what makes dgrad an NT GEMM).
"""

import torch

from ..ops._ext import load_ext


class SGD:
    def __init__(self, parameters, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0):
        self.params = [p for p in parameters if p.requires_grad]
        self.lr = float(lr)
        self.momentum = float(momentum)
        self.weight_decay = float(weight_decay)
        self._vel = None
        if self.momentum:
            self._vel = [torch.zeros_like(p.data) for p in self.params]
        self._desc = None  # device-side descriptor table, built lazily

    def _build_desc(self):
        # Static descriptor: [master_ptr, grad_ptr, lp_ptr, lpt_ptr,
        # numel, cols, start] per tensor, int64, lives on device.
        # Pointers are stable because parameter storage is persistent.
        rows, start = [], 0
        for i, p in enumerate(self.params):
            t = p.data
            cols = t.shape[1] if t.dim() == 2 else 1
            rows.append([
                t.data_ptr(), p.grad.data_ptr(),
                p.lp.data_ptr() if p.lp is not None else 0,
                p.lp_t.data_ptr() if p.lp_t is not None else 0,
                t.numel(), cols, start,
                self._vel[i].data_ptr() if self._vel is not None else 0,
            ])
            start += t.numel()
        self._total = start
        cpu = torch.tensor(rows, dtype=torch.int64)
        self._desc = cpu.to(self.params[0].data.device)
        self._desc_cpu = cpu

    def step(self):
        if not self.params:
            return
        if self.params[0].data.is_cuda:
            ext = load_ext(required=True)
            if self._desc is None:
                self._build_desc()
            ext.sgd_multi(self._desc, self.lr, self._total, self.momentum,
                          self.weight_decay)
        else:
            for i, p in enumerate(self.params):
                g = p.grad
                if self.weight_decay:
                    g = g + self.weight_decay * p.data
                if self._vel is not None:
                    self._vel[i].mul_(self.momentum).add_(g)
                    g = self._vel[i]
                p.data -= self.lr * g
