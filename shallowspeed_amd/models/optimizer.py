"""Optimizers.

SGD — stateless by default (reference: shallowspeed/optimizer.py:4-13),
with optional momentum and weight decay (beyond-reference extension).
AdamW — beyond-reference: decoupled-weight-decay Adam for modern
training recipes.

CPU: elementwise torch updates per parameter (the numerics oracle).
GPU: ONE fused multi-tensor HIP kernel launch per step updates every
parameter of the stage — f32 master update + optimizer state, and
re-emits the bf16 compute copy AND the transposed bf16 copy in the
same pass (the transposed copy is what makes dgrad the same NT MFMA
kernel as forward).

Both expose state_dict()/load_state_dict() so checkpoints carry the
full optimization trajectory (momentum velocities / Adam moments and
step count).
"""

import torch

from ..ops._ext import load_ext


def _split_tiled_transposes(params):
    """Large 2-D weights whose transposed bf16 copy should be emitted
    by the tiled transpose kernel instead of the fused optimizer's
    in-kernel scatter (which stores 2 B at stride rows·2 — uncoalesced;
    measured 7x off roofline at 4096-wide).  Returns [(lp, lp_t)];
    also stashes the id set on the caller via the return's .ids attr
    convention (see _build_desc)."""
    out = []
    for p in params:
        if p.lp_t is None or p.lp is None:
            continue
        o, i = p.lp.shape
        if i >= 1024 and o % 64 == 0 and i % 64 == 0:
            out.append((p, p.lp, p.lp_t))
    return out


def _build_blockmap(params, device, epb=16384):
    """Block→(tensor, elem_base) map for the v2 fused optimizer
    kernels: each block owns `epb` contiguous elements of one tensor,
    replacing the v1 kernels' per-element binary search (a dependent
    6-level desc walk that held the 523M-param wide-model step at
    ~0.9 TB/s — profiles/r02_kernel_stats.md)."""
    rows = []
    for i, p in enumerate(params):
        n = p.data.numel()
        for b in range(0, n, epb):
            rows.append([i, b])
    return torch.tensor(rows, dtype=torch.int64).to(device)


def _clip_flat(flat_grad, clip_norm):
    """Global-norm gradient clip on the stage's flat f32 grad buffer.
    Runs AFTER the DP all-reduce (grads are the summed global grads),
    so every replica computes the identical scale — sync-free and
    deterministic.  Returns the pre-clip norm (a 0-d tensor; never
    host-synced on the hot path)."""
    n = torch.linalg.vector_norm(flat_grad)
    scale = (clip_norm / (n + 1e-6)).clamp(max=1.0)
    flat_grad.mul_(scale)
    return n


class SGD:
    def __init__(self, parameters, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, clip_norm=None, flat_grad=None):
        self.params = [p for p in parameters if p.requires_grad]
        self.lr = float(lr)
        self.momentum = float(momentum)
        self.weight_decay = float(weight_decay)
        self.clip_norm = clip_norm
        self.flat_grad = flat_grad  # model._flat_grad (clipping substrate)
        self._vel = None
        if self.momentum:
            self._vel = [torch.zeros_like(p.data) for p in self.params]
        self._desc = None  # device-side descriptor table, built lazily

    def _build_desc(self):
        # Static descriptor: [master_ptr, grad_ptr, lp_ptr, lpt_ptr,
        # numel, cols, start, vel_ptr] per tensor, int64, on device.
        # Pointers are stable because parameter storage is persistent.
        rows, start = [], 0
        self._tiled_t = _split_tiled_transposes(self.params)
        tiled_ids = {id(p) for p, _, _ in self._tiled_t}
        for i, p in enumerate(self.params):
            t = p.data
            cols = t.shape[1] if t.dim() == 2 else 1
            rows.append([
                t.data_ptr(), p.grad.data_ptr(),
                p.lp.data_ptr() if p.lp is not None else 0,
                p.lp_t.data_ptr()
                if p.lp_t is not None and id(p) not in tiled_ids else 0,
                t.numel(), cols, start,
                self._vel[i].data_ptr() if self._vel is not None else 0,
            ])
            start += t.numel()
        self._total = start
        cpu = torch.tensor(rows, dtype=torch.int64)
        self._desc = cpu.to(self.params[0].data.device)
        self._desc_cpu = cpu
        self._bmap = _build_blockmap(self.params, self._desc.device)

    def step(self):
        if not self.params:
            return
        if self.clip_norm is not None and self.flat_grad is not None:
            _clip_flat(self.flat_grad, self.clip_norm)
        if self.params[0].data.is_cuda:
            ext = load_ext(required=True)
            if self._desc is None:
                self._build_desc()
            if self._total >= (1 << 22):
                # blockmap kernel: wins at large param counts (binary
                # search chain removed; 8.2→~0.5 ms at 523M params)
                ext.sgd_multi2(self._desc, self._bmap, self.lr,
                               self.momentum, self.weight_decay)
            else:
                # v1 grid-stride kernel: wins at small counts (6.8 vs
                # 18.6 µs at the 270K-param flagship — measured)
                ext.sgd_multi(self._desc, self.lr, self._total,
                              self.momentum, self.weight_decay)
            for _, lp, lpt in self._tiled_t:
                ext.transpose_bf16(lp, lpt)
        else:
            for i, p in enumerate(self.params):
                g = p.grad
                if self.weight_decay:
                    g = g + self.weight_decay * p.data
                if self._vel is not None:
                    self._vel[i].mul_(self.momentum).add_(g)
                    g = self._vel[i]
                p.data -= self.lr * g

    # ------------------------------------------------- checkpoint state
    def state_dict(self):
        d = {"kind": "sgd", "lr": self.lr, "momentum": self.momentum,
             "weight_decay": self.weight_decay}
        if self._vel is not None:
            d["velocity"] = [v.detach().cpu() for v in self._vel]
        return d

    def load_state_dict(self, d):
        assert d.get("kind", "sgd") == "sgd", d.get("kind")
        self.lr = d.get("lr", self.lr)
        self.weight_decay = d.get("weight_decay", self.weight_decay)
        if "velocity" in d:
            self.momentum = d.get("momentum", self.momentum)
            if self._vel is None:
                self._vel = [torch.zeros_like(p.data) for p in self.params]
                self._desc = None  # descriptor must pick up vel pointers
            assert len(d["velocity"]) == len(self._vel)
            for dst, src in zip(self._vel, d["velocity"]):
                assert dst.shape == src.shape, (dst.shape, src.shape)
                dst.copy_(src.to(dst.device))


class AdamW:
    """Decoupled-weight-decay Adam (beyond-reference).  GPU: one fused
    multi-tensor HIP launch (adamw_multi_kernel) updates masters +
    moments and re-emits bf16/bf16ᵀ copies; CPU path is the oracle
    (matches torch.optim.AdamW semantics)."""

    def __init__(self, parameters, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.01,
                 clip_norm=None, flat_grad=None):
        self.params = [p for p in parameters if p.requires_grad]
        self.lr = float(lr)
        self.betas = (float(betas[0]), float(betas[1]))
        self.eps = float(eps)
        self.weight_decay = float(weight_decay)
        self.clip_norm = clip_norm
        self.flat_grad = flat_grad
        self.step_count = 0
        self._m = [torch.zeros_like(p.data) for p in self.params]
        self._v = [torch.zeros_like(p.data) for p in self.params]
        self._desc = None

    def _build_desc(self):
        rows, start = [], 0
        self._tiled_t = _split_tiled_transposes(self.params)
        tiled_ids = {id(p) for p, _, _ in self._tiled_t}
        for i, p in enumerate(self.params):
            t = p.data
            cols = t.shape[1] if t.dim() == 2 else 1
            rows.append([
                t.data_ptr(), p.grad.data_ptr(),
                p.lp.data_ptr() if p.lp is not None else 0,
                p.lp_t.data_ptr()
                if p.lp_t is not None and id(p) not in tiled_ids else 0,
                t.numel(), cols, start,
                self._m[i].data_ptr(), self._v[i].data_ptr(),
            ])
            start += t.numel()
        self._total = start
        cpu = torch.tensor(rows, dtype=torch.int64)
        self._desc = cpu.to(self.params[0].data.device)
        self._bmap = _build_blockmap(self.params, self._desc.device)

    def step(self):
        if not self.params:
            return
        if self.clip_norm is not None and self.flat_grad is not None:
            _clip_flat(self.flat_grad, self.clip_norm)
        self.step_count += 1
        b1, b2 = self.betas
        inv_bc1 = 1.0 / (1.0 - b1 ** self.step_count)
        inv_bc2 = 1.0 / (1.0 - b2 ** self.step_count)
        if self.params[0].data.is_cuda:
            ext = load_ext(required=True)
            if self._desc is None:
                self._build_desc()
            if self._total >= (1 << 22):
                ext.adamw_multi2(self._desc, self._bmap, self.lr, b1, b2,
                                 self.eps, self.weight_decay, inv_bc1,
                                 inv_bc2)
            else:
                ext.adamw_multi(self._desc, self.lr, self._total, b1, b2,
                                self.eps, self.weight_decay, inv_bc1,
                                inv_bc2)
            for _, lp, lpt in self._tiled_t:
                ext.transpose_bf16(lp, lpt)
        else:
            for i, p in enumerate(self.params):
                g = p.grad
                self._m[i].mul_(b1).add_(g, alpha=1 - b1)
                self._v[i].mul_(b2).addcmul_(g, g, value=1 - b2)
                if self.weight_decay:
                    p.data.mul_(1 - self.lr * self.weight_decay)
                mhat = self._m[i] * inv_bc1
                vhat = self._v[i] * inv_bc2
                p.data -= self.lr * mhat / (vhat.sqrt() + self.eps)

    # ------------------------------------------------- checkpoint state
    def state_dict(self):
        return {
            "kind": "adamw", "lr": self.lr, "betas": self.betas,
            "eps": self.eps, "weight_decay": self.weight_decay,
            "step": self.step_count,
            "exp_avg": [m.detach().cpu() for m in self._m],
            "exp_avg_sq": [v.detach().cpu() for v in self._v],
        }

    def load_state_dict(self, d):
        assert d.get("kind") == "adamw", d.get("kind")
        self.lr = d.get("lr", self.lr)
        self.betas = tuple(d.get("betas", self.betas))
        self.eps = d.get("eps", self.eps)
        self.weight_decay = d.get("weight_decay", self.weight_decay)
        self.step_count = d.get("step", 0)
        for dst, src in zip(self._m, d["exp_avg"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self._v, d["exp_avg_sq"]):
            dst.copy_(src.to(dst.device))
