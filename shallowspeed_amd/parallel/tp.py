"""Tensor parallelism (beyond-reference): Megatron-style column/row-
parallel Linear pair over RCCL/xGMI.

The reference has no layer-internal sharding (TP absent);
this extends the framework beyond capability parity.  Design:

  * ColumnParallelLinear shards W by OUTPUT rows: each rank computes a
    column slice of y locally (bias shard local, ReLU local).  Backward
    all-reduduces dx (each rank contributes its shard's term).
  * RowParallelLinear shards W by INPUT columns and consumes the
    column-sharded activation directly (no comm between the pair);
    forward all-reduces the partial products then adds the replicated
    bias.  Backward is comm-free (dx comes out column-sharded for the
    preceding ColumnParallel layer).

  ⇒ one all-reduce forward + one backward per (col,row) pair — the
  classic Megatron f/g operators — over `torch.distributed` (nccl ==
  RCCL on ROCm; gloo for CPU tests).

Shape-seeded determinism is preserved the same way the pipeline
stages preserve it (reference layers.py:106-113 property): every rank
materializes the FULL seeded weight and takes its slice, so a TP-
sharded model is numerically the same function as the serial model —
the serial-equivalence test strategy carries over intact.

Compute runs on the SAME HIP kernels as Linear (functional.linear_*);
there is no TP-specific kernel — sharding only changes shapes and
inserts collectives.
"""

import math

import torch
import torch.distributed as dist

from ..models.layers import Module, Parameter, Sequential, _shape_seed
from ..ops import functional as F


def _full_seeded_weight(in_dims, out_dims):
    g = torch.Generator().manual_seed(_shape_seed(in_dims, out_dims))
    w = torch.randn(out_dims, in_dims, generator=g, dtype=torch.float32)
    return w / math.sqrt(in_dims)


def _allreduce(t, group):
    if group is not None and dist.is_initialized() and \
            dist.get_world_size(group=group) > 1:
        t = t.contiguous()
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return t


class ColumnParallelLinear(Module):
    """y_local = x @ W_shardᵀ + b_shard (+ReLU); W sharded by out rows.
    Input x is replicated; output is column-sharded."""

    def __init__(self, in_dims, out_dims, tp_group=None, tp_rank=0,
                 tp_world=1, activation=None):
        super().__init__()
        assert out_dims % tp_world == 0, (out_dims, tp_world)
        assert activation in (None, "relu")
        self.in_dims, self.out_dims = in_dims, out_dims
        self.activation = activation
        self.tp_group, self.tp_rank, self.tp_world = tp_group, tp_rank, tp_world
        shard = out_dims // tp_world
        w_full = _full_seeded_weight(in_dims, out_dims)
        self._params["weight"] = Parameter(
            w_full[tp_rank * shard:(tp_rank + 1) * shard].contiguous())
        self._params["bias"] = Parameter(torch.zeros(shard))
        self.weight = self._params["weight"]
        self.bias = self._params["bias"]

    def forward(self, inputs, mubatch_id: int = 0):
        self._stash("x", mubatch_id, inputs)
        y = F.linear_fwd(inputs, self.weight.compute(), self.bias.compute(),
                         relu=(self.activation == "relu"))
        if self.activation == "relu":
            self._stash("y", mubatch_id, y)
        return y

    def backward(self, dout, mubatch_id: int = 0, need_dx: bool = True):
        x = self._unstash("x", mubatch_id)
        mask_src = None
        if self.activation == "relu":
            mask_src = self._unstash("y", mubatch_id)
        dx = None
        if need_dx:
            dx = F.linear_dgrad(dout, self.weight.compute(),
                                self.weight.compute_t(), mask_src)
            # each rank holds a partial dx (its output-shard's term):
            # the Megatron g-operator
            dx = _allreduce(dx, self.tp_group)
        F.linear_wgrad_acc(dout, x, self.weight.grad, self.bias.grad,
                           mask_src)
        return dx


class RowParallelLinear(Module):
    """y = Σ_ranks x_local @ W_shardᵀ (+ replicated bias)(+ReLU); W
    sharded by in columns, input column-sharded, output replicated."""

    def __init__(self, in_dims, out_dims, tp_group=None, tp_rank=0,
                 tp_world=1, activation=None):
        super().__init__()
        assert in_dims % tp_world == 0, (in_dims, tp_world)
        assert activation in (None, "relu")
        self.in_dims, self.out_dims = in_dims, out_dims
        self.activation = activation
        self.tp_group, self.tp_rank, self.tp_world = tp_group, tp_rank, tp_world
        shard = in_dims // tp_world
        w_full = _full_seeded_weight(in_dims, out_dims)
        self._params["weight"] = Parameter(
            w_full[:, tp_rank * shard:(tp_rank + 1) * shard].contiguous())
        # bias is REPLICATED (added once, post-reduce); every rank
        # computes the identical bias grad from the replicated dz, so
        # replicas stay in sync like DP replicas do
        self._params["bias"] = Parameter(torch.zeros(out_dims))
        self.weight = self._params["weight"]
        self.bias = self._params["bias"]

    def forward(self, inputs, mubatch_id: int = 0):
        self._stash("x", mubatch_id, inputs)
        y = F.linear_fwd(inputs, self.weight.compute(), None, relu=False)
        y = _allreduce(y, self.tp_group)  # the Megatron f-operator
        y = y + self.bias.compute().to(y.dtype)
        if self.activation == "relu":
            y = F.relu_fwd(y)
            self._stash("y", mubatch_id, y)
        return y

    def backward(self, dout, mubatch_id: int = 0, need_dx: bool = True):
        x = self._unstash("x", mubatch_id)
        if self.activation == "relu":
            y = self._unstash("y", mubatch_id)
            dout = F.relu_bwd(dout, y)
        dx = None
        if need_dx:
            # dz is replicated; dx comes out column-sharded — no comm
            dx = F.linear_dgrad(dout, self.weight.compute(),
                                self.weight.compute_t(), None)
        F.linear_wgrad_acc(dout, x, self.weight.grad, self.bias.grad, None)
        return dx


def tp_mlp_layers(sizes, tp_group=None, tp_rank=0, tp_world=1):
    """Build a TP-sharded MLP layer list from the same `sizes`
    boundaries MLP uses: consecutive Linears alternate
    column-parallel / row-parallel (hidden dims sharded, boundary
    activations replicated every 2 layers), with a replicated head.
    Requires an even number of hidden Linears and hidden dims % tp."""
    from ..models.layers import Linear

    n_linear = len(sizes) - 1
    layers = []
    i = 0
    while i < n_linear - 1:  # pair up all but the head
        a, h, b = sizes[i], sizes[i + 1], sizes[i + 2]
        layers.append(ColumnParallelLinear(
            a, h, tp_group, tp_rank, tp_world, activation="relu"))
        layers.append(RowParallelLinear(
            h, b, tp_group, tp_rank, tp_world,
            activation="relu" if i + 2 < n_linear else None))
        i += 2
    if i == n_linear - 1:  # odd count: replicated head Linear
        layers.append(Linear(sizes[-2], sizes[-1], activation=None))
    return layers


class TPMLP(Sequential):
    """TP-sharded single-stage MLP with the standard loss head —
    drop-in for MLP when dp == pp == 1 and --tp > 1."""

    def __init__(self, sizes, tp_group, tp_rank, tp_world,
                 global_batch_size, loss="xent"):
        from ..models.layers import SoftmaxMSE, SoftmaxXent

        layers = tp_mlp_layers(sizes, tp_group, tp_rank, tp_world)
        head = {"xent": SoftmaxXent, "mse": SoftmaxMSE}[loss](
            global_batch_size)
        layers.append(head)
        super().__init__(layers)
        self.in_dim, self.out_dim = sizes[0], sizes[-1]
        self.tp_world = tp_world
