"""Model layer: Parameter / Module / Linear / loss heads / Sequential / MLP.

Reference: shallowspeed/layers.py (270 LoC, NumPy).  Behavioral parity:

  * explicit forward/backward (no autograd), per-µbatch activation
    stash keyed by mubatch_id (layers.py:70,117) so multiple µbatches
    can be in flight under pipeline schedules,
  * deterministic SHAPE-SEEDED weight init (layers.py:106-113): the
    same (in,out) shape produces bit-identical weights regardless of
    DP/PP partitioning — the load-bearing trick behind replica-sync
    and pipeline-vs-serial equivalence testing,
  * gradient ACCUMULATION in backward (layers.py:135-136) — wgrad does
    `grad +=`, enabling µbatch accumulation,
  * Sequential grad-hook API (layers.py:182-213): after each layer's
    backward the per-parameter hooks fire (this is where the Worker
    injects the DP all-reduce, pipe.py:389-400), then post-grad hooks,
  * MLP stage construction with one-element overlap slicing and no
    activation on the very last Linear (layers.py:242-263).

MI355X-native differences:
  * tensors are torch; on GPU the compute dtype is bf16 and every hot
    op dispatches to the in-tree HIP/CDNA4 extension,
  * parameters keep an f32 master + bf16 copy + transposed bf16 copy
    (the transposed copy makes dgrad the same NT MFMA kernel as fwd),
  * all f32 grads of a model live in ONE flat buffer (views per param)
    — zero_grad is one memset and DP bucketing operates on flat slices
    (the reference's own docstring asks for bucketing, pipe.py:309-310),
  * the loss head is a fused softmax-cross-entropy (or softmax-MSE for
    reference parity) producing d(logits) in one op.
"""

import math

import torch

from ..ops import functional as F


_DZ_FIRST = None


def _dz_first_enabled() -> bool:
    global _DZ_FIRST
    if _DZ_FIRST is None:
        import os

        # default OFF: A/B'd neutral at the flagship (0.2072 vs 0.2081
        # ms within one call) — the mask re-reads are L2-absorbed there
        _DZ_FIRST = os.environ.get("SS_DZ_FIRST", "0") == "1"
    return _DZ_FIRST


def _shape_seed(in_dims: int, out_dims: int) -> int:
    # Same spirit as reference layers.py:106-108 (seed from the layer
    # shape only): partition-invariant deterministic init.
    return (in_dims * 1_000_003 + out_dims * 7919) & 0x7FFFFFFF


class Parameter:
    """f32 master tensor + f32 grad (+ bf16 / transposed-bf16 device
    copies on GPU).  Reference: layers.py:17-28."""

    def __init__(self, data: torch.Tensor, requires_grad: bool = True):
        self.data = data.float()
        self.requires_grad = requires_grad
        self.grad = torch.zeros_like(self.data) if requires_grad else None
        self.lp = None    # bf16 compute copy (GPU)
        self.lp_t = None  # transposed bf16 copy (GPU, 2-D weights only)
        # DP all-reduce bookkeeping (reference parks the MPI request on
        # the parameter, pipe.py:314): we park the async work handle.
        self._comm_handle = None

    @property
    def shape(self):
        return self.data.shape

    def compute(self) -> torch.Tensor:
        """The tensor GEMMs consume: bf16 copy on GPU, master on CPU."""
        return self.lp if self.lp is not None else self.data

    def compute_t(self):
        return self.lp_t

    def materialize_device(self, device, compute_dtype):
        self.data = self.data.to(device)
        if self.grad is not None:
            self.grad = self.grad.to(device)
        if compute_dtype == torch.bfloat16:
            self.lp = self.data.to(torch.bfloat16)
            if self.data.dim() == 2:
                self.lp_t = self.lp.t().contiguous()

    def sync_lp(self):
        """Refresh low-precision copies from the master (used by the CPU
        SGD path and by checkpoint load; the fused GPU SGD kernel does
        this in-kernel)."""
        if self.lp is not None:
            self.lp.copy_(self.data.to(torch.bfloat16))
            if self.lp_t is not None:
                self.lp_t.copy_(self.lp.t())


class Module:
    """Stateful op with per-µbatch cache.  Reference: layers.py:31-64."""

    def __init__(self):
        self._params = {}
        self._cache = {}
        self._training = True

    def forward(self, inputs: torch.Tensor, mubatch_id: int = 0):
        raise NotImplementedError

    def backward(self, dout: torch.Tensor, mubatch_id: int = 0):
        raise NotImplementedError

    def parameters(self):
        return list(self._params.values())

    def zero_grad(self):
        for p in self._params.values():
            if p.grad is not None:
                p.grad.zero_()

    def train(self):
        self._training = True

    def eval(self):
        self._training = False

    def _stash(self, name, mubatch_id, value):
        if self._training:
            self._cache[(name, mubatch_id)] = value

    def _unstash(self, name, mubatch_id):
        return self._cache.pop((name, mubatch_id))

    def materialize_device(self, device, compute_dtype):
        for p in self._params.values():
            p.materialize_device(device, compute_dtype)


class Linear(Module):
    """y = x @ W^T + b with optional FUSED ReLU (GEMM epilogue).

    Reference: layers.py:99-142 (module-level ReLU fusion at
    layers.py:120-122 becomes a kernel epilogue; backward unwinds the
    activation first via the stashed post-ReLU output, then runs
    dgrad/wgrad — layers.py:124-139).
    Init: W ~ N(0,1)/√in_dims f32, b = 0, shape-seeded
    (layers.py:106-113).
    """

    def __init__(self, in_dims: int, out_dims: int, activation=None):
        super().__init__()
        assert activation in (None, "relu", "gelu")
        self.in_dims, self.out_dims = in_dims, out_dims
        self.activation = activation
        g = torch.Generator().manual_seed(_shape_seed(in_dims, out_dims))
        w = torch.randn(out_dims, in_dims, generator=g, dtype=torch.float32)
        w /= math.sqrt(in_dims)
        self._params["weight"] = Parameter(w)
        self._params["bias"] = Parameter(torch.zeros(out_dims))
        self.weight = self._params["weight"]
        self.bias = self._params["bias"]
        # deferred-µbatch wgrad (GPU): collect (dy, x, mask) per
        # µbatch and launch ONE chunked kernel at flush time.  The
        # WINDOW caps how many µbatches are retained (holding every
        # µbatch's dout/x would break 1F1B's bounded activation stash
        # — the schedule's whole memory story): when the window fills,
        # the pending chunks flush early (grads accumulate atomically,
        # so partial flushes compose).
        self._defer_wgrad = False
        self._wgrad_window = 1 << 30
        self._wgrad_pending = []
        # MX-fp8 forward (training): quantize activations per µbatch
        # and keep an fp8 copy of the weight, refreshed lazily after
        # each optimizer step (invalidate_fp8()).  Backward stays bf16
        # (dgrad/wgrad on the bf16 stash); the f32 master is untouched
        # — this is the training-forward extension of the round-1
        # serving tier (csrc/fp8.hip, ~1.7x the bf16 8-phase GEMM).
        self.fp8_fwd = False
        self._wq = None
        self._ws = None

    def _fp8_ok(self, rows: int) -> bool:
        return (self.fp8_fwd and self.activation in (None, "relu")
                and rows % 256 == 0
                and self.out_dims % 256 == 0 and self.in_dims % 256 == 0)

    def forward(self, inputs, mubatch_id: int = 0):
        self._stash("x", mubatch_id, inputs)
        if inputs.is_cuda and self._fp8_ok(inputs.shape[0]):
            if self._wq is None:
                self._wq, self._ws = F.fp8_quantize(self.weight.compute())
            xq, xs = F.fp8_quantize(inputs)
            y = F.linear_fwd_fp8(xq, xs, self._wq, self._ws,
                                 self.bias.compute(),
                                 relu=(self.activation == "relu"))
            if self.activation == "relu":
                self._stash("y", mubatch_id, y)
            return y
        if self.activation == "gelu":
            z = F.linear_fwd(inputs, self.weight.compute(),
                             self.bias.compute(), relu=False)
            self._stash("z", mubatch_id, z)
            return F.gelu_fwd(z)
        y = F.linear_fwd(
            inputs, self.weight.compute(), self.bias.compute(),
            relu=(self.activation == "relu"),
        )
        if self.activation == "relu":
            # post-ReLU output doubles as the backward mask source
            # (out>0 ⟺ pre-act>0); replaces the reference's separate
            # bitmask stash (layers.py:70).
            self._stash("y", mubatch_id, y)
        return y

    def backward(self, dout, mubatch_id: int = 0, need_dx: bool = True):
        x = self._unstash("x", mubatch_id)
        mask_src = None
        if self.activation == "relu":
            mask_src = self._unstash("y", mubatch_id)
            if dout.is_cuda:
                B, O = dout.shape
                I = self.in_dims
                # WIDE aligned layers: materializing dz once lets
                # dgrad take the unmasked glds tier and strips the
                # mask re-reads from wgrad (mask fusion wins at small
                # shapes where L2 absorbs the re-reads — measured).
                # Second disjunct: any shape the 256²-tile wgrad tier
                # can take (O%256, I%256, B%128) unmasks too — the
                # masked path would silently fall back to the 128²
                # wgrad tier (~0.65 PF vs 1.1-1.2 PF), which costs far
                # more than one elementwise dz pass.  dz is shared by
                # dgrad and (deferred) wgrad, so the pass amortizes
                # over both GEMMs.
                wide_glds = (B % 128 == 0 and I % 128 == 0 and O % 32 == 0
                             and (B // 128) * (I // 128) >= 512
                             and O >= 512 and I >= 512)
                wgrad256_ok = (O % 256 == 0 and I % 256 == 0
                               and B % 128 == 0 and B >= 128
                               and O >= 512 and I >= 512
                               and (O // 256) * (I // 256) >= 64)
                # dgrad-skipped layers (pipeline stage 0's first
                # Linear): wgrad is the mask's only consumer and
                # re-reads it once per N-tile — one elementwise dz
                # pass is cheaper (A/B'd on the flagship first layer,
                # SS_DZ_FIRST=0 reverts)
                first_layer_dz = (not need_dx and B >= 4096
                                  and _dz_first_enabled())
                if wide_glds or wgrad256_ok or first_layer_dz:
                    dout = F.relu_bwd(dout, mask_src)
                    mask_src = None
        elif self.activation == "gelu":
            z = self._unstash("z", mubatch_id)
            dout = F.gelu_bwd(dout, z)
        dx = None
        if need_dx:
            dx = F.linear_dgrad(dout, self.weight.compute(),
                                self.weight.compute_t(), mask_src)
        if self._defer_wgrad:
            self._wgrad_pending.append((dout, x, mask_src))
            if len(self._wgrad_pending) >= self._wgrad_window:
                self.flush_wgrad()
        else:
            F.linear_wgrad_acc(dout, x, self.weight.grad, self.bias.grad,
                               mask_src)
        return dx

    def flush_wgrad(self):
        """Launch the deferred µbatch weight gradients as ONE chunked
        kernel (see functional.linear_wgrad_multi)."""
        if self._wgrad_pending:
            F.linear_wgrad_multi(self._wgrad_pending, self.weight.grad,
                                 self.bias.grad)
            self._wgrad_pending = []


class ReLU(Module):
    """Standalone ReLU (reference layers.py:67-80); the hot path uses
    Linear's fused epilogue instead."""

    def forward(self, inputs, mubatch_id: int = 0):
        y = F.relu_fwd(inputs)
        self._stash("y", mubatch_id, y)
        return y

    def backward(self, dout, mubatch_id: int = 0):
        y = self._unstash("y", mubatch_id)
        return F.relu_bwd(dout, y)


class GELU(Module):
    """Standalone GELU (tanh approximation) — beyond-reference module
    for modern MLP blocks; the fused path is Linear(activation="gelu")."""

    def forward(self, inputs, mubatch_id: int = 0):
        self._stash("z", mubatch_id, inputs)
        return F.gelu_fwd(inputs)

    def backward(self, dout, mubatch_id: int = 0):
        z = self._unstash("z", mubatch_id)
        return F.gelu_bwd(dout, z)


class LayerNorm(Module):
    """Rowwise LayerNorm with learnable scale/shift — beyond-reference
    module (HIP kernels: csrc/norm.hip; wave-per-row f32 statistics)."""

    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.dim, self.eps = dim, eps
        self._params["gamma"] = Parameter(torch.ones(dim))
        self._params["beta"] = Parameter(torch.zeros(dim))
        self.gamma = self._params["gamma"]
        self.beta = self._params["beta"]

    def forward(self, inputs, mubatch_id: int = 0):
        y, mean, rstd = F.layernorm_fwd(
            inputs, self.gamma.compute(), self.beta.compute(), self.eps)
        self._stash("x", mubatch_id, inputs)
        self._stash("stats", mubatch_id, (mean, rstd))
        return y

    def backward(self, dout, mubatch_id: int = 0):
        x = self._unstash("x", mubatch_id)
        mean, rstd = self._unstash("stats", mubatch_id)
        return F.layernorm_bwd(dout, x, self.gamma.compute(), mean, rstd,
                               self.gamma.grad, self.beta.grad)


class Softmax(Module):
    """Standalone row softmax (reference layers.py:83-96).  Stashes the
    OUTPUT (reference stashes input and recomputes — functional.py:31-32
    notes that as wasteful; we fix it)."""

    def forward(self, inputs, mubatch_id: int = 0):
        s = F.softmax_fwd(inputs)
        self._stash("s", mubatch_id, s)
        return s

    def backward(self, dout, mubatch_id: int = 0):
        s = self._unstash("s", mubatch_id)
        return F.softmax_bwd(dout, s)


class MSELoss(Module):
    """Standalone MSE loss head for API parity with the reference
    (layers.py:145-166): forward is the IDENTITY (the loss value is
    never computed in training, layers.py:150-155); backward takes the
    TARGET as its dout argument and emits −2(t−x)/global_batch
    (functional.py:43-44).  The fused SoftmaxMSE head below is the hot
    path; this module exists for hand-built Sequential stacks.

    GPU note: −2(t−x)/GB == (x−t)/(GB/2), so the backward reuses the
    fused xent elementwise kernel with a halved batch scale."""

    def __init__(self, global_batch_size: int):
        super().__init__()
        self.global_batch_size = global_batch_size

    def forward(self, inputs, mubatch_id: int = 0):
        self._stash("x", mubatch_id, inputs)
        return inputs

    def backward(self, target, mubatch_id: int = 0):
        x = self._unstash("x", mubatch_id)
        if target.dtype != x.dtype:
            target = target.to(x.dtype)
        return F.head_softmax_xent_bwd(x, target, self.global_batch_size / 2.0)


class _LossHead(Module):
    """Fused softmax+loss head.

    Forward emits PROBS (so the pipeline output buffer holds softmax
    probabilities, same as the reference where MSELoss.forward is the
    identity after Softmax — layers.py:150-155, and eval argmaxes the
    output buffer, train.py:40-43).  Backward takes the TARGET as its
    dout argument (reference layers.py:157-163) and emits d(logits)
    in one fused op, scaled by the GLOBAL batch size so µbatch/DP
    gradients sum to the sequential gradient (layers.py:146-148).
    """

    def __init__(self, global_batch_size: int):
        super().__init__()
        self.global_batch_size = global_batch_size

    def forward(self, inputs, mubatch_id: int = 0):
        s = F.softmax_fwd(inputs)
        self._stash("s", mubatch_id, s)
        return s

    def _bwd(self, probs, target):
        raise NotImplementedError

    def backward(self, target, mubatch_id: int = 0):
        s = self._unstash("s", mubatch_id)
        if target.dtype != s.dtype:
            target = target.to(s.dtype)
        return self._bwd(s, target)


class SoftmaxMSE(_LossHead):
    """Softmax → MSE, matching the reference's Softmax+MSELoss pair
    (layers.py:83-96,145-166) as ONE fused head."""

    def _bwd(self, probs, target):
        return F.head_softmax_mse_bwd(probs, target, self.global_batch_size)


class SoftmaxXent(_LossHead):
    """Fused softmax-cross-entropy head: dz = (s − t)/GB.  The improved
    default loss head (BASELINE.json north star)."""

    def _bwd(self, probs, target):
        return F.head_softmax_xent_bwd(probs, target, self.global_batch_size)


class Sequential(Module):
    """Layer chain with the grad-hook API.  Reference: layers.py:169-233.

    backward() reverse-chains layers and, after each layer's backward,
    fires the per-parameter grad hooks (layers.py:201-208) — the
    injection point for the DP all-reduce overlap — then the post-grad
    hooks over all parameters (layers.py:210-211).
    """

    def __init__(self, layers):
        super().__init__()
        self.layers = list(layers)
        self._grad_hooks = []
        self._post_grad_hooks = []
        # set True on pipeline stage 0: the first Linear's input grad is
        # dL/d(data) — never consumed — so its dgrad GEMM is skipped
        # (the most expensive dgrad: K = input width).
        self._skip_input_grad = False
        # set by the Worker for the temporally-LAST backward of a batch
        # under deferred wgrad: each layer's pending µbatch wgrad chunks
        # flush the moment that layer's backward completes, so its grads
        # are final MID-backward and the DP bucket all-reduce fired from
        # the grad hooks overlaps the remaining layers' dgrad/wgrad
        # kernels (reference pipe.py:302-316 semantics, restored for the
        # deferred path — round-1 deferred all grads to OptimizerStep).
        self._flush_in_backward = False

    def forward(self, inputs, mubatch_id: int = 0):
        x = inputs
        for layer in self.layers:
            x = layer.forward(x, mubatch_id)
        return x

    def backward(self, dout, mubatch_id: int = 0):
        d = dout
        for i in range(len(self.layers) - 1, -1, -1):
            layer = self.layers[i]
            if i == 0 and self._skip_input_grad and isinstance(layer, Linear):
                d = layer.backward(d, mubatch_id, need_dx=False)
            else:
                d = layer.backward(d, mubatch_id)
            # deferred-wgrad final backward: complete this layer's grads
            # NOW (before its grad hooks fire) so param_done sees final
            # values and the bucket all-reduce launches mid-backward
            if self._flush_in_backward and hasattr(layer, "flush_wgrad"):
                layer.flush_wgrad()
            for hook in self._grad_hooks:
                for p in layer.parameters():
                    if p.requires_grad:
                        hook(p)
        for hook in self._post_grad_hooks:
            hook(self.parameters())
        return d

    # hook (de)registration — reference layers.py:182-199
    def register_grad_hook(self, fn):
        self._grad_hooks.append(fn)

    def reset_grad_hooks(self):
        self._grad_hooks = []

    def register_post_grad_hook(self, fn):
        self._post_grad_hooks.append(fn)

    def reset_post_grad_hooks(self):
        self._post_grad_hooks = []

    def parameters(self):
        return [p for layer in self.layers for p in layer.parameters()]

    def zero_grad(self):
        if getattr(self, "_flat_grad", None) is not None:
            self._flat_grad.zero_()
        else:
            for layer in self.layers:
                layer.zero_grad()

    def train(self):
        self._training = True
        for l in self.layers:
            l.train()

    def eval(self):
        self._training = False
        for l in self.layers:
            l.eval()

    def set_fp8_fwd(self, flag: bool):
        """Enable the MX-fp8 training forward on every qualifying
        Linear (O%256, I%256; per-call µbatch%256 check in _fp8_ok).
        Backward stays bf16; masters stay f32."""
        n = 0
        for layer in self.layers:
            if isinstance(layer, Linear) and layer.activation in (None, "relu") \
                    and layer.out_dims % 256 == 0 and layer.in_dims % 256 == 0:
                layer.fp8_fwd = flag
                layer._wq = layer._ws = None
                n += 1
        return n

    def invalidate_fp8(self):
        """Drop cached fp8 weight copies (call after every optimizer
        step / checkpoint load so the next forward re-quantizes)."""
        for layer in self.layers:
            if getattr(layer, "_wq", None) is not None:
                layer._wq = layer._ws = None

    def set_defer_wgrad(self, flag: bool, window: int = 1 << 30):
        """Enable/disable deferred-µbatch wgrad on every Linear
        (pipeline schedules with >1 µbatch; flushed before the
        optimizer step).  window caps retained µbatches — pass the
        schedule's max_in_flight so 1F1B's activation-memory bound
        survives deferral."""
        for layer in self.layers:
            if hasattr(layer, "_defer_wgrad"):
                layer._defer_wgrad = flag
                layer._wgrad_window = max(1, window)

    def flush_wgrads(self, per_layer_hook=None):
        """Flush any REMAINING deferred weight gradients layer by layer
        in backward order; returns how many layers still had pending
        chunks.  On the normal path this is a no-op safety net — the
        final (AllReduce) backward flushes in-backward via
        _flush_in_backward; leftovers here mean a schedule never issued
        BackwardGradAllReduce."""
        n = 0
        for layer in reversed(self.layers):
            if hasattr(layer, "flush_wgrad") and layer._wgrad_pending:
                layer.flush_wgrad()
                n += 1
            if per_layer_hook is not None:
                per_layer_hook(layer)
        return n

    def materialize_device(self, device, compute_dtype=None):
        """Move to device, set compute dtype, and re-point every
        parameter grad into ONE flat f32 buffer (bucketing substrate;
        the reference's per-param MPI messages are called out as
        wasteful in its own docstring, pipe.py:309-310)."""
        device = torch.device(device)
        if compute_dtype is None:
            compute_dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
        F.clear_wgrad_tables()  # grad buffers are about to be repointed
        self.device = device
        self.compute_dtype = compute_dtype
        for layer in self.layers:
            layer.materialize_device(device, compute_dtype)
        params = [p for p in self.parameters() if p.requires_grad]
        total = sum(p.data.numel() for p in params)
        flat = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        for p in params:
            n = p.data.numel()
            p.grad = flat[off:off + n].view(p.data.shape)
            off += n
        self._flat_grad = flat
        return self


class MLP(Sequential):
    """Stage-sliced deep MLP.  Reference: layers.py:236-270.

    sizes: full list of layer boundaries.  Each pipeline stage takes its
    slice with ONE-ELEMENT OVERLAP (layers.py:247-250), builds
    Linear(..., relu) per consecutive pair with NO activation on the
    very last Linear of the last stage (layers.py:251-260), and the last
    stage appends the fused loss head (reference: Softmax+MSELoss,
    layers.py:261-263; default here: softmax-cross-entropy).
    """

    def __init__(self, sizes, stage_idx=0, n_stages=1, global_batch_size=1,
                 loss="xent"):
        assert len(sizes) % n_stages == 0, (
            f"len(sizes)={len(sizes)} must divide into {n_stages} stages"
        )
        stage_size = len(sizes) // n_stages
        lo = stage_idx * stage_size
        bounds = sizes[lo:lo + stage_size + 1]
        is_last = stage_idx == n_stages - 1
        layers = []
        for i, (a, b) in enumerate(zip(bounds[:-1], bounds[1:])):
            last_linear = is_last and i == len(bounds) - 2
            layers.append(Linear(a, b, activation=None if last_linear else "relu"))
        if is_last:
            head = {"xent": SoftmaxXent, "mse": SoftmaxMSE}[loss](global_batch_size)
            layers.append(head)
        super().__init__(layers)
        self.stage_idx, self.n_stages = stage_idx, n_stages
        self._skip_input_grad = stage_idx == 0
        self.in_dim = bounds[0]
        self.out_dim = bounds[-1] if not is_last else sizes[-1]
