"""Stateless math ops, dual-backend.

Reference: shallowspeed/functional.py:4-44 — 8 pure NumPy functions
(linear/relu/softmax/mse + grads).  Here every op has

  * a CPU reference implementation in plain torch float32 — used by
    host-only tests and as the numerics oracle for the HIP kernels, and
  * a GPU implementation dispatching to the in-tree HIP/CDNA4 extension
    (csrc/) — MFMA bf16 GEMMs, fused epilogues.  On a CUDA/HIP device
    the extension is REQUIRED: there is no silent torch fallback.

Differences from the reference, by design:
  * reference softmax uses a GLOBAL max and a +1e-7 denominator fudge
    (functional.py:26); we use the correct per-row max and exact rowsum.
  * the backward of the loss head is fused: softmax∘MSE and
    softmax∘cross-entropy each produce d(logits) in one op instead of
    chaining mse_loss_grad (functional.py:43-44) through softmax_grad
    (functional.py:30-35).
"""

import torch

from ._ext import load_ext


_DETERMINISTIC = False


def set_deterministic(flag: bool):
    """Bitwise run-to-run reproducible GPU training.

    Forces the single-owner (split_k=1) wgrad path so f32 atomicAdd
    ARRIVAL ORDER cannot perturb gradient bits; all other hot kernels
    (fwd/dgrad GEMMs, fused SGD, loss heads) are deterministic by
    construction (fixed tile ownership, no atomics), and the DP bucket
    all-reduce order is fixed by construction in GradReducer.  Matches
    the reference's exact-equality replica gate (train.py:154-155) at
    the bitwise level on GPU.  Costs wgrad parallelism on small-M
    shapes (split-K is what fills the chip there) — opt-in via
    --deterministic."""
    global _DETERMINISTIC
    _DETERMINISTIC = bool(flag)


def deterministic() -> bool:
    return _DETERMINISTIC


def _is_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


def _ext_for(t: torch.Tensor):
    return load_ext(required=True) if _is_gpu(t) else None


# ---------------------------------------------------------------- linear

def linear_fwd(x, w, b=None, relu=False):
    """y = x @ W^T (+ b) (+ ReLU).

    x: (M, K); w: (N, K) row-major ("NT" GEMM — both operands row-major
    with contiguous K, the natural MFMA layout); b: (N,) or None.
    Reference: functional.py:13-17 with the module-level fused ReLU of
    layers.py:120-122 moved into the GEMM epilogue.
    """
    if _is_gpu(x):
        ext = _ext_for(x)
        empty = torch.Tensor()
        return ext.gemm_nt(x, w, b if b is not None else empty, empty, bool(relu))
    y = x @ w.t()
    if b is not None:
        y = y + b
    if relu:
        y = torch.clamp(y, min=0)
    return y


def fp8_quantize(x):
    """bf16 (R, K) -> (e4m3 u8 data (R, K), E8M0 scales (K/128, R, 4)).

    MX block quantization matching the gfx950 scaled-MFMA's hardware
    scale-group partition (csrc/fp8.hip; probes in
    scripts/probe_mx*.hip).  GPU only; K % 128 == 0.
    """
    return _ext_for(x).fp8_quantize(x)


def linear_fwd_fp8(xq, xs, wq, ws, b=None, relu=False):
    """y = dequant(xq) @ dequant(wq)^T (+ b)(+ ReLU) on the MX-fp8
    scaled-MFMA tier (~1.7x the bf16 8-phase kernel).  Opt-in serving
    precision (beyond the reference's scope); shapes must satisfy
    M%256 == N%256 == 0, K%256 == 0."""
    ext = _ext_for(xq)
    empty = torch.Tensor()
    return ext.gemm_nt_f8(xq, xs, wq, ws, b if b is not None else empty,
                          bool(relu))


def linear_dgrad(dy, w, w_t=None, mask_src=None):
    """dx = (dy ⊙ 1[mask_src>0]) @ W.

    The ReLU backward (reference functional.py:8-10, layers.py:75) is
    fused into the GEMM A-operand read: when mask_src (the stashed
    post-ReLU output of THIS layer) is given, dy elements where
    mask_src<=0 are zeroed during LDS staging.

    GPU path multiplies by the transposed bf16 weight copy w_t ((K, N)
    row-major) so the kernel is the same NT GEMM as the forward;
    w_t is maintained by the fused SGD step.
    Reference: functional.py:20-21 first return value.
    """
    if _is_gpu(dy):
        ext = _ext_for(dy)
        assert w_t is not None, "GPU dgrad needs the transposed weight copy"
        # Tiered dispatch: a PLAIN wide dgrad (no mask, no epilogue) is
        # exactly the "plain library GEMM" case where hipBLASLt belongs
        # — measured 1362-1665 TF vs our 256-tile kernel's 1155-1305 at
        # >=1024-wide shapes (scripts/test_gemm256.py ladder); at the
        # flagship 256-wide shapes our kernel wins and keeps the lane.
        # Fused ops (bias/ReLU epilogues, mask prologue, wgrad's f32
        # atomic accumulate) always stay on the HIP kernels.
        M, O = dy.shape
        if (mask_src is None and not _DETERMINISTIC
                and M >= 2048 and O >= 1024 and w.shape[1] >= 1024
                and _dgrad_lib_enabled()):
            # NT layout: w_t is (I, O) row-major (K=O contiguous), so
            # w_t.t() presents hipBLASLt the transposed-B problem — the
            # fast MFMA-native layout (NN with a row-major B measured
            # only ~+1% end-to-end; NT is the library's 1.4-1.66 PF lane)
            return torch.matmul(dy, w_t.t())
        empty = torch.Tensor()
        return ext.gemm_nt(dy, w_t, empty, mask_src if mask_src is not None else empty, False)
    if mask_src is not None:
        dy = dy * (mask_src > 0).to(dy.dtype)
    return dy @ w


_DGRAD_LIB = None


def _dgrad_lib_enabled() -> bool:
    global _DGRAD_LIB
    if _DGRAD_LIB is None:
        import os

        _DGRAD_LIB = os.environ.get("SS_DGRAD_LIB", "1") == "1"
    return _DGRAD_LIB


def linear_wgrad_acc(dy, x, grad_w, grad_b=None, mask_src=None, split_k=0):
    """grad_w += (dy ⊙ mask)^T @ x ; grad_b += colsum(dy ⊙ mask).

    Accumulates IN PLACE into f32 grad buffers — this is how both
    µbatch gradient accumulation (reference layers.py:135-136) and
    split-K parallelism work: the GPU kernel uses f32 atomicAdd, so a
    K-split over the batch dimension and accumulation across µbatches
    compose for free.  split_k=0 lets the kernel pick; split_k=1 is the
    deterministic single-owner path (forced by set_deterministic).
    Reference: functional.py:21 (dW = dout.T @ x, db = dout.sum(0)).
    """
    if split_k == 0 and _DETERMINISTIC:
        split_k = 1
    if _is_gpu(dy):
        ext = _ext_for(dy)
        ext.wgrad_tn(
            dy,
            x,
            grad_w,
            grad_b if grad_b is not None else torch.Tensor(),
            mask_src if mask_src is not None else torch.Tensor(),
            int(split_k),
        )
        return
    if mask_src is not None:
        dy = dy * (mask_src > 0).to(dy.dtype)
    grad_w += (dy.t() @ x).to(grad_w.dtype)
    if grad_b is not None:
        grad_b += dy.sum(dim=0).to(grad_b.dtype)


# ---------------------------------------------------------------- relu

def relu_fwd(x):
    """Standalone ReLU (reference functional.py:4-5); the hot path uses
    the fused GEMM epilogue instead."""
    if _is_gpu(x):
        return _ext_for(x).relu_fwd(x)
    return torch.clamp(x, min=0)


def relu_bwd(dy, y):
    """dx = dy ⊙ 1[y>0] from the stashed ReLU OUTPUT (for relu,
    out>0 ⟺ in>0, so stashing the output replaces the reference's
    bitmask stash at layers.py:70)."""
    if _is_gpu(dy):
        return _ext_for(dy).relu_bwd(dy, y)
    return dy * (y > 0).to(dy.dtype)


# ---------------------------------------------------------------- softmax

def softmax_fwd(x):
    """Row softmax with per-row max subtraction.

    Reference: functional.py:24-27 (which uses a global max and +1e-7;
    we use the numerically correct form — divergence documented)."""
    if _is_gpu(x):
        return _ext_for(x).softmax_fwd(x)
    m = x.max(dim=-1, keepdim=True).values
    e = torch.exp(x - m)
    return e / e.sum(dim=-1, keepdim=True)


def softmax_bwd(dy, s):
    """ds/dx given softmax OUTPUT s (we stash the output; the reference
    stashes the input and recomputes, noted wasteful at
    functional.py:31-32).  dx = s*(dy - rowsum(s*dy))."""
    if _is_gpu(dy):
        return _ext_for(dy).softmax_bwd(dy, s)
    dot = (s * dy).sum(dim=-1, keepdim=True)
    return s * (dy - dot)


# ---------------------------------------------------------------- loss heads

def mse_loss(x, t, global_batch):
    """Σ(t−x)²/global_batch (reference functional.py:38-40; only used
    by tests/logging — training fwd is identity, layers.py:150-155)."""
    return ((t - x) ** 2).sum() / global_batch


def head_softmax_mse_bwd(probs, target, global_batch):
    """Fused backward of Softmax→MSELoss w.r.t. logits.

    g = −2(t−s)/GB (mse_loss_grad, functional.py:43-44), then the
    softmax Jacobian (functional.py:30-35):
      dz = s ⊙ (g − rowsum(s ⊙ g)).
    global_batch is the GLOBAL batch size so µbatch/DP gradients sum to
    the sequential gradient (reference layers.py:146-148).
    """
    if _is_gpu(probs):
        return _ext_for(probs).head_mse_bwd(probs, target, float(global_batch))
    g = -2.0 * (target - probs) / global_batch
    dot = (probs * g).sum(dim=-1, keepdim=True)
    return probs * (g - dot)


def head_softmax_xent_bwd(probs, target, global_batch):
    """Fused backward of softmax cross-entropy: dz = (s − t)/GB."""
    if _is_gpu(probs):
        return _ext_for(probs).head_xent_bwd(probs, target, float(global_batch))
    return (probs - target) / global_batch


def xent_loss(probs, target, global_batch, eps=1e-9):
    """−Σ t·log(p)/GB (logging only)."""
    return -(target * torch.log(probs.float() + eps)).sum() / global_batch


# ----------------------------------------------- layernorm / gelu (extras)

def layernorm_fwd(x, gamma, beta, eps=1e-5):
    """Rowwise LayerNorm.  Returns (y, mean, rstd) — the f32 row stats
    feed the backward.  Beyond-reference extension (modern-MLP blocks);
    HIP kernel: csrc/norm.hip."""
    if _is_gpu(x):
        return tuple(_ext_for(x).ln_fwd(x, gamma, beta, float(eps)))
    xf = x.float()
    mu = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    rstd = (var + eps).rsqrt()
    y = ((xf - mu) * rstd * gamma.float() + beta.float()).to(x.dtype)
    return y, mu.squeeze(-1), rstd.squeeze(-1)


def layernorm_bwd(dy, x, gamma, mean, rstd, dgamma, dbeta):
    """dx plus in-place accumulation of dγ/dβ (f32 grads)."""
    if _is_gpu(dy):
        e = _ext_for(dy)
        dx = e.ln_bwd_dx(dy, x, gamma, mean, rstd)
        e.ln_bwd_dparam(dy, x, mean, rstd, dgamma, dbeta)
        return dx
    xf, dyf = x.float(), dy.float()
    xh = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    g = dyf * gamma.float()
    C = x.shape[-1]
    dx = rstd.unsqueeze(-1) * (
        g - g.mean(-1, keepdim=True) - xh * (g * xh).mean(-1, keepdim=True))
    dgamma += (dyf * xh).sum(0)
    dbeta += dyf.sum(0)
    return dx.to(dy.dtype)


def gelu_fwd(z):
    """tanh-approximation GELU (elementwise HIP kernel on GPU)."""
    if _is_gpu(z):
        return _ext_for(z).gelu_fwd(z)
    return torch.nn.functional.gelu(z.float(), approximate="tanh").to(z.dtype)


def gelu_bwd(dy, z):
    if _is_gpu(dy):
        return _ext_for(dy).gelu_bwd(dy, z)
    zf = z.float().detach().requires_grad_(True)
    y = torch.nn.functional.gelu(zf, approximate="tanh")
    y.backward(dy.float())
    return zf.grad.to(dy.dtype)


from collections import OrderedDict  # noqa: E402

# LRU-capped cache of (pinned-host, device) pointer tables for the
# chunked wgrad kernel, keyed by (grad_w ptr, n_chunks).  The cap plus
# clear_wgrad_tables() (called from Sequential.materialize_device)
# keeps long-lived processes that build/destroy many models from
# leaking pinned+device entries; a freed-then-reallocated grad buffer
# at the same address is safe regardless (rows are re-checked), the
# eviction is purely about memory growth.
_WGRAD_TABLES = OrderedDict()
_WGRAD_TABLES_CAP = 64


def clear_wgrad_tables():
    _WGRAD_TABLES.clear()


def linear_wgrad_multi(chunks, grad_w, grad_b=None, split_k=0):
    """Deferred-µbatch weight gradients: ONE kernel launch accumulates
    every (dy, x[, mask]) chunk of a layer into grad_w/grad_b.

    chunks: list of (dy, x, mask_or_None) with identical shapes.
    Replaces num_µbatches separate wgrad launches under pipeline
    schedules (each launch re-pays prologue + atomic epilogue).
    GPU only; the CPU path loops (it has no launch cost to amortize).
    """
    if split_k == 0 and _DETERMINISTIC:
        split_k = 1
    dy0 = chunks[0][0]
    if not _is_gpu(dy0) or split_k == 1:
        # split_k==1 (deterministic mode): the chunked kernel would
        # still run one block PER CHUNK per tile, all atomicAdd-ing
        # the same gW element — chunk ARRIVAL order is scheduling-
        # dependent, which broke bitwise reproducibility (flaky GPU
        # determinism test).  Sequential per-chunk launches fix the
        # accumulation order at a small launch-count cost; perf mode
        # (split_k=0) keeps the single fused launch.
        for dy, x, m in chunks:
            linear_wgrad_acc(dy, x, grad_w, grad_b, m, split_k)
        return
    ext = _ext_for(dy0)
    has_mask = chunks[0][2] is not None
    # Pinned-host staging + cached device table: the H2D becomes an
    # async (capture-legal) copy, so the flush works INSIDE hipGraph
    # capture — the pinned buffer stays alive with stable content, and
    # replays re-copy it (chunk pointers are stable under the graph
    # memory pool).
    rows = []
    for dy, x, m in chunks:
        assert dy.shape == dy0.shape and (m is not None) == has_mask
        rows.append(dy.data_ptr())
        rows.append(x.data_ptr())
        rows.append(m.data_ptr() if m is not None else 0)
    key = (grad_w.data_ptr(), len(chunks))
    entry = _WGRAD_TABLES.get(key)
    if entry is None:
        while len(_WGRAD_TABLES) >= _WGRAD_TABLES_CAP:
            _WGRAD_TABLES.popitem(last=False)
        pinned = torch.empty(len(chunks), 3, dtype=torch.int64,
                             pin_memory=True)
        dev = torch.empty(len(chunks), 3, dtype=torch.int64,
                          device=dy0.device)
        entry = _WGRAD_TABLES[key] = [pinned, dev, None]
    else:
        _WGRAD_TABLES.move_to_end(key)
    pinned, table, last_rows = entry
    if rows != last_rows:
        # chunk pointers are usually STABLE step to step (caching
        # allocator steady state / graph pool), so the host-side table
        # rebuild + H2D is skipped on the hot path
        pinned.view(-1).copy_(torch.tensor(rows, dtype=torch.int64))
        table.copy_(pinned, non_blocking=True)
        entry[2] = rows
    ext.wgrad_tn_multi(table, len(chunks), has_mask, grad_w,
                       grad_b if grad_b is not None else torch.Tensor(),
                       dy0.shape[1], chunks[0][1].shape[1], dy0.shape[0],
                       int(split_k))


def row_argmax(x):
    """Per-row argmax (eval/serving).  torch's ROCm argmax on skinny
    bf16 tensors is pathologically slow (~1.3 ms at 16384x10); the HIP
    wave-per-row kernel is ~5 µs.  CPU falls back to torch."""
    if _is_gpu(x) and x.dtype == torch.bfloat16:
        return _ext_for(x).row_argmax(x).long()
    return x.argmax(dim=-1)
