"""Stage-sharded checkpoint save/load.

The reference library has no checkpointing; only its side script saves
a state_dict pickle (scripts/DDP_PyTorch_MNIST.py:157-161, naming
precedent `model_p{size}.pkl`).  BASELINE.json requires a checkpoint
layout, so: each pipeline stage writes its own file, written only by
DP rank 0 of that stage (every DP replica is identical — enforced by
assert_sync).  Layout:

    <dir>/meta.pt                      (world/dp/pp, sizes, step)
    <dir>/stage_{s:02d}.pt             (f32 master params + optimizer
                                        state — momentum velocities —
                                        in parameter order)

Resuming a momentum>0 run restores the exact optimization trajectory
(velocities + hyperparameters are saved with the stage shard); loading
validates the checkpoint's PP/TP topology against the current one so a
repartitioned resume fails loudly instead of via per-tensor shape
asserts alone.  TP shards write one file per TP rank
(stage_XX_tpYY.pt).  If ranks load a checkpoint IMMEDIATELY after a
collective save in the same run, put a dist.barrier() between save and
load — rank 0 writes meta.pt and other ranks must not race it (the
normal save-at-exit / resume-at-start flow never hits this).
"""

import os

import torch


def _shard_name(topo):
    """stage_{s}.pt, plus a _tp{t} suffix when tensor-parallel shards
    exist (every TP rank owns different weights)."""
    name = f"stage_{topo.stage_id:02d}"
    if getattr(topo, "tp", 1) > 1:
        name += f"_tp{topo.tp_rank:02d}"
    return name + ".pt"


def save_checkpoint(path, model, topo, step: int = 0, extra=None,
                    optimizer=None):
    os.makedirs(path, exist_ok=True)
    if topo.dp_rank == 0:
        state = {
            "params": [p.data.detach().cpu() for p in model.parameters()],
            "stage_id": topo.stage_id,
            "step": step,
        }
        if optimizer is not None and hasattr(optimizer, "state_dict"):
            opt_state = optimizer.state_dict()
            # stateless SGD (no momentum): nothing worth persisting
            if any(k in opt_state for k in ("velocity", "exp_avg")):
                state["opt"] = opt_state
        torch.save(state, os.path.join(path, _shard_name(topo)))
    if topo.rank == 0:
        meta = {"dp": topo.dp, "pp": topo.pp,
                "tp": getattr(topo, "tp", 1), "step": step}
        if extra:
            meta.update(extra)
        torch.save(meta, os.path.join(path, "meta.pt"))


def load_checkpoint(path, model, topo, optimizer=None):
    meta_f = os.path.join(path, "meta.pt")
    meta = torch.load(meta_f, map_location="cpu", weights_only=False) \
        if os.path.exists(meta_f) else {}
    if "pp" in meta:
        assert meta["pp"] == topo.pp, (
            f"checkpoint was written with pp={meta['pp']}; current "
            f"topology has pp={topo.pp} — stage shards do not repartition"
        )
    if meta.get("tp", 1) != getattr(topo, "tp", 1):
        raise AssertionError(
            f"checkpoint was written with tp={meta.get('tp', 1)}; current "
            f"topology has tp={getattr(topo, 'tp', 1)} — TP shards do "
            f"not repartition")
    f = os.path.join(path, _shard_name(topo))
    state = torch.load(f, map_location="cpu", weights_only=False)
    params = model.parameters()
    assert len(params) == len(state["params"]), (
        f"checkpoint has {len(state['params'])} tensors, "
        f"model stage has {len(params)}"
    )
    for p, saved in zip(params, state["params"]):
        assert p.data.shape == saved.shape, (p.data.shape, saved.shape)
        p.data.copy_(saved.to(p.data.device))
        p.sync_lp()
    if optimizer is not None and "opt" in state:
        optimizer.load_state_dict(state["opt"])
    return meta
