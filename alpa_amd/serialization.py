"""Checkpoint save/restore in the reference's on-disk format.

Format (mirrors ``alpa/serialization.py:75-189`` + the shard writer in
``device_mesh.py:302-338,1582-1616``):

  <ckpt_dir>/checkpoint_<step>        msgpack-packed state tree whose
                                      tensor leaves are replaced by
                                      relative directory paths
  <ckpt_dir>/<tensor_dir>/shard_<host>.<k>    .npy shard files
  <ckpt_dir>/<tensor_dir>/metadata_<host>     pickle:
      {"global_shape", "dtype", "shard_names", "shard_indices"}

"host" here is the rank (one process per GPU).  Restore reassembles each
rank's target slice from whichever saved shards overlap it — resharding
on load, so a checkpoint written under one (dp, tp) placement loads under
another (reference restore_checkpoint:137 guided by PlacementSpec).
"""
from __future__ import annotations

import os
import pickle
from typing import Any, Dict, List, Optional, Sequence, Tuple

import msgpack
import numpy as np
import torch

from .mesh import is_distributed, rank, world_size


class ShardSpec:
    """This rank's slice of a global tensor: (global_shape, index)."""

    def __init__(self, global_shape: Tuple[int, ...],
                 index: Tuple[slice, ...], is_writer: bool = True):
        self.global_shape = tuple(global_shape)
        self.index = tuple(index)
        self.is_writer = is_writer

    @staticmethod
    def full(shape, is_writer: bool = True) -> "ShardSpec":
        return ShardSpec(tuple(shape), tuple(slice(0, s) for s in shape),
                         is_writer)


def _flatten(tree: Any, prefix: str = "") -> Dict[str, Any]:
    out = {}
    if isinstance(tree, dict):
        for k, v in tree.items():
            out.update(_flatten(v, f"{prefix}{k}." if prefix or True else k))
    else:
        out[prefix[:-1]] = tree
    return out


def _tensor_dir_name(path_key: str) -> str:
    return path_key.replace("/", "_").replace(".", "_")


def _replace_leaves(tree: Any, fn) -> Any:
    if isinstance(tree, dict):
        return {k: _replace_leaves(v, fn) for k, v in tree.items()}
    return fn(tree)


def save_checkpoint(ckpt_dir: str, state_tree: Any, step: int,
                    shard_specs: Optional[Dict[str, ShardSpec]] = None
                    ) -> None:
    """Save a (possibly nested-dict) state tree.  Tensor leaves become
    shard directories; non-tensors are stored inline in the msgpack tree.

    shard_specs maps flattened leaf paths (dot-joined) to ShardSpec; absent
    leaves are treated as replicated full tensors written by rank 0 only.
    """
    shard_specs = shard_specs or {}
    host = rank()
    os.makedirs(ckpt_dir, exist_ok=True)

    flat_pos: List[str] = []

    def leaf_to_entry(path: str, leaf):
        if not torch.is_tensor(leaf):
            return leaf
        dirname = _tensor_dir_name(path)
        spec = shard_specs.get(path)
        if spec is None:
            spec = ShardSpec.full(leaf.shape, is_writer=(host == 0 or
                                                         not is_distributed()))
        tdir = os.path.join(ckpt_dir, dirname)
        os.makedirs(tdir, exist_ok=True)
        if spec.is_writer:
            # clone before .numpy(): a shared export would pin the
            # source storage (ZeRO-3 resizes param storages to 0)
            arr = leaf.detach().to(torch.float32).cpu().numpy() \
                if leaf.dtype == torch.bfloat16 \
                else leaf.detach().cpu().clone().numpy()
            shard_name = f"shard_{host}.0"
            np.save(os.path.join(tdir, shard_name + ".npy"), arr)
            meta = {
                "global_shape": spec.global_shape,
                "dtype": str(leaf.dtype).replace("torch.", ""),
                "shard_names": [shard_name],
                "shard_indices": [tuple((s.start, s.stop)
                                        for s in spec.index)],
            }
            with open(os.path.join(tdir, f"metadata_{host}"), "wb") as f:
                pickle.dump(meta, f)
        return {"__tensor_dir__": dirname}

    def walk(tree, prefix=""):
        if isinstance(tree, dict):
            return {k: walk(v, f"{prefix}{k}.") for k, v in tree.items()}
        return leaf_to_entry(prefix[:-1], tree)

    packed_tree = walk(state_tree)
    if host == 0 or not is_distributed():
        with open(os.path.join(ckpt_dir, f"checkpoint_{step}"), "wb") as f:
            f.write(msgpack.packb(packed_tree, use_bin_type=True))
    if is_distributed():
        # a checkpoint is complete only when every rank's shards (and rank
        # 0's tree) are on disk — ranks may restore immediately after
        import torch.distributed as dist
        dist.barrier()


def _load_slice(tdir: str, want_index: Tuple[slice, ...],
                global_shape, np_dtype) -> np.ndarray:
    """Assemble want_index of the global tensor from saved shards
    (resharding on load)."""
    # layout conversion: when the SAVED global shape differs from the
    # requested one but the element count matches (serial p-shaped Adam
    # moments vs ZeRO's flat per-param ranges), assemble the full stored
    # tensor and reindex through a reshape.
    metas = sorted(f for f in os.listdir(tdir) if f.startswith("metadata_"))
    if metas and global_shape is not None:
        with open(os.path.join(tdir, metas[0]), "rb") as f:
            stored_gs = tuple(pickle.load(f)["global_shape"])
        if stored_gs != tuple(global_shape) and \
                int(np.prod(stored_gs)) == int(np.prod(global_shape)):
            full = _load_slice(tdir, tuple(slice(0, s) for s in stored_gs),
                               stored_gs, np_dtype)
            return np.ascontiguousarray(
                full.reshape(tuple(global_shape))[tuple(want_index)])
    out_shape = tuple(s.stop - s.start for s in want_index)
    out = np.zeros(out_shape, dtype=np_dtype)
    filled = np.zeros(out_shape, dtype=bool) if out.size else None
    for fname in sorted(os.listdir(tdir)):
        if not fname.startswith("metadata_"):
            continue
        with open(os.path.join(tdir, fname), "rb") as f:
            meta = pickle.load(f)
        for shard_name, idx in zip(meta["shard_names"],
                                   meta["shard_indices"]):
            # intersection of shard idx with want_index
            inter = []
            src = []
            dst = []
            ok = True
            for d, ((a0, a1), w) in enumerate(zip(idx, want_index)):
                lo = max(a0, w.start)
                hi = min(a1, w.stop)
                if lo >= hi:
                    ok = False
                    break
                src.append(slice(lo - a0, hi - a0))
                dst.append(slice(lo - w.start, hi - w.start))
            if not ok:
                continue
            arr = np.load(os.path.join(tdir, shard_name + ".npy"))
            out[tuple(dst)] = arr[tuple(src)]
            if filled is not None:
                filled[tuple(dst)] = True
    if filled is not None and not filled.all():
        raise RuntimeError(f"checkpoint {tdir}: missing regions for "
                           f"{want_index}")
    return out


def restore_checkpoint(ckpt_dir: str, step: int, target_tree: Any,
                       shard_specs: Optional[Dict[str, ShardSpec]] = None
                       ) -> Any:
    """Load into `target_tree` (tensors are filled in place with this
    rank's slice per shard_specs; non-tensor leaves are replaced).
    Returns the tree."""
    shard_specs = shard_specs or {}
    with open(os.path.join(ckpt_dir, f"checkpoint_{step}"), "rb") as f:
        packed_tree = msgpack.unpackb(f.read(), raw=False)

    def walk(target, packed, prefix=""):
        if isinstance(target, dict):
            return {k: walk(v, (packed or {}).get(k), f"{prefix}{k}.")
                    for k, v in target.items()}
        path = prefix[:-1]
        if torch.is_tensor(target):
            if isinstance(packed, dict) and "__tensor_dir__" in packed:
                dirname = packed["__tensor_dir__"]
            else:
                # the tree writer (rank 0) may not own this leaf (e.g. a
                # later pipeline stage wrote its shards); fall back to the
                # deterministic directory naming
                dirname = _tensor_dir_name(path)
            tdir = os.path.join(ckpt_dir, dirname)
            assert os.path.isdir(tdir), \
                f"{path}: no checkpoint data at {tdir}"
            spec = shard_specs.get(path) or ShardSpec.full(target.shape)
            want = spec.index
            np_dtype = np.float32 if target.dtype in (torch.bfloat16,
                                                      torch.float32) \
                else target.detach().cpu().numpy().dtype
            arr = _load_slice(tdir, want, spec.global_shape, np_dtype)
            with torch.no_grad():
                target.copy_(torch.from_numpy(arr).to(target.device,
                                                      target.dtype))
            return target
        return packed

    return walk(target_tree, packed_tree)


# ----------------------------------------------------------------------
# TrainState integration
# ----------------------------------------------------------------------


def model_shard_specs(model: torch.nn.Module,
                      owner_mesh=None) -> Dict[str, ShardSpec]:
    """Derive per-parameter ShardSpecs from the parallel layer modules
    (Column/Row/VocabParallel know their global shape + slice; everything
    else is replicated, written by the global writer rank of its dp
    group)."""
    from .parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                                  VocabParallelEmbedding)
    from .parallel.expert import ExpertParallelMLP
    specs: Dict[str, ShardSpec] = {}
    for mod_name, mod in model.named_modules():
        prefix = f"{mod_name}." if mod_name else ""
        mesh = getattr(mod, "mesh", None)
        axis = getattr(mod, "axis", 1)
        tp = mesh.axis_size(axis) if mesh is not None else 1
        idx = mesh.axis_index(axis) if (mesh is not None and mesh.is_member) \
            else 0
        # dp writer: only the first rank along every non-tp axis writes
        dp_writer = True
        if mesh is not None and mesh.is_member:
            other = 1 - axis
            dp_writer = mesh.axis_index(other) == 0
        if isinstance(mod, ColumnParallelLinear) and tp > 1:
            o = mod.out_per_rank
            specs[prefix + "weight"] = ShardSpec(
                (mod.out_features, mod.in_features),
                (slice(idx * o, (idx + 1) * o), slice(0, mod.in_features)),
                dp_writer)
            if mod.bias is not None:
                specs[prefix + "bias"] = ShardSpec(
                    (mod.out_features,), (slice(idx * o, (idx + 1) * o),),
                    dp_writer)
        elif isinstance(mod, RowParallelLinear) and tp > 1:
            i = mod.in_per_rank
            specs[prefix + "weight"] = ShardSpec(
                (mod.out_features, mod.in_features),
                (slice(0, mod.out_features), slice(idx * i, (idx + 1) * i)),
                dp_writer)
            if mod.bias is not None:
                specs[prefix + "bias"] = ShardSpec(
                    (mod.out_features,), (slice(0, mod.out_features),),
                    dp_writer and idx == 0)
        elif isinstance(mod, ExpertParallelMLP) and mod.ep > 1:
            # expert weights: dim-0 shards of the global [E, ...] stack
            for wname, w in (("w1", mod.w1), ("w2", mod.w2)):
                gshape = (mod.E,) + tuple(w.shape[1:])
                idx = (slice(mod.e_start, mod.e_start + mod.e_local),) + \
                    tuple(slice(0, s) for s in w.shape[1:])
                specs[prefix + wname] = ShardSpec(gshape, idx, dp_writer)
        elif isinstance(mod, VocabParallelEmbedding) and tp > 1:
            v = mod.vocab_per_rank
            emb = mod.weight.shape[1]
            specs[prefix + "weight"] = ShardSpec(
                (mod.num_embeddings, emb),
                (slice(idx * v, (idx + 1) * v), slice(0, emb)), dp_writer)
    # default for unsharded params: replicated across the OWNING mesh
    # (the stage mesh for pipeline states) — its first rank writes
    for name, p in model.named_parameters():
        if name not in specs:
            if owner_mesh is not None:
                writer = owner_mesh.is_member and \
                    rank() == owner_mesh.ranks[0]
            else:
                writer = rank() == 0 or not is_distributed()
            specs[name] = ShardSpec.full(p.shape, writer)
    return specs


def _train_state_tree_and_specs(state):
    """(tree, specs) for a TrainState: params + Adam moments.

    Raw AdamW moments share the parameter sharding (2-D specs, TP-
    reshardable).  ZeRO-2/3 moments are bucket/block flat shards: they are
    saved per parameter as 1-D ranges of the param's own flat index space
    (ShardSpec over (numel,)), which reshards correctly across dp sizes
    and — via the layout-conversion fallback in ``_load_slice`` — to and
    from the serial p-shaped layout.  TP-sharded params under ZeRO get a
    ``@tp<offsets>`` key qualifier (per-tp-rank flat spaces are disjoint);
    such moments restore only under the same tp topology.
    """
    specs = model_shard_specs(state.model, getattr(state, "mesh", None))
    tree = {"params": dict(state.model.state_dict()),
            "step": state.step_count}
    full_specs = {f"params.{k}": v for k, v in specs.items()}
    name_of = {id(p): n for n, p in state.model.named_parameters()}
    opt = state.optimizer

    def add_dense(adam):
        m_tree = tree.setdefault("opt_m", {})
        v_tree = tree.setdefault("opt_v", {})
        for p, m, v in zip(adam.params, adam.exp_avgs, adam.exp_avg_sqs):
            n = name_of.get(id(p))
            if n is None:
                continue
            m_tree[n], v_tree[n] = m, v
            full_specs[f"opt_m.{n}"] = specs[n]
            full_specs[f"opt_v.{n}"] = specs[n]

    def add_sharded(z):
        m_tree = tree.setdefault("opt_m", {})
        v_tree = tree.setdefault("opt_v", {})
        for (p, lo, hi, m, v) in z.moment_slices():
            n = name_of.get(id(p))
            if n is None:
                continue
            key = n
            sp = specs.get(n)
            if sp is not None and sp.index != tuple(
                    slice(0, s) for s in sp.global_shape):
                starts = "_".join(str(s.start) for s in sp.index)
                key = f"{n}@tp{starts}"
            m_tree[key], v_tree[key] = m, v
            fsp = ShardSpec((p.numel(),), (slice(lo, hi),), True)
            full_specs[f"opt_m.{key}"] = fsp
            full_specs[f"opt_v.{key}"] = fsp

    if opt is None:
        pass
    elif hasattr(opt, "moment_slices"):     # ZeroOptimizer (ZeRO-2)
        add_sharded(opt)
    elif hasattr(opt, "z3"):                # ZeRO-3 composite
        add_sharded(opt.z3)
        if opt.rest_opt is not None:
            add_dense(opt.rest_opt)
    elif hasattr(opt, "exp_avgs") and hasattr(opt, "params"):
        add_dense(opt)                      # raw AdamW (incl. pipeline)
    return tree, full_specs


def _maybe_materialized(state):
    """ZeRO-3 states keep param storages released between uses; gather
    them around checkpoint IO (and write shards back on exit)."""
    mgr = getattr(state, "zero3_manager", None)
    if mgr is not None:
        return mgr.materialized()
    import contextlib
    return contextlib.nullcontext()


def save_train_state(ckpt_dir: str, state, step: Optional[int] = None
                     ) -> None:
    """Save model + optimizer of a TrainState (reference
    save_checkpoint, serialization.py:75)."""
    step = step if step is not None else state.step_count
    with _maybe_materialized(state):
        tree, full_specs = _train_state_tree_and_specs(state)
        save_checkpoint(ckpt_dir, tree, step, full_specs)


def restore_train_state(ckpt_dir: str, state, step: int) -> None:
    """Restore in place, resharding to this rank's placement (reference
    restore_checkpoint, serialization.py:137)."""
    with _maybe_materialized(state):
        tree, full_specs = _train_state_tree_and_specs(state)
        loaded = restore_checkpoint(ckpt_dir, step, tree, full_specs)
    if isinstance(loaded.get("step"), int):
        st = loaded["step"]
        state.optimizer.step_count = st
        # ZeRO-3 composite: bias correction is driven by the nested
        # optimizers' own counters
        for sub in ("z3", "rest_opt"):
            o = getattr(state.optimizer, sub, None)
            if o is not None:
                o.step_count = st
        state.step_count = st
