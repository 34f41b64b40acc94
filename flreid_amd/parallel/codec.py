"""Tensor codec for client-state synchronisation (C2/C3 in SURVEY.md §2.9).

Upload states are nested python dicts of tensors + scalars.  The generic
`all_gather_object` path pickles ~100 MB of parameters per client through
host memory; this codec instead:

  1. exchanges a tiny metadata skeleton (names, shapes, dtypes, scalars)
     via object gather;
  2. packs every client's tensors into ONE flat device buffer (sorted key
     order), pads ranks to the same client count, and runs a single RCCL
     `all_gather_into_tensor` over xGMI;
  3. rebuilds per-client states as DEVICE tensors on every rank, so server
     aggregation stays at HBM bandwidth.

Falls back to the object path when client schemas differ across ranks (e.g.
methods with data-dependent state shapes).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Tuple

import torch

_SENTINEL = "__flreid_tensor__"
_SCALAR = "__flreid_scalar__"

# steady-state schema cache: the full skeleton gather (an O(world_size)
# object serialisation) runs only when some rank's upload STRUCTURE changes
# (first contact, task switch growing a state, ...).  Per round, only a
# fingerprint + the scalar leaves (train_cnt etc.) ride the object hop.
_META_CACHE: Dict[str, Any] = {"fps": None, "parts": None}


def reset_schema_cache() -> None:
    _META_CACHE["fps"] = None
    _META_CACHE["parts"] = None


def _split_scalars(skeleton: Any, values: List[Any]) -> Any:
    """Replace every non-tensor leaf with a numbered placeholder, collecting
    the values — the structure half is cacheable across rounds, the values
    half changes every round."""
    if isinstance(skeleton, tuple) and len(skeleton) == 3 and skeleton[0] == _SENTINEL:
        return skeleton
    if isinstance(skeleton, dict):
        return {k: _split_scalars(v, values) for k, v in skeleton.items()}
    if isinstance(skeleton, (list, tuple)):
        out = [_split_scalars(v, values) for v in skeleton]
        return tuple(out) if isinstance(skeleton, tuple) else out
    values.append(skeleton)
    return (_SCALAR, len(values) - 1)


def _subst_scalars(structure: Any, values: List[Any]) -> Any:
    if isinstance(structure, tuple) and len(structure) == 2 and structure[0] == _SCALAR:
        return values[structure[1]]
    if isinstance(structure, tuple) and len(structure) == 3 and structure[0] == _SENTINEL:
        return structure
    if isinstance(structure, dict):
        return {k: _subst_scalars(v, values) for k, v in structure.items()}
    if isinstance(structure, (list, tuple)):
        out = [_subst_scalars(v, values) for v in structure]
        return tuple(out) if isinstance(structure, tuple) else out
    return structure


def _wire_dtype() -> torch.dtype:
    """Wire format for the flat gather: fp32 (default, bit-transparent) or
    bf16 (FLREID_COMM_DTYPE=bf16 — halves xGMI traffic at the cost of one
    bf16 round-trip on the exchanged states)."""
    return (torch.bfloat16 if os.environ.get("FLREID_COMM_DTYPE", "fp32")
            == "bf16" else torch.float32)


def _flatten(state: Any, path: str, tensors: List[Tuple[str, torch.Tensor]]):
    """Returns a skeleton with tensors replaced by (sentinel, shape, dtype)."""
    if torch.is_tensor(state):
        tensors.append((path, state))
        return (_SENTINEL, tuple(state.shape), str(state.dtype))
    if isinstance(state, dict):
        return {k: _flatten(state[k], f"{path}.{k}", tensors)
                for k in sorted(state.keys(), key=str)}
    if isinstance(state, (list, tuple)):
        out = [_flatten(v, f"{path}[{i}]", tensors) for i, v in enumerate(state)]
        return tuple(out) if isinstance(state, tuple) else out
    return state


def _rebuild(skeleton: Any, chunks: List[torch.Tensor], pos: List[int]) -> Any:
    if isinstance(skeleton, tuple) and len(skeleton) == 3 and skeleton[0] == _SENTINEL:
        t = chunks[pos[0]]
        pos[0] += 1
        return t
    if isinstance(skeleton, dict):
        return {k: _rebuild(v, chunks, pos) for k, v in skeleton.items()}
    if isinstance(skeleton, (list, tuple)):
        out = [_rebuild(v, chunks, pos) for v in skeleton]
        return out if isinstance(skeleton, list) else tuple(out)
    return skeleton


def _dtype_of(name: str) -> torch.dtype:
    return getattr(torch, name.replace("torch.", ""))


def sync_client_states(ctx, local_uploads: Dict[str, Any]) -> Dict[str, Any]:
    """All ranks end with the union of every rank's {client: state}."""
    if not ctx.is_distributed:
        return dict(local_uploads)

    device = ctx._comm_device()

    # 1. flatten locally
    metas: Dict[str, Any] = {}
    flats: Dict[str, torch.Tensor] = {}
    local_has_nonfloat = False
    for cname in sorted(local_uploads.keys()):
        tensors: List[Tuple[str, torch.Tensor]] = []
        skeleton = _flatten(local_uploads[cname], cname, tensors)
        # integer/bool tensors must not round-trip through the float wire
        # dtype (precision loss above 2^24 in fp32, far sooner in bf16):
        # such states take the object-gather fallback below
        local_has_nonfloat = local_has_nonfloat or any(
            not t.is_floating_point() for _n, t in tensors)
        wire = _wire_dtype()
        if tensors and not local_has_nonfloat:
            flat = torch.cat([t.detach().reshape(-1).to(wire)
                              for _n, t in tensors]).to(device)
        else:
            flat = torch.zeros(0, dtype=wire, device=device)
        metas[cname] = skeleton
        flats[cname] = flat

    # 2. metadata hop: per round only a schema fingerprint + the scalar
    # leaves are serialised; the full skeleton gather runs on schema change
    import hashlib
    import pickle

    per_client_sizes = {c: f.numel() for c, f in flats.items()}
    structures: Dict[str, Any] = {}
    scalars: Dict[str, List[Any]] = {}
    for cname, skeleton in metas.items():
        vals: List[Any] = []
        structures[cname] = _split_scalars(skeleton, vals)
        scalars[cname] = vals
    local_fp = hashlib.sha1(pickle.dumps(
        (structures, per_client_sizes, local_has_nonfloat))).hexdigest()

    small = ctx.all_gather_object((local_fp, scalars))
    fps = [fp for fp, _sc in small]
    if _META_CACHE["fps"] == fps:
        all_parts = _META_CACHE["parts"]
    else:
        all_parts = ctx.all_gather_object(
            (structures, per_client_sizes, local_has_nonfloat))
        _META_CACHE["fps"] = fps
        _META_CACHE["parts"] = all_parts

    all_metas = []
    for rank, (structures_r, sizes_r, nf_r) in enumerate(all_parts):
        scal_r = small[rank][1]
        metas_r = {c: _subst_scalars(s, scal_r[c])
                   for c, s in structures_r.items()}
        all_metas.append((metas_r, sizes_r, nf_r))

    strides = {n for _m, sizes, _nf in all_metas for n in sizes.values()}
    counts = [len(sizes) for _m, sizes, _nf in all_metas]
    any_nonfloat = any(nf for _m, _s, nf in all_metas)

    # equal-stride fast path: every client's payload has the same numel
    if len(strides) == 1 and any(counts) and not any_nonfloat:
        stride = next(iter(strides))
        max_clients = max(counts)
        local = torch.zeros(max_clients * stride, dtype=_wire_dtype(),
                            device=device)
        for i, cname in enumerate(sorted(flats.keys())):
            local[i * stride:(i + 1) * stride] = flats[cname]
        gathered = ctx.all_gather_flat(local)          # [W, max_clients*stride]

        out: Dict[str, Any] = {}
        for rank, (rank_metas, rank_sizes, _nf) in enumerate(all_metas):
            for i, cname in enumerate(sorted(rank_sizes.keys())):
                flat = gathered[rank, i * stride:(i + 1) * stride]
                out[cname] = _unflatten_state(rank_metas[cname], flat)
        return out

    # fallback: ragged schemas -> object path (host round trip)
    gathered_obj = ctx.all_gather_object({c: _to_cpu(s) for c, s in local_uploads.items()})
    merged: Dict[str, Any] = {}
    for rank_uploads in gathered_obj:
        merged.update(rank_uploads)
    return merged


def _unflatten_state(skeleton: Any, flat: torch.Tensor) -> Any:
    chunks: List[torch.Tensor] = []
    offset = 0

    def walk(node):
        nonlocal offset
        if isinstance(node, tuple) and len(node) == 3 and node[0] == _SENTINEL:
            shape, dtype = node[1], _dtype_of(node[2])
            numel = 1
            for s in shape:
                numel *= s
            chunks.append(flat[offset:offset + numel].view(shape).to(dtype))
            offset += numel
            return
        if isinstance(node, dict):
            for v in node.values():
                walk(v)
        elif isinstance(node, (list, tuple)):
            for v in node:
                walk(v)

    walk(skeleton)
    return _rebuild(skeleton, chunks, [0])


def _to_cpu(state: Any) -> Any:
    if torch.is_tensor(state):
        return state.detach().cpu()
    if isinstance(state, dict):
        return {k: _to_cpu(v) for k, v in state.items()}
    if isinstance(state, (list, tuple)):
        out = [_to_cpu(v) for v in state]
        return out if isinstance(state, list) else tuple(out)
    return state
