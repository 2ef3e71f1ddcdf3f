"""Mid-run selector checkpointing.

The reference's unit of resume is a whole seed run (SURVEY.md section 5.4:
all durable state lives in the tracking DB). At large scales a single run
is expensive, so selectors can be checkpointed mid-run: a selector's full
state is small and explicit (for CODA: dirichlets (H,C,C), the
labeled/unlabeled bookkeeping, step counter - coda/coda.py:196-203).

Covers every built-in selector via a field registry; sharded CODA
checkpoints its LOCAL Dirichlet shard (each rank writes
<path>.rank<k> and restores its own).
"""
from __future__ import annotations

import os
from typing import Any, Dict

import torch

# selector attribute names that form the restorable state, per class name
_STATE_FIELDS = {
    "CODA": ["dirichlets", "labeled_idxs", "labels", "unlabeled_idxs",
             "q_vals", "stochastic", "step"],
    "IID": ["d_l_idxs", "d_l_ys", "d_u_idxs", "_loss_sum", "stochastic"],
    "Uncertainty": ["d_l_idxs", "d_l_ys", "d_u_idxs", "_loss_sum",
                    "stochastic"],
    "ActiveTesting": ["d_l_idxs", "d_l_ys", "d_u_idxs", "_loss_sum",
                      "losses", "qs", "M", "stochastic"],
    "VMA": ["d_l_idxs", "d_l_ys", "d_u_idxs", "_loss_sum", "losses", "qs",
            "M", "stochastic"],
    "ModelPicker": ["d_l_idxs", "d_l_ys", "d_u_idxs", "posterior",
                    "correct_counts", "stochastic"],
}


def state_dict(selector) -> Dict[str, Any]:
    cls = type(selector).__name__
    fields = _STATE_FIELDS.get(cls)
    if fields is None:
        raise ValueError(f"No checkpoint registry for selector {cls}")
    out = {"__class__": cls}
    for f in fields:
        v = getattr(selector, f)
        if torch.is_tensor(v):
            v = v.detach().cpu().clone()
        elif isinstance(v, list) and v and torch.is_tensor(v[0]):
            v = [t.detach().cpu().clone() for t in v]
        elif isinstance(v, (list, tuple)) or type(v).__name__ == "SortedList":
            v = list(v)
        out[f] = v
    return out


def load_state_dict(selector, state: Dict[str, Any]):
    cls = type(selector).__name__
    if state.get("__class__") != cls:
        raise ValueError(f"Checkpoint is for {state.get('__class__')}, "
                         f"selector is {cls}")
    device = getattr(selector, "device", "cpu")
    for f in _STATE_FIELDS[cls]:
        v = state[f]
        cur = getattr(selector, f, None)
        if torch.is_tensor(cur):
            v = v.to(device=device, dtype=cur.dtype)
        elif isinstance(v, list) and v and torch.is_tensor(v[0]):
            v = [t.to(device) for t in v]
        elif type(cur).__name__ == "SortedList":
            from sortedcontainers import SortedList
            v = SortedList(v)
        setattr(selector, f, v)
    # derived state that depends on the posterior
    if cls == "CODA":
        selector._tables = None       # force a fresh v2 table build
        selector._tables_dirty = set()
        selector._posterior_version += 1
        selector._pbest_rows_cache = (-1, None)
        selector._label_graph = None  # graph buffers alias replaced state
        selector._acq_graph = None
        selector._acq_out = None      # sized for the stale pair set
        selector._acq_fresh = False
        selector._acq_saved = None
        selector._merged_acq = False
        selector._pairs_static = None  # hit structure covers a stale set
        selector._pair_row_of = None
        selector._active_mask = None
        from sortedcontainers import SortedList
        selector._active_candidates = SortedList(
            i for i in selector.unlabeled_idxs
            if selector._disagreement_host[i])
        if getattr(selector, "_replicated", False):
            from coda_amd import ops
            a_l, b_l = ops.dirichlet_to_beta(selector.dirichlets)
            sizes = selector.comm.shard_sizes(selector.H)
            order = selector.comm.unshard_order(selector.H) \
                .to(selector.device)
            selector._alpha_g = selector.comm.all_gather_cat(
                a_l, dim=0, sizes=sizes)[order].contiguous()
            selector._beta_g = selector.comm.all_gather_cat(
                b_l, dim=0, sizes=sizes)[order].contiguous()
        selector.update_pi_hat()
    return selector


def save(selector, path: str, rank: int = 0, world: int = 1):
    if world > 1:
        path = f"{path}.rank{rank}"
    tmp = path + ".tmp"
    torch.save(state_dict(selector), tmp)
    os.replace(tmp, path)
    return path


def load(selector, path: str, rank: int = 0, world: int = 1):
    if world > 1:
        path = f"{path}.rank{rank}"
    state = torch.load(path, map_location="cpu", weights_only=False)
    return load_state_dict(selector, state)


def exists(path: str, rank: int = 0, world: int = 1) -> bool:
    if world > 1:
        path = f"{path}.rank{rank}"
    return os.path.exists(path)
