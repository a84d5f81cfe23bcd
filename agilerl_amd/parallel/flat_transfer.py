"""Flat-tensor cross-rank agent-state transfer over RCCL/xGMI.

The reference ships winner weights between ranks as pickled checkpoint
dicts through ``broadcast_object_list`` (``agilerl/hpo/tournament.py:179``)
— a host round-trip plus pickle of every parameter.  Here the transfer is
split:

1. a tiny pickled *skeleton* — the checkpoint dict with every tensor leaf
   replaced by a placeholder (init dicts, hyperparameters, tensor layout);
2. ONE flat tensor per dtype carrying every parameter / buffer /
   optimizer moment, broadcast with ``dist.broadcast`` so RCCL moves the
   payload GPU-to-GPU over the xGMI mesh without serialization.

Dtypes are preserved exactly (per-dtype flat buffers), so int64 step
counters and bool buffers round-trip bit-exactly.
"""

from __future__ import annotations

from typing import Any, Dict, List, Tuple

import torch
import torch.distributed as dist

__all__ = ["split_tensor_tree", "merge_tensor_tree", "broadcast_checkpoint"]

_PLACEHOLDER = "__flatxfer_tensor__"


def split_tensor_tree(tree: Any) -> Tuple[Any, List[torch.Tensor]]:
    """Replace every tensor leaf in a nested dict/list/tuple with a
    placeholder index; return (skeleton, tensors-in-visit-order)."""
    tensors: List[torch.Tensor] = []

    def walk(node: Any) -> Any:
        if torch.is_tensor(node):
            tensors.append(node)
            return (_PLACEHOLDER, len(tensors) - 1)
        if isinstance(node, dict):
            return {k: walk(v) for k, v in node.items()}
        if isinstance(node, (list, tuple)):
            out = [walk(v) for v in node]
            return out if isinstance(node, list) else tuple(out)
        return node

    return walk(tree), tensors


def merge_tensor_tree(skeleton: Any, tensors: List[torch.Tensor]) -> Any:
    """Inverse of :func:`split_tensor_tree`."""

    def walk(node: Any) -> Any:
        if isinstance(node, tuple) and len(node) == 2 and node[0] == _PLACEHOLDER:
            return tensors[node[1]]
        if isinstance(node, dict):
            return {k: walk(v) for k, v in node.items()}
        if isinstance(node, list):
            return [walk(v) for v in node]
        if isinstance(node, tuple):
            return tuple(walk(v) for v in node)
        return node

    return walk(skeleton)


def broadcast_checkpoint(
    ckpt: Dict[str, Any] | None,
    src: int,
    rank: int,
    device: str = "cpu",
    backend: str = "gloo",
    group=None,
) -> Dict[str, Any]:
    """Broadcast an agent checkpoint dict from ``src`` to all ranks.

    On ``src``, ``ckpt`` is the dict from ``get_checkpoint_dict()``; other
    ranks pass None.  The bulk payload travels as one flat tensor per dtype
    (on the GPU for the nccl/RCCL backend, so it rides xGMI directly).
    """
    src_tensors: List[torch.Tensor] = []
    if rank == src:
        skeleton, src_tensors = split_tensor_tree(ckpt)
        layout = [(tuple(t.shape), str(t.dtype).replace("torch.", "")) for t in src_tensors]
        meta = [skeleton, layout]
    else:
        meta = [None, None]
    dist.broadcast_object_list(meta, src=src, group=group)
    skeleton, layout = meta

    comm_device = device if backend == "nccl" else "cpu"
    # group tensor indices by dtype -> one flat broadcast per dtype
    by_dtype: Dict[str, List[int]] = {}
    for i, (_, dt) in enumerate(layout):
        by_dtype.setdefault(dt, []).append(i)

    out_tensors: List[torch.Tensor | None] = [None] * len(layout)
    for dt, indices in sorted(by_dtype.items()):
        dtype = getattr(torch, dt)
        numels = [int(torch.Size(layout[i][0]).numel()) for i in indices]
        total = sum(numels)
        if rank == src:
            if total:
                flat = torch.cat(
                    [src_tensors[i].detach().reshape(-1).to(comm_device, dtype) for i in indices]
                )
            else:
                flat = torch.empty(0, dtype=dtype, device=comm_device)
        else:
            flat = torch.empty(total, dtype=dtype, device=comm_device)
        if total:
            dist.broadcast(flat, src=src, group=group)
        flat = flat.cpu()
        off = 0
        for i, n in zip(indices, numels):
            out_tensors[i] = flat[off : off + n].reshape(layout[i][0]).clone()
            off += n
    return merge_tensor_tree(skeleton, out_tensors)
