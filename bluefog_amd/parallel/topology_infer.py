# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Dynamic-topology inference: given each rank's send (or recv) set, derive
the mirror set by inverting the allgathered adjacency (reference analog:
bluefog/torch/topology_util.py:22-108). Collective calls."""

import collections
from typing import Any, List, Tuple, Union

import numpy as np
import torch

__all__ = ["InferSourceFromDestinationRanks", "InferDestinationFromSourceRanks"]


def _check_ranks(rank_list: List[Any], self_rank: int, size: int) -> Tuple[bool, str]:
    for rank in rank_list:
        if not isinstance(rank, int):
            return False, "contain element that is not integer."
        if (rank < 0) or (rank >= size):
            return False, "contain element that is not between 0 and size-1."
    if len(set(rank_list)) != len(rank_list):
        return False, "contain duplicated elements."
    if self_rank in rank_list:
        return False, "contain self rank."
    return True, ""


def InferSourceFromDestinationRanks(
    dst_ranks: List[int], construct_adjacency_matrix: bool = False
) -> Union[List[int], Tuple[List[int], np.ndarray]]:
    """Source ranks whose dst set names this rank. With
    ``construct_adjacency_matrix`` also return the column-normalized W."""
    from bluefog_amd.ops.context import ctx

    is_valid, error_msg = _check_ranks(dst_ranks, ctx().rank(), ctx().size())
    assert is_valid, f"The format of dst_ranks is wrong: {error_msg}"
    return _infer_topo(dst_ranks, transpose=False,
                       construct_adjacency_matrix=construct_adjacency_matrix)


def InferDestinationFromSourceRanks(
    src_ranks: List[int], construct_adjacency_matrix: bool = False
) -> Union[List[int], Tuple[List[int], np.ndarray]]:
    """Destination ranks whose src set names this rank."""
    from bluefog_amd.ops.context import ctx

    is_valid, error_msg = _check_ranks(src_ranks, ctx().rank(), ctx().size())
    assert is_valid, f"The format of src_ranks is wrong: {error_msg}"
    return _infer_topo(src_ranks, transpose=True,
                       construct_adjacency_matrix=construct_adjacency_matrix)


def _infer_topo(rank_list: List[int], transpose: bool, construct_adjacency_matrix: bool):
    from bluefog_amd.ops import collective
    from bluefog_amd.ops.context import ctx

    degree = len(rank_list)
    all_degree_list = collective.allgather(
        torch.tensor([degree], dtype=torch.int32)
    ).numpy()
    all_rank_list = collective.allgather(
        torch.tensor(rank_list, dtype=torch.int32)
    ).numpy()
    adjacency_dict = {}
    displacement = 0
    for i, deg in enumerate(all_degree_list):
        adjacency_dict[i] = sorted(all_rank_list[displacement : displacement + deg])
        displacement += deg

    inv_adjacency_dict = collections.defaultdict(list)
    for k, adj in adjacency_dict.items():
        for v in adj:
            inv_adjacency_dict[v].append(k)
    return_list = inv_adjacency_dict.get(ctx().rank())
    if return_list is None:
        return_list = []

    if not construct_adjacency_matrix:
        return return_list

    W = np.eye(ctx().size())
    for k, adj in adjacency_dict.items():
        W[k, adj] = 1
    if transpose:
        W = W.T
    return return_list, W / W.sum(axis=1)
