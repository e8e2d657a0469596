"""Synthetic graph shortest-path task (parity: reference
examples/randomwalks/randomwalks.py).

A random directed graph over ``n_nodes`` (letters a, b, c, ...; goal node is
'a'); samples are random walks; the metric scores a generated walk by how
close it is to the shortest path from its start node (optimality in [0, 1]).
No network, no files — the canonical offline regression workload.

BFS replaces networkx (not installed); semantics are unchanged.
"""

from collections import deque
from typing import Callable, Dict, List, Optional, Tuple

import numpy as np
import torch


def _rand_int_excluding(rng: np.random.RandomState, maxval: int, exclude: int) -> int:
    while True:
        x = rng.randint(maxval)
        if x != exclude:
            return x


def _bfs_shortest_lengths(adjacency: np.ndarray, goal: int, max_length: int) -> List[int]:
    """Shortest path length in NODES (incl. start and goal) from each non-goal
    node to the goal; ``max_length`` when unreachable."""
    n = adjacency.shape[0]
    # reverse BFS from goal
    dist = [-1] * n
    dist[goal] = 0
    q = deque([goal])
    while q:
        v = q.popleft()
        for u in range(n):
            if adjacency[u, v] and dist[u] == -1:
                dist[u] = dist[v] + 1
                q.append(u)
    out = []
    for start in range(n):
        if start == goal:
            continue
        if dist[start] == -1:
            out.append(max_length)
        else:
            out.append(min(dist[start] + 1, max_length))
    return out


def generate_random_walks(
    n_nodes: int = 21,
    max_length: int = 10,
    n_walks: int = 1000,
    p_edge: float = 0.1,
    seed: int = 1002,
) -> Tuple[Callable, List[str], List[str], torch.Tensor]:
    """Returns (metric_fn, eval_prompts, sample_walks, logit_mask)."""
    rng = np.random.RandomState(seed)

    while True:
        adjacency = rng.rand(n_nodes, n_nodes) > (1 - p_edge)
        np.fill_diagonal(adjacency, 0)
        # every source node must have at least one outgoing edge
        if np.all(adjacency.sum(1)):
            break

    goal = 0
    adjacency[goal, :] = 0
    adjacency[goal, goal] = 1

    char_to_node = {chr(ix + ord("a")): ix for ix in range(n_nodes)}
    node_to_char = {ix: chr(ix + ord("a")) for ix in range(n_nodes)}

    sample_walks: List[str] = []
    for _ in range(n_walks):
        node = _rand_int_excluding(rng, n_nodes, goal)
        walk_nodes = [node]
        for _step in range(max_length - 1):
            node = rng.choice(np.nonzero(adjacency[node])[0])
            walk_nodes.append(node)
            if node == goal:
                break
        sample_walks.append("".join(node_to_char[ix] for ix in walk_nodes))

    shortest_lengths = _bfs_shortest_lengths(adjacency, goal, max_length)

    def metric_fn(samples: List[str], **kwargs) -> Dict[str, List[float]]:
        invalid_path_length = 100
        lengths: List[float] = []
        sample_optimal_lengths: List[int] = []
        for sample_str in samples:
            sample = [char_to_node.get(c, 1000) for c in sample_str]
            length: Optional[float] = None
            for node in range(len(sample)):
                if sample[node] >= n_nodes or (node > 0 and not adjacency[sample[node - 1], sample[node]]):
                    length = invalid_path_length
                    break
                elif sample[node] == 0:
                    length = node + 1
                    break
            if length is None:
                length = invalid_path_length
            lengths.append(float(length))
            start = sample[0] if sample and sample[0] < n_nodes else 1
            sample_optimal_lengths.append(shortest_lengths[start - 1])

        lengths_t = torch.tensor(lengths, dtype=torch.float)
        bound = torch.where(lengths_t.eq(invalid_path_length), torch.tensor(float(max_length)), lengths_t).abs()
        optimal = torch.as_tensor(sample_optimal_lengths)
        optimality = (max_length - bound) / (max_length - optimal)
        return {"lengths": lengths, "optimality": optimality.tolist()}

    logit_mask = torch.tensor(adjacency)

    eval_prompts = sorted(set(w[0] for w in sample_walks))

    return metric_fn, eval_prompts, sample_walks, logit_mask
