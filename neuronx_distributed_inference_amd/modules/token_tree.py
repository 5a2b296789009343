"""Static token trees for tree-structured speculation.

Analog of the reference's `modules/eagle/token_tree.py` (:8-560: per-level
top-k config, path enumeration, tree attention masks, rotary position ids,
scatter indices). A tree is specified by per-level branching factors, e.g.
[4, 2] = root proposes 4 children, each proposes 2 -> 1 + 4 + 8 = 13 nodes.

The draft proposes tokens for every node level by level; the target
verifies ALL nodes in one forward using the tree attention mask (each node
attends its ancestors + the full prior); acceptance walks the deepest path
whose tokens all match the target's predictions.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Tuple

import torch


@dataclass
class TokenTree:
    branching: List[int]                      # per-level top-k
    parent: List[int] = field(init=False)     # node -> parent (-1 for root)
    depth: List[int] = field(init=False)      # node -> depth (root = 0)
    level_nodes: List[List[int]] = field(init=False)
    paths: List[List[int]] = field(init=False)  # root-to-leaf node paths

    def __post_init__(self):
        self.parent = [-1]
        self.depth = [0]
        self.level_nodes = [[0]]
        for level, k in enumerate(self.branching):
            new_level = []
            for p in self.level_nodes[level]:
                for _ in range(k):
                    idx = len(self.parent)
                    self.parent.append(p)
                    self.depth.append(level + 1)
                    new_level.append(idx)
            self.level_nodes.append(new_level)
        # enumerate root->leaf paths
        children = [[] for _ in range(self.num_nodes)]
        for n, p in enumerate(self.parent):
            if p >= 0:
                children[p].append(n)
        self.paths = []

        def walk(n, acc):
            acc = acc + [n]
            if not children[n]:
                self.paths.append(acc)
            for c in children[n]:
                walk(c, acc)

        walk(0, [])

    @property
    def num_nodes(self) -> int:
        return len(self.parent)

    def ancestors(self, n: int) -> List[int]:
        out = []
        while n >= 0:
            out.append(n)
            n = self.parent[n]
        return list(reversed(out))

    def attention_mask(self, prior_len: int, device=None) -> torch.Tensor:
        """(1, 1, N, prior_len + N) bool: node n attends the full prior and
        its ancestor chain (incl. itself) — the tree analog of the causal
        mask (ref token_tree.py tree masks)."""
        N = self.num_nodes
        m = torch.zeros(N, prior_len + N, dtype=torch.bool, device=device)
        m[:, :prior_len] = True
        for n in range(N):
            for a in self.ancestors(n):
                m[n, prior_len + a] = True
        return m.view(1, 1, N, prior_len + N)

    def position_ids(self, base: int, device=None) -> torch.Tensor:
        """(N,) rotary positions: base + depth(n) (ref rotary position ids
        for trees)."""
        return torch.tensor([base + d for d in self.depth], device=device)

    def accept(self, node_tokens: torch.Tensor, target_tokens: torch.Tensor
               ) -> Tuple[List[int], int]:
        """Greedy tree acceptance for ONE sequence.

        node_tokens: (N,) the drafted token at each non-root node (root holds
        the committed token); target_tokens: (N,) target's greedy prediction
        AT each node. Returns (accepted_node_path_excluding_root, bonus_node)
        where bonus_node is the node whose target prediction extends the
        accepted path.
        """
        best: List[int] = []
        best_node = 0
        for path in self.paths:
            acc = []
            node = 0
            for child in path[1:]:
                # child's token must equal target's prediction at its parent
                if int(node_tokens[child]) == int(target_tokens[self.parent[child]]):
                    acc.append(child)
                    node = child
                else:
                    break
            if len(acc) > len(best):
                best = acc
                best_node = node
        return best, best_node


def ancestor_matrix(tree: "TokenTree") -> torch.Tensor:
    """(N, N) bool: anc[n, a] = a is on n's root path (incl. n)."""
    N = tree.num_nodes
    anc = torch.zeros(N, N, dtype=torch.bool)
    for n in range(N):
        for a in tree.ancestors(n):
            anc[n, a] = True
    return anc


def tree_attention_mask_batch(tree: "TokenTree", anc: torch.Tensor,
                              nodes, lengths: torch.Tensor, Smax: int,
                              device) -> torch.Tensor:
    """(B, 1, len(nodes), Smax) bool tree mask with PER-ROW prior
    lengths: node n attends cache cols [0, lengths_b) plus its ancestor
    rows at lengths_b + ancestor_idx (the batched form of
    TokenTree.attention_mask; used by EAGLE tree TKG and Medusa tree
    verification)."""
    B = lengths.shape[0]
    N = tree.num_nodes
    cols = torch.arange(Smax, device=device)
    rel = cols.view(1, -1) - lengths.view(B, 1)
    in_tree = (rel >= 0) & (rel < N)
    relc = rel.clamp(0, N - 1)
    anc_sel = anc.to(device)[list(nodes)]
    tree_m = anc_sel[:, relc].permute(1, 0, 2) & in_tree.unsqueeze(1)
    prior_m = (rel < 0).unsqueeze(1).expand(-1, len(nodes), -1)
    return (prior_m | tree_m).unsqueeze(1)
