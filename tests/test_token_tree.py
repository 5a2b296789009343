"""Token-tree structure: parents/paths/masks/positions/acceptance."""

import torch

from neuronx_distributed_inference_amd.modules.token_tree import TokenTree


def test_tree_structure():
    t = TokenTree([2, 2])
    assert t.num_nodes == 1 + 2 + 4
    assert t.parent == [-1, 0, 0, 1, 1, 2, 2]
    assert t.depth == [0, 1, 1, 2, 2, 2, 2]
    assert len(t.paths) == 4
    assert t.paths[0] == [0, 1, 3]


def test_tree_mask():
    t = TokenTree([2])
    m = t.attention_mask(prior_len=3)[0, 0]
    assert m.shape == (3, 6)
    assert m[:, :3].all()                      # full prior
    assert m[1, 3] and m[1, 4] and not m[1, 5]  # node1: root + itself
    assert m[2, 3] and m[2, 5] and not m[2, 4]


def test_tree_positions():
    t = TokenTree([2, 1])
    assert t.position_ids(10).tolist() == [10, 11, 11, 12, 12]


def test_tree_accept():
    t = TokenTree([2, 1])        # nodes: 0 root; 1,2 level1; 3(of 1),4(of 2)
    target = torch.tensor([7, 9, 0, 5, 0])   # predictions AT each node
    drafted = torch.tensor([0, 7, 3, 9, 9])  # node tokens (root unused)
    # path 0-1-3: node1 token 7 == target[0] ok; node3 token 9 == target[1] ok
    acc, bonus = t.accept(drafted, target)
    assert acc == [1, 3]
    assert bonus == 3
    # break the second hop
    drafted2 = torch.tensor([0, 7, 3, 8, 9])
    acc2, bonus2 = t.accept(drafted2, target)
    assert acc2 == [1]
    assert bonus2 == 1
