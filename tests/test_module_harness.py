"""Module/function test harness (utils/testing.py) coverage — the
reference's module_test tier (SURVEY §4 tier 2)."""

import torch

from neuronx_distributed_inference_amd.utils.testing import (
    build_function,
    validate_accuracy,
)


def test_build_function_and_validate_accuracy():
    def fn(a, b):
        return torch.nn.functional.silu(a) * b

    ex = (torch.randn(4, 16), torch.randn(4, 16))
    mod, inputs = build_function(fn, ex)
    out = mod(*inputs)
    assert out.shape == (4, 16)

    validate_accuracy(
        lambda a, b: fn(a, b),
        lambda a, b: a.float().mul(torch.sigmoid(a.float())).mul(b.float()),
        example_inputs=ex, atol=1e-5, rtol=1e-5,
    )


def test_validate_accuracy_detects_mismatch():
    import pytest

    with pytest.raises(AssertionError):
        validate_accuracy(
            lambda a: a + 1,
            lambda a: a + 2,
            example_inputs=(torch.randn(3, 3),), atol=1e-6, rtol=1e-6,
        )
