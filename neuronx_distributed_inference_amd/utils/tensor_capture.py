"""Tensor capture & replacement for divergence debugging.

Analogs of the reference's TensorCaptureConfig (`config.py:1121-1169`,
collection `model_base.py:1026-1149`) and tensor replacement
(`utils/tensor_replacement/registry.py`, `config.py:1172-1202`):

- `TensorCapture`: forward hooks on chosen modules record their outputs per
  step (eager path; with hipGraphs, capture forces the eager fallback for
  the instrumented run exactly like the reference disables NEFF reuse).
- `TensorReplacement`: forward pre-hooks that substitute golden inputs into
  chosen modules, isolating which submodule diverges.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional

import torch
import torch.nn as nn


class TensorCapture:
    def __init__(self, model: nn.Module, module_names: List[str],
                 max_steps: int = 16):
        self.model = model
        self.module_names = module_names
        self.max_steps = max_steps
        self.captured: Dict[str, List[torch.Tensor]] = {n: [] for n in module_names}
        self._handles = []

    def __enter__(self):
        mods = dict(self.model.named_modules())
        for name in self.module_names:
            if name not in mods:
                raise KeyError(f"module {name!r} not found")

            def hook(mod, args, out, name=name):
                if len(self.captured[name]) < self.max_steps:
                    t = out[0] if isinstance(out, (tuple, list)) else out
                    if isinstance(t, torch.Tensor):
                        self.captured[name].append(t.detach().to("cpu", copy=True))

            self._handles.append(mods[name].register_forward_hook(hook))
        return self

    def __exit__(self, *exc):
        for h in self._handles:
            h.remove()
        self._handles = []
        return False

    def save(self, path: str):
        torch.save(self.captured, path)


class TensorReplacement:
    """Substitute golden INPUTS into chosen modules (forward pre-hook)."""

    def __init__(self, model: nn.Module, replacements: Dict[str, torch.Tensor]):
        self.model = model
        self.replacements = replacements
        self._handles = []

    def __enter__(self):
        mods = dict(self.model.named_modules())
        for name, golden in self.replacements.items():
            if name not in mods:
                raise KeyError(f"module {name!r} not found")

            def pre(mod, args, golden=golden):
                if not args:
                    return args
                new0 = golden.to(args[0].device, args[0].dtype)
                return (new0,) + tuple(args[1:])

            self._handles.append(mods[name].register_forward_pre_hook(pre))
        return self

    def __exit__(self, *exc):
        for h in self._handles:
            h.remove()
        self._handles = []
        return False
