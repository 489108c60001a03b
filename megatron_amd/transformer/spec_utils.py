"""Declarative layer composition: ModuleSpec trees.

Capability analog of reference megatron/core/transformer/spec_utils.py:13-142
(ModuleSpec, build_module, import_module): a layer is described by a spec
tree (module class or dotted import path + init params + submodule specs)
so model variants swap components without subclassing.  Layer-spec providers
live in megatron_amd/transformer/layer_specs.py (the analog of
models/gpt/gpt_layer_specs.py).
"""

from __future__ import annotations

import types
from dataclasses import dataclass, field
from typing import Any, Optional, Tuple, Union


@dataclass
class ModuleSpec:
    """module: a class, or a ("package.module", "ClassName") import path.
    params: extra kwargs for __init__.  submodules: arbitrary spec payload
    passed through as the `submodules=` kwarg when the target accepts it."""

    module: Union[Tuple[str, str], type]
    params: dict = field(default_factory=dict)
    submodules: Optional[object] = None

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return build_module(self, *args, **kwargs)


def import_module(module_path: Tuple[str, str]):
    base_path, name = module_path
    module = __import__(base_path, globals(), locals(), [name])
    return vars(module)[name]


def get_module(spec_or_module: Union[ModuleSpec, type]):
    if isinstance(spec_or_module, (type, types.FunctionType)):
        return spec_or_module
    if isinstance(spec_or_module.module, (type, types.FunctionType)):
        return spec_or_module.module
    return import_module(spec_or_module.module)


def build_module(spec_or_module: Union[ModuleSpec, type], *args, **kwargs):
    """Instantiate a spec (or plain class).  Spec params are merged with call
    kwargs (call kwargs win); `submodules` is forwarded when the target's
    __init__ accepts it and the spec carries one."""
    cls = get_module(spec_or_module)
    if isinstance(spec_or_module, ModuleSpec):
        merged = dict(spec_or_module.params)
        merged.update(kwargs)
        if spec_or_module.submodules is not None:
            import inspect

            if "submodules" in inspect.signature(cls.__init__).parameters:
                merged.setdefault("submodules", spec_or_module.submodules)
        return cls(*args, **merged)
    return cls(*args, **kwargs)
