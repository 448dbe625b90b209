"""Flat fast-weight arena.

The reference carries inner-loop fast weights as a nested name->tensor dict
(``meta_neural_network_architectures.py:11-38``) and updates them tensor by
tensor.  MI355X-first design instead lays every inner-loop-adapted
parameter out in **one contiguous fp32 buffer** per task batch:

* ``theta`` (the meta-learned initialization) is a single ``[P]`` parameter;
* per iteration the fast weights are one ``[T, P]`` tensor (``T`` = tasks
  resident on this GPU), so
* ``autograd.grad`` of the support loss returns ONE ``[T, P]`` tensor,
* the LSLR update is ONE fused elementwise kernel over ``[T, P]``
  (reference: a Python loop over ~10 tensors,
  ``inner_loop_optimizers.py:99-113``),
* the outer meta-gradient all-reduce is ONE flat RCCL bucket.

Layers see the arena through zero-copy views produced by :meth:`views`.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Sequence, Tuple

import torch


@dataclass(frozen=True)
class ParamSpec:
    name: str            # reference-style dotted name, e.g. layer_dict.conv0.conv.weight
    shape: Tuple[int, ...]
    offset: int          # element offset into the flat arena
    numel: int
    slot: int            # LSLR slot id (one learnable per-step LR per spec)


class ParamArena:
    def __init__(self, named_shapes: Sequence[Tuple[str, Tuple[int, ...]]]):
        self.specs: List[ParamSpec] = []
        off = 0
        for slot, (name, shape) in enumerate(named_shapes):
            n = 1
            for s in shape:
                n *= int(s)
            self.specs.append(ParamSpec(name, tuple(int(s) for s in shape), off, n, slot))
            off += n
        self.numel = off
        self._by_name = {s.name: s for s in self.specs}

    @property
    def num_slots(self) -> int:
        return len(self.specs)

    def spec(self, name: str) -> ParamSpec:
        return self._by_name[name]

    def names(self) -> List[str]:
        return [s.name for s in self.specs]

    def slot_index(self, device=None, dtype=torch.long) -> torch.Tensor:
        """[P] tensor mapping each arena element to its LSLR slot — used to
        gather the per-element learning-rate vector for the fused update."""
        idx = torch.empty(self.numel, dtype=dtype)
        for s in self.specs:
            idx[s.offset:s.offset + s.numel] = s.slot
        return idx.to(device) if device is not None else idx

    def pack(self, named: Dict[str, torch.Tensor]) -> torch.Tensor:
        """dict of [shape] tensors -> flat [P] (fp32)."""
        flat = torch.empty(self.numel, dtype=torch.float32)
        for s in self.specs:
            flat[s.offset:s.offset + s.numel] = named[s.name].detach().reshape(-1).float()
        return flat

    def views(self, arena: torch.Tensor) -> Dict[str, torch.Tensor]:
        """Zero-copy per-parameter views.

        arena [P]   -> name -> [*shape]
        arena [T,P] -> name -> [T, *shape]
        """
        out: Dict[str, torch.Tensor] = {}
        if arena.dim() == 1:
            for s in self.specs:
                out[s.name] = arena[s.offset:s.offset + s.numel].view(*s.shape)
        elif arena.dim() == 2:
            T = arena.shape[0]
            for s in self.specs:
                out[s.name] = arena[:, s.offset:s.offset + s.numel].view(T, *s.shape)
        else:
            raise ValueError("arena must be [P] or [T, P], got %s" % (arena.shape,))
        return out

    def unpack(self, arena: torch.Tensor) -> Dict[str, torch.Tensor]:
        return {k: v.clone() for k, v in self.views(arena).items()}
