"""Flat parameter buffer — the wire/checkpoint format contract.

The reference's wire format is a list of fp32 ndarrays in sorted-name order,
filtered to names containing ``transformer`` (photon/utils.py:640-670,
SURVEY.md §3.5); its SHM layer describes the same list as one flat segment
with per-array bounds (photon/shm/utils.py:138-247).

MI355X-native equivalent: ONE flat fp32 torch tensor resident in HBM, with
per-parameter views in the same sorted-name order. The flat buffer is what
RCCL broadcast/all-reduce operate on (one large collective instead of
per-layer messages — sized for the 7-link xGMI topology), and `.npz`
import/export keeps checkpoint compatibility.
"""

from __future__ import annotations

from pathlib import Path

import numpy as np
import torch


def trainable_param_names(
    model: torch.nn.Module, filter_key: str | None = "transformer"
) -> list[str]:
    """Sorted trainable parameter names, optionally filtered by substring
    (the reference's set_trainer_key_to_filter contract)."""
    names = [n for n, p in model.named_parameters() if p.requires_grad]
    if filter_key:
        filtered = [n for n in names if filter_key in n]
        if filtered:
            names = filtered
    return sorted(names)


class FlatParams:
    """Flat fp32 buffer + named views over a model's trainable params."""

    def __init__(self, model: torch.nn.Module, filter_key: str | None = "transformer",
                 device=None):
        self.names = trainable_param_names(model, filter_key)
        params = dict(model.named_parameters())
        self.shapes = [tuple(params[n].shape) for n in self.names]
        self.numels = [int(np.prod(s)) if s else 1 for s in self.shapes]
        self.total = sum(self.numels)
        dev = device if device is not None else next(model.parameters()).device
        self.flat = torch.zeros(self.total, dtype=torch.float32, device=dev)
        self._views = {}
        off = 0
        for n, shape, numel in zip(self.names, self.shapes, self.numels):
            self._views[n] = self.flat[off : off + numel].view(shape)
            off += numel

    # -- model <-> buffer ---------------------------------------------------
    @torch.no_grad()
    def copy_from_model(self, model: torch.nn.Module) -> "FlatParams":
        params = dict(model.named_parameters())
        for n in self.names:
            self._views[n].copy_(params[n].detach().to(torch.float32))
        return self

    @torch.no_grad()
    def copy_to_model(self, model: torch.nn.Module) -> None:
        params = dict(model.named_parameters())
        for n in self.names:
            params[n].data.copy_(self._views[n].to(params[n].dtype))

    def view(self, name: str) -> torch.Tensor:
        return self._views[name]

    def views(self) -> list[torch.Tensor]:
        return [self._views[n] for n in self.names]

    def clone_flat(self) -> torch.Tensor:
        return self.flat.clone()

    def like(self) -> torch.Tensor:
        """A zero flat tensor with the same layout."""
        return torch.zeros_like(self.flat)

    def layer_views_of(self, flat: torch.Tensor) -> list[torch.Tensor]:
        """Per-parameter views of an arbitrary flat tensor with this layout."""
        out, off = [], 0
        for shape, numel in zip(self.shapes, self.numels):
            out.append(flat[off : off + numel].view(shape))
            off += numel
        return out

    # -- ndarray/npz wire + checkpoint format -------------------------------
    def to_ndarrays(self) -> list[np.ndarray]:
        return [self._views[n].detach().cpu().numpy().copy() for n in self.names]

    @torch.no_grad()
    def from_ndarrays(self, arrays: list[np.ndarray]) -> "FlatParams":
        assert len(arrays) == len(self.names), (
            f"expected {len(self.names)} arrays, got {len(arrays)}"
        )
        for n, a in zip(self.names, arrays):
            self._views[n].copy_(torch.from_numpy(np.ascontiguousarray(a, dtype=np.float32)))
        return self

    def save_npz(self, path: str | Path, flat: torch.Tensor | None = None) -> None:
        """Reference server-checkpoint layout: positional .npz of fp32 arrays
        (photon/server/s3_utils.py:392-548)."""
        src = self.layer_views_of(flat) if flat is not None else self.views()
        arrays = [v.detach().cpu().numpy() for v in src]
        np.savez(Path(path), *arrays)

    def load_npz(self, path: str | Path) -> list[np.ndarray]:
        with np.load(Path(path)) as z:
            return [z[k] for k in sorted(z.files, key=lambda s: int(s.split("_")[1]))]
