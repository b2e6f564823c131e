"""Flat parameter substrate — the MI355X-native wire format.

The reference exchanges a model as an ordered list of per-layer NumPy arrays
(fl4health/parameter_exchange/full_exchanger.py:10-48); every aggregation then
loops layer-by-layer in Python. Here the exchanged set lives in ONE contiguous
fp32 device tensor:

- one RCCL all-reduce / broadcast moves the whole model (bucketing handled by
  the communicator, sized for 7 xGMI links),
- one fused HIP kernel implements each server/client hot op over the buffer,
- module params/buffers are rebound as VIEWS of the flat buffer where dtypes
  permit, so push/pull are zero-copy on the hot path.

Non-fp32 state_dict entries (e.g. BatchNorm ``num_batches_tracked`` int64) are
carried in the flat buffer as casted fp32 slots and cast back on pull — same
aggregation semantics as the reference (flwr averages them as float arrays).
"""
from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.nn as nn


@dataclass
class ParameterSpec:
    """Layout of a flat buffer: state_dict-ordered (name, shape, dtype, offset)."""

    names: list[str]
    shapes: list[torch.Size]
    dtypes: list[torch.dtype]
    offsets: list[int] = field(default_factory=list)
    total: int = 0

    def __post_init__(self) -> None:
        if not self.offsets:
            off = 0
            for s in self.shapes:
                self.offsets.append(off)
                off += int(torch.Size(s).numel())
            self.total = off

    @classmethod
    def from_named_tensors(cls, named: list[tuple[str, torch.Tensor]]) -> "ParameterSpec":
        return cls(
            names=[n for n, _ in named],
            shapes=[t.shape for _, t in named],
            dtypes=[t.dtype for _, t in named],
        )

    def slice_of(self, flat: torch.Tensor, i: int) -> torch.Tensor:
        n = int(torch.Size(self.shapes[i]).numel())
        return flat[self.offsets[i] : self.offsets[i] + n].view(self.shapes[i])

    def index_of(self, name: str) -> int:
        return self.names.index(name)


class FlatParameterView:
    """A flat fp32 buffer over a chosen subset of a module's state_dict.

    ``bind=True`` additionally rebinds the module's fp32 params/buffers to be
    views of the flat buffer (zero-copy push/pull + fused flat optimizer ops).
    """

    def __init__(self, module: nn.Module, names: list[str] | None = None, device: torch.device | str | None = None, bind: bool = False) -> None:
        sd = module.state_dict()
        if names is None:
            # params-first ordering: all trainable parameters form one
            # contiguous leading region of the flat buffer, so the fused
            # optimizer/penalty kernels (prox-SGD, SCAFFOLD correction, DP
            # clip) run over flat[:params_numel] in a single pass while
            # buffers (BN running stats...) are exchanged but never stepped.
            # Within the params region, multi-dim (conv/linear) params come
            # FIRST so an optional bf16 compute mirror covers one contiguous
            # leading slice (1D affine params stay fp32 for the BN kernels).
            all_params = [(n, p) for n, p in module.named_parameters()]
            nd_names = [n for n, p in all_params if p.dim() >= 2]
            oned_names = [n for n, p in all_params if p.dim() < 2]
            param_names = nd_names + oned_names
            param_set = set(param_names)
            names = param_names + [n for n in sd.keys() if n not in param_set]
            self.params_numel = sum(sd[n].numel() for n in param_names)
            self.mirror_numel = sum(sd[n].numel() for n in nd_names)
        else:
            self.params_numel = None  # unknown for custom subsets
            self.mirror_numel = None
        self.bf16_mirror: torch.Tensor | None = None
        self.bf16_grad: torch.Tensor | None = None
        named = [(n, sd[n]) for n in names]
        self.spec = ParameterSpec.from_named_tensors(named)
        dev = device if device is not None else (named[0][1].device if named else "cpu")
        self.flat = torch.zeros(self.spec.total, dtype=torch.float32, device=dev)
        self.module = module
        self.bound = False
        self.pull_into_flat()
        if bind:
            self._bind_views()

    # ---- data movement -------------------------------------------------
    def pull_into_flat(self) -> None:
        """Copy current module tensors into the flat buffer (no-op when bound)."""
        sd = self.module.state_dict()
        for i, name in enumerate(self.spec.names):
            t = sd[name]
            if self.bf16_mirror is not None and t.dtype == torch.bfloat16:
                continue  # fp32 master is authoritative for mirrored params
            dst = self.spec.slice_of(self.flat, i)
            if self.bound and dst.data_ptr() == t.data_ptr():
                continue
            dst.copy_(t.detach().to(torch.float32))

    def push_into_module(self) -> None:
        """Copy flat buffer values back into the module (no-op for bound views)."""
        sd = self.module.state_dict()
        with torch.no_grad():
            for i, name in enumerate(self.spec.names):
                t = sd[name]
                if self.bf16_mirror is not None and t.dtype == torch.bfloat16:
                    continue  # refreshed in one pass by sync_mirror_
                src = self.spec.slice_of(self.flat, i)
                if self.bound and src.data_ptr() == t.data_ptr():
                    continue
                t.copy_(src.to(t.dtype))

    def clone_flat(self) -> torch.Tensor:
        return self.flat.detach().clone()

    def load_flat(self, flat: torch.Tensor) -> None:
        self.flat.copy_(flat.to(self.flat.device, torch.float32))
        self.push_into_module()
        self.sync_mirror_()

    # ---- view binding (zero-copy hot path) -----------------------------
    def _bind_views(self) -> None:
        """Rebind fp32 module params/buffers as views of the flat buffer.

        4D params of channels-last modules are bound with NHWC strides over
        their flat slice (view as (N,H,W,C) then permute), so MIOpen's NHWC
        bf16 igemm kernels run without per-step layout transposes. The flat
        buffer then stores such weights in NHWC element order — identical on
        every rank, so collective aggregation is unaffected.
        """
        name_to_idx = {n: i for i, n in enumerate(self.spec.names)}
        for mod_name, mod in self.module.named_modules():
            prefix = mod_name + "." if mod_name else ""
            for pname, p in list(mod.named_parameters(recurse=False)):
                full = prefix + pname
                i = name_to_idx.get(full)
                if i is None or p.dtype != torch.float32:
                    continue
                view = self.spec.slice_of(self.flat, i).reshape(-1)
                if p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last):
                    n_, c_, h_, w_ = p.shape
                    nhwc = view.view(n_, h_, w_, c_).permute(0, 3, 1, 2)
                    with torch.no_grad():
                        nhwc.copy_(p.detach())
                    new_p = nn.Parameter(nhwc, requires_grad=p.requires_grad)
                else:
                    new_p = nn.Parameter(view.view(p.shape), requires_grad=p.requires_grad)
                setattr(mod, pname, new_p)
            for bname, b in list(mod.named_buffers(recurse=False)):
                full = prefix + bname
                i = name_to_idx.get(full)
                if i is None or b is None or b.dtype != torch.float32:
                    continue
                mod._buffers[bname] = self.spec.slice_of(self.flat, i)
        self.bound = True

    # ---- flat gradient buffer ------------------------------------------
    @property
    def params_region(self) -> torch.Tensor:
        """Contiguous trainable-parameter slice of the flat buffer."""
        assert self.params_numel is not None, "params-first ordering required"
        return self.flat[: self.params_numel]

    # ---- persistent bf16 compute mirror --------------------------------
    def _strided_view(self, base: torch.Tensor, i: int, like: torch.Tensor) -> torch.Tensor:
        n = int(torch.Size(self.spec.shapes[i]).numel())
        sl = base[self.spec.offsets[i] : self.spec.offsets[i] + n]
        p = like
        if p.dim() == 4 and not p.is_contiguous() and p.is_contiguous(memory_format=torch.channels_last):
            n_, c_, h_, w_ = p.shape
            return sl.view(n_, h_, w_, c_).permute(0, 3, 1, 2)
        return sl.view(p.shape)

    def enable_bf16_mirror(self) -> None:
        """Rebind multi-dim params as bf16 views of a persistent mirror buffer.

        The fp32 master stays the source of truth (exchange, optimizer,
        penalties); forwards read the bf16 mirror directly so autocast's
        per-step weight-cast kernels disappear. The fused prox-SGD kernel
        consumes the resulting bf16 grads and re-casts updated weights into
        the mirror in the same pass. Requires bound views + autocast training.
        """
        assert self.bound and self.mirror_numel, "bind=True and multi-dim params required"
        self.bf16_mirror = torch.empty(self.mirror_numel, dtype=torch.bfloat16, device=self.flat.device)
        self.bf16_grad = torch.zeros(self.mirror_numel, dtype=torch.bfloat16, device=self.flat.device)
        with torch.no_grad():
            self.bf16_mirror.copy_(self.flat[: self.mirror_numel])
        name_to_idx = {n: i for i, n in enumerate(self.spec.names)}
        for mod_name, mod in self.module.named_modules():
            prefix = mod_name + "." if mod_name else ""
            for pname, p in list(mod.named_parameters(recurse=False)):
                i = name_to_idx.get(prefix + pname)
                if i is None or p.dim() < 2 or p.dtype != torch.float32:
                    continue
                view = self._strided_view(self.bf16_mirror, i, p)
                new_p = nn.Parameter(view, requires_grad=p.requires_grad)
                new_p.grad = self._strided_view(self.bf16_grad, i, p)
                setattr(mod, pname, new_p)

    def sync_mirror_(self) -> None:
        if self.bf16_mirror is not None:
            with torch.no_grad():
                self.bf16_mirror.copy_(self.flat[: self.mirror_numel])

    def make_grad_buffer(self) -> torch.Tensor:
        """Allocate a flat grad buffer over the params region and point each
        bound fp32 param's .grad at its slice.

        Autograd accumulates into existing .grad in place, so after backward the
        flat grad buffer holds all gradients contiguously (fused optimizer ops,
        single-collective gradient reduction).
        """
        assert self.params_numel is not None, "params-first ordering required"
        gbuf = torch.zeros(self.params_numel, dtype=torch.float32, device=self.flat.device)
        name_to_idx = {n: i for i, n in enumerate(self.spec.names)}
        for pname, p in self.module.named_parameters():
            i = name_to_idx.get(pname)
            if i is None or p.dtype != torch.float32:
                continue
            n = p.numel()
            gslice = gbuf[self.spec.offsets[i] : self.spec.offsets[i] + n]
            if p.dim() == 4 and not p.is_contiguous() and p.is_contiguous(memory_format=torch.channels_last):
                # grad memory order must MATCH the param's NHWC flat order so
                # the fused elementwise optimizer kernels stay aligned
                n_, c_, h_, w_ = p.shape
                p.grad = gslice.view(n_, h_, w_, c_).permute(0, 3, 1, 2)
            else:
                p.grad = gslice.view(p.shape)
        return gbuf
