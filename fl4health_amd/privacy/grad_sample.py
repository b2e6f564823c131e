"""Per-sample gradient engine (Opacus GradSampleModule equivalent, K7).

The reference delegates DP-SGD to Opacus (fl4health/clients/
instance_level_dp_client.py:64-114 + utils/privacy_utilities.py:11-71); this
is a from-scratch MI355X implementation: forward/backward hooks capture
activations and output-grads, per-sample grads come from batched einsums
(rocBLAS/MFMA GEMMs), and the clip+noise+reduce path runs through the fused
HIP kernels per_sample_sqnorm / clip_rowsum / gaussian_noise (flat_ops.hip).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn

SUPPORTED_LAYERS = (nn.Linear, nn.Conv1d, nn.Conv2d, nn.GroupNorm, nn.LayerNorm, nn.Embedding)


def validate_module(module: nn.Module) -> None:
    """BatchNorm mixes samples -> incompatible with per-sample DP (same rule
    as Opacus; reference privacy_utilities.py). Additionally, any trainable
    parameter living on a module type we cannot hook (Conv3d, RNNs,
    MultiheadAttention, custom containers with direct nn.Parameters...) would
    silently keep its ordinary .grad and reach the optimizer UNCLIPPED and
    UN-NOISED — a privacy leak, so it is a hard error (Opacus raises too).
    Freeze such parameters (requires_grad=False) to exclude them explicitly."""
    for name, m in module.named_modules():
        if isinstance(m, (nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d)):
            raise ValueError(
                f"module {name} is BatchNorm: incompatible with instance-level DP; "
                "call convert_batchnorm_modules(model) first"
            )
        if isinstance(m, SUPPORTED_LAYERS):
            continue
        direct_trainable = [pn for pn, p in m.named_parameters(recurse=False) if p.requires_grad]
        if direct_trainable:
            raise ValueError(
                f"module {name or '<root>'} ({type(m).__name__}) holds trainable parameters "
                f"{direct_trainable} but is not a supported per-sample-grad layer "
                f"({', '.join(t.__name__ for t in SUPPORTED_LAYERS)}); its gradients would "
                "bypass DP clipping/noising. Freeze them or replace the module."
            )


def convert_batchnorm_modules(model: nn.Module) -> nn.Module:
    """Replace BatchNorm with GroupNorm (reference privacy_utilities.py:44-71).
    Also disables in-place activations: full backward hooks (the per-sample
    grad capture) forbid in-place mutation of their outputs."""
    for name, child in list(model.named_children()):
        if isinstance(child, (nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d)):
            setattr(model, name, nn.GroupNorm(min(32, child.num_features), child.num_features, affine=True))
        else:
            if getattr(child, "inplace", False):
                child.inplace = False
            convert_batchnorm_modules(child)
    return model


class GradSampleModule(nn.Module):
    """Wraps a module; after backward, each supported layer's parameters carry
    .grad_sample of shape [B, *param_shape]."""

    def __init__(self, module: nn.Module, ghost_clipping: bool = True) -> None:
        super().__init__()
        validate_module(module)
        self._module = module
        self._hooks: list = []
        self._activations: dict[nn.Module, torch.Tensor] = {}
        self.hooks_enabled = True
        # Ghost clipping (Linear, single-token): ||g_b (x) a_b||_F = ||g_b||*||a_b||,
        # so the per-sample norm needs only the captured (act, grad_out) pair and
        # the clipped sum is ONE rocBLAS GEMM (c_b*g_b)^T A — the [B, out, in]
        # per-sample grad tensor is never materialized (for a CNN's big FC layer
        # that is ~B*|W|*4 bytes of HBM traffic per step saved).
        self.ghost_clipping = ghost_clipping
        self._ghost: dict[nn.Module, tuple[torch.Tensor, torch.Tensor]] = {}
        self._register_hooks()

    @property
    def wrapped_module(self) -> nn.Module:
        return self._module

    def forward(self, *args, **kwargs):
        return self._module(*args, **kwargs)

    # ------------------------------------------------------------------
    def _register_hooks(self) -> None:
        for m in self._module.modules():
            if isinstance(m, SUPPORTED_LAYERS) and any(p.requires_grad for p in m.parameters(recurse=False)):
                self._hooks.append(m.register_forward_hook(self._fwd_hook))
                self._hooks.append(m.register_full_backward_hook(self._bwd_hook))

    def remove_hooks(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def _fwd_hook(self, module, inputs, output) -> None:  # noqa: ARG002
        if self.hooks_enabled and module.training:
            self._activations[module] = inputs[0].detach()

    def _bwd_hook(self, module, grad_input, grad_output) -> None:  # noqa: ARG002
        if not self.hooks_enabled:
            return
        act = self._activations.pop(module, None)
        if act is None:
            return
        go = grad_output[0].detach()
        self._compute_grad_sample(module, act, go)

    # ------------------------------------------------------------------
    def _store(self, param: torch.Tensor, gs: torch.Tensor) -> None:
        if getattr(param, "grad_sample", None) is None:
            param.grad_sample = gs
        else:
            param.grad_sample = param.grad_sample + gs

    def _compute_grad_sample(self, m: nn.Module, act: torch.Tensor, go: torch.Tensor) -> None:
        if isinstance(m, nn.Linear):
            if self.ghost_clipping and act.dim() == 2 and m not in self._ghost and m.weight.requires_grad:
                self._ghost[m] = (act, go)
                return
            if m in self._ghost:
                # module fired more than once this step: materialize the stored
                # pair and fall through to the per-sample-grad path for both
                a0, g0 = self._ghost.pop(m)
                self._store(m.weight, torch.einsum("bo,bi->boi", g0, a0))
                if m.bias is not None:
                    self._store(m.bias, g0)
            a2 = act.reshape(act.shape[0], -1, act.shape[-1])  # [B, T, in]
            g2 = go.reshape(go.shape[0], -1, go.shape[-1])  # [B, T, out]
            self._store(m.weight, torch.einsum("bto,bti->boi", g2, a2))
            if m.bias is not None:
                self._store(m.bias, g2.sum(dim=1))
        elif isinstance(m, (nn.Conv1d, nn.Conv2d)):
            if m in self._ghost:
                # module fired more than once this step: the ghost-norm trick
                # cannot see the cross-term between firings, so materialize the
                # stored pair and fall through to the per-sample-grad path for
                # both (mirrors the Linear fallback above)
                a0, g0 = self._ghost.pop(m)
                self._materialize_conv(m, a0, g0)
            elif self.ghost_clipping and isinstance(m, nn.Conv2d) and m.groups == 1:
                # ghost-norm pays when the [L, L] Grams are smaller than the
                # per-sample grad itself: L^2 < |W| (deep ResNet blocks: L=64,
                # |W|=590k). Large-spatial early convs stay materialized.
                l_spatial = go.shape[2:].numel()
                if l_spatial * l_spatial < m.weight.numel():
                    self._ghost[m] = (act, go)
                    return
            self._materialize_conv(m, act, go)
        elif isinstance(m, (nn.GroupNorm, nn.LayerNorm)):
            if isinstance(m, nn.GroupNorm):
                normed = Fn.group_norm(act, m.num_groups, eps=m.eps)
                dims = tuple(range(2, act.dim()))
                gw = (go * normed).sum(dim=dims) if dims else go * normed
                gb = go.sum(dim=dims) if dims else go
            else:
                normed = Fn.layer_norm(act, m.normalized_shape, eps=m.eps)
                n_norm_dims = len(m.normalized_shape)
                lead = tuple(range(1, act.dim() - n_norm_dims))
                gw = (go * normed).sum(dim=lead) if lead else go * normed
                gb = go.sum(dim=lead) if lead else go
            if m.weight is not None:
                self._store(m.weight, gw)
            if m.bias is not None:
                self._store(m.bias, gb)
        elif isinstance(m, nn.Embedding):
            b = act.shape[0]
            gs = torch.zeros(b, *m.weight.shape, device=go.device, dtype=go.dtype)
            idx = act.reshape(b, -1, 1).expand(-1, -1, m.embedding_dim)
            gs.scatter_add_(1, idx.long(), go.reshape(b, -1, m.embedding_dim))
            self._store(m.weight, gs)

    def _materialize_conv(self, m: nn.Module, act: torch.Tensor, go: torch.Tensor) -> None:
        """Explicit per-sample conv grads via unfold + batched einsum."""
        b = act.shape[0]
        if isinstance(m, nn.Conv2d):
            unfolded = Fn.unfold(act, m.kernel_size, m.dilation, m.padding, m.stride)  # [B, Cin*k*k, L]
        else:
            # conv1d as a (1, k) conv2d for unfold
            a4 = act.unsqueeze(2)  # [B, C, 1, L]
            unfolded = Fn.unfold(
                a4, (1, m.kernel_size[0]), (1, m.dilation[0]), (0, m.padding[0]), (1, m.stride[0])
            )  # [B, Cin*k, L]
        g2 = go.reshape(b, go.shape[1], -1)  # [B, Cout, L]
        if m.groups == 1:
            gs = torch.einsum("bol,bil->boi", g2, unfolded)  # [B, Cout, Cin*k*k]
        else:
            cin_per_g = act.shape[1] // m.groups
            cout_per_g = go.shape[1] // m.groups
            uf = unfolded.reshape(b, m.groups, cin_per_g * int(torch.tensor(m.kernel_size).prod()), -1)
            gg = g2.reshape(b, m.groups, cout_per_g, -1)
            gs = torch.einsum("bgol,bgil->bgoi", gg, uf).reshape(b, go.shape[1], -1)
        self._store(m.weight, gs.reshape(b, *m.weight.shape))
        if m.bias is not None:
            self._store(m.bias, g2.sum(dim=2))

    # ------------------------------------------------------------------
    def clear_grad_samples(self) -> None:
        for p in self._module.parameters():
            if hasattr(p, "grad_sample"):
                p.grad_sample = None
        self._ghost.clear()

    def per_sample_params(self) -> list[torch.Tensor]:
        return [p for p in self._module.parameters() if p.requires_grad]
