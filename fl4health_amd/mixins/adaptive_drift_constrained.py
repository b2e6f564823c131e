"""Mixin system for composing personalization behaviors onto clients.

Capability of reference fl4health/mixins/ (base.py:11, core_protocols.py,
adaptive_drift_constrained.py:35-204, personalized/ditto.py:47,
personalized/mr_mtl.py:35): protocol-checked mixins plus dynamic class
factories that graft a behavior onto an existing client class.
"""
from __future__ import annotations

from typing import Protocol, runtime_checkable

import torch

from fl4health_amd.clients.adaptive_drift_constraint_client import AdaptiveDriftConstraintClient
from fl4health_amd.clients.basic_client import BasicClient


@runtime_checkable
class BasicClientProtocol(Protocol):
    """Minimal surface a client must expose for mixins (reference core_protocols.py:15)."""

    def get_model(self, config): ...

    def get_data_loaders(self, config): ...

    def get_optimizer(self, config): ...

    def get_criterion(self, config): ...


class BaseFlexibleMixin:
    """Validates at subclass time that the mixin lands on a client class
    (reference base.py:11)."""

    def __init_subclass__(cls, **kwargs) -> None:
        super().__init_subclass__(**kwargs)
        # pure mixin classes (not yet composed with a client) are exempt
        if cls.__name__.endswith("Mixin") or cls.__name__.startswith("_"):
            return
        if not any(issubclass(b, BasicClient) for b in cls.__mro__ if b not in (cls, BaseFlexibleMixin, object)):
            import warnings

            warnings.warn(f"{cls.__name__} mixes into a non-client class; protocol conformance not guaranteed")


class AdaptiveDriftConstrainedMixin(BaseFlexibleMixin):
    """Adds the packed-loss / adaptive-mu / drift-penalty behavior. Concrete
    math lives in AdaptiveDriftConstraintClient; the mixin form lets users
    compose it onto their own client classes."""

    def apply_adaptive_drift(self) -> None:
        assert isinstance(self, BasicClient)


def apply_adaptive_drift_to_client(client_cls: type[BasicClient]) -> type[BasicClient]:
    """Dynamic class factory (reference adaptive_drift_constrained.py:204):
    returns a subclass of `client_cls` with adaptive drift-constraint behavior."""
    name = f"AdaptiveDrift{client_cls.__name__}"
    return type(name, (AdaptiveDriftConstraintClient, client_cls), {})


def make_it_personal(client_cls: type[BasicClient], mode: str = "ditto") -> type[BasicClient]:
    """Personalization factory (reference personalized/__init__.py:19-41);
    the hook-driven implementation lives in mixins/personalized.py."""
    from fl4health_amd.mixins.personalized import make_it_personal as _factory

    return _factory(client_cls, mode)
