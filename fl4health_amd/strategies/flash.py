"""Flash strategy (reference fl4health/strategies/flash.py:21-170).

Server-side adaptive optimization with a drift-aware third moment:
  m_t = b1*m + (1-b1)*delta
  v_t = b2*v + (1-b2)*delta^2
  beta_3 = |v_prev| / (|delta^2 - v_t| + |v_prev|)   (per element)
  d_t = beta_3*d + (1-beta_3)*(delta^2 - v_t)
  x  += eta * m_t / (sqrt(v_t) - d_t + tau)
One fused kernel pass over the flat buffer (kind=flash in server_opt_kernel).
"""
from __future__ import annotations

from fl4health_amd.strategies.fedopt import _FedOptBase


class Flash(_FedOptBase):
    kind = "flash"

    def __init__(self, *, eta: float = 1e-1, beta_1: float = 0.9, beta_2: float = 0.99, tau: float = 1e-9, **kwargs) -> None:
        super().__init__(eta=eta, beta_1=beta_1, beta_2=beta_2, tau=tau, **kwargs)
