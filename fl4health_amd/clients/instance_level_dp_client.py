"""Instance-level DP client (reference fl4health/clients/instance_level_dp_client.py:17-114).

DP-SGD with per-sample gradient clipping + Gaussian noise via the from-scratch
GradSampleModule/DpSgdEngine (Opacus-equivalent on CDNA4 kernels, K7).
"""
from __future__ import annotations

import secrets

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.privacy.dp_sgd import DpSgdEngine
from fl4health_amd.privacy.grad_sample import GradSampleModule, convert_batchnorm_modules


class InstanceLevelDpClient(BasicClient):
    def __init__(self, *args, clipping_bound: float = 1.0, noise_multiplier: float = 1.0, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.clipping_bound = clipping_bound
        self.noise_multiplier = noise_multiplier
        self.dp_engine: DpSgdEngine | None = None

    def setup_client(self, config: Config) -> None:
        self.clipping_bound = float(config.get("clipping_bound", self.clipping_bound))
        self.noise_multiplier = float(config.get("noise_multiplier", self.noise_multiplier))
        super().setup_client(config)
        # make private: BN -> GN, hook per-sample grads, wrap optimizer
        self.model = convert_batchnorm_modules(self.model)
        # rebuild the flat view after module surgery
        from fl4health_amd.parameter_exchange.flat import FlatParameterView
        from fl4health_amd.parameter_exchange.exchangers import FullParameterExchanger

        self.flat_view = FlatParameterView(self.model, bind=True)
        if isinstance(self.parameter_exchanger, FullParameterExchanger):
            self.parameter_exchanger._view = self.flat_view
        self.set_optimizer(config)
        gsm = GradSampleModule(self.model)
        self._gsm = gsm
        self.model = gsm
        # DP noise MUST be unpredictable: a fixed/broadcast seed lets an
        # observer regenerate and subtract the noise stream, voiding the DP
        # guarantee. Default to a cryptographically random per-client per-run
        # seed; a fixed seed is honored only via the explicit debug/test key
        # `dp_noise_seed` (never the shared training `seed`).
        if "dp_noise_seed" in config:
            noise_seed = int(config["dp_noise_seed"])
        else:
            noise_seed = secrets.randbits(63)
        self.dp_engine = DpSgdEngine(
            gsm, self.optimizers["global"], self.noise_multiplier, self.clipping_bound,
            seed=noise_seed,
        )

    def set_optimizer_zero_grad(self) -> None:
        if self.dp_engine is not None:
            self.dp_engine.zero_grad()
        else:
            super().set_optimizer_zero_grad()

    def step_optimizers(self) -> None:
        assert self.dp_engine is not None
        self.dp_engine.step()
