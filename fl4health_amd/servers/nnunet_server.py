"""nnU-Net segmentation server
(reference fl4health/servers/nnunet_server.py:54-264): pre-fit bootstrap asks
ONE client to generate training plans from its local dataset fingerprint; the
pickled plans blob is injected into every subsequent config (fit, evaluate
AND the init-parameters handshake); client-reported channel/head counts drive
late server-side model initialization for checkpointing."""
from __future__ import annotations

import logging
import pickle
import random

from fl4health_amd.common import GetPropertiesIns
from fl4health_amd.servers.base_server import FlServer

log = logging.getLogger(__name__)


class NnunetServer(FlServer):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.nnunet_plans_bytes: bytes | None = None
        self.num_input_channels: int | None = None
        self.num_segmentation_heads: int | None = None
        self.enable_deep_supervision: bool | None = None

    def update_before_fit(self, num_rounds: int, timeout: float | None) -> None:
        """Plans bootstrap (reference update_before_fit :156-250): poll one
        client WITHOUT plans in the config — it elects plans from its local
        dataset and returns them (with model-shape properties) — then inject
        the blob into every config produced from here on."""
        if self.nnunet_plans_bytes is None:
            assert self.transport is not None
            elected = random.choice(list(self.client_manager.all().values()))
            ins = GetPropertiesIns(config={**self.fl_config})
            results = self.transport.poll_clients([(elected, ins)], timeout)
            props = results[0][1].properties
            plans = props["nnunet_plans"]
            self.nnunet_plans_bytes = plans if isinstance(plans, bytes) else pickle.dumps(plans)
            self.num_input_channels = int(props.get("num_input_channels", 1))
            self.num_segmentation_heads = int(props.get("num_segmentation_heads", 2))
            self.enable_deep_supervision = bool(props.get("enable_deep_supervision", True))
            log.info(
                "Elected nnunet plans: %s (%d channels, %d heads)",
                pickle.loads(self.nnunet_plans_bytes).get("plans_name"),
                self.num_input_channels,
                self.num_segmentation_heads,
            )
        plans = self.nnunet_plans_bytes
        self.fl_config = {**self.fl_config, "nnunet_plans": plans}
        base_fn = self.on_init_parameters_config_fn
        self.on_init_parameters_config_fn = lambda r: {**base_fn(r), "nnunet_plans": plans}
        if self.strategy.on_fit_config_fn is not None:
            fit_fn = self.strategy.on_fit_config_fn
            self.strategy.on_fit_config_fn = lambda r: {**fit_fn(r), "nnunet_plans": plans}
        else:
            self.strategy.on_fit_config_fn = lambda r: {"current_server_round": r, "nnunet_plans": plans}
        if getattr(self.strategy, "on_evaluate_config_fn", None) is not None:
            ev_fn = self.strategy.on_evaluate_config_fn
            self.strategy.on_evaluate_config_fn = lambda r: {**ev_fn(r), "nnunet_plans": plans}
        else:
            self.strategy.on_evaluate_config_fn = lambda r: {"current_server_round": r, "nnunet_plans": plans}
        self.initialize_server_model()

    def initialize_server_model(self) -> None:
        """Late server model init from the elected plans (reference :133-154):
        the architecture is only knowable after the bootstrap poll."""
        if self.checkpoint_and_state_module.model is not None or self.nnunet_plans_bytes is None:
            return
        from fl4health_amd.models.unet3d import UNet3D

        plans = pickle.loads(self.nnunet_plans_bytes)
        net = plans.get("network")
        if net is None:
            cfgs = plans["configurations"]
            key = "3d_fullres" if "3d_fullres" in cfgs else next(iter(cfgs))
            net = {
                "in_channels": self.num_input_channels or 1,
                "num_classes": self.num_segmentation_heads or 2,
                "base_channels": int(cfgs[key].get("UNet_base_num_features", 32)),
                "num_levels": int(cfgs[key].get("n_stages", 4)),
            }
        self.checkpoint_and_state_module.model = UNet3D(
            in_channels=net["in_channels"],
            num_classes=net["num_classes"],
            base_channels=min(net["base_channels"], 32),
            num_levels=net["num_levels"],
        )
