"""PerFCL client (reference fl4health/clients/perfcl_client.py:20-278):
PerFclModel + dual contrastive losses vs previous-round snapshots."""
from __future__ import annotations

from fl4health_amd.clients.constrained_fenda_client import ConstrainedFendaClient
from fl4health_amd.losses.fenda_loss_config import ConstrainedFendaLossContainer
from fl4health_amd.losses.perfcl_loss import PerFclLoss


class PerFclClient(ConstrainedFendaClient):
    def __init__(
        self,
        *args,
        global_feature_loss_temperature: float = 0.5,
        local_feature_loss_temperature: float = 0.5,
        global_feature_contrastive_loss_weight: float = 1.0,
        local_feature_contrastive_loss_weight: float = 1.0,
        **kwargs,
    ) -> None:
        container = ConstrainedFendaLossContainer(
            perfcl_loss=PerFclLoss(None, global_feature_loss_temperature, local_feature_loss_temperature),
            perfcl_global_loss_weight=global_feature_contrastive_loss_weight,
            perfcl_local_loss_weight=local_feature_contrastive_loss_weight,
        )
        super().__init__(*args, loss_container=container, **kwargs)
