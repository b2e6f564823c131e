"""Behavioral unit tests for model bases (mirrors reference tests/models/:
apfl_base, feature_extractor_buffer, fedsimclr, ensemble, gpfl, fenda,
autoencoders, pca — the smoke scenarios exercise them end-to-end; these pin
the per-component math and contracts)."""
import pytest
import torch
import torch.nn as nn


def tiny_net():
    return nn.Sequential(nn.Flatten(), nn.Linear(12, 8), nn.ReLU(), nn.Linear(8, 3))


# ---------------------------------------------------------------------------
# APFL
# ---------------------------------------------------------------------------

def test_apfl_layers_to_exchange_are_global_prefixed():
    from fl4health_amd.model_bases.apfl_base import ApflModule

    m = ApflModule(tiny_net())
    names = m.layers_to_exchange()
    assert names and all(n.startswith("global_model.") for n in names)
    assert set(names) == {n for n in m.state_dict() if n.startswith("global_model.")}


def test_apfl_forward_convex_combination():
    from fl4health_amd.model_bases.apfl_base import ApflModule

    m = ApflModule(tiny_net(), alpha=0.25)
    x = torch.randn(4, 12)
    out = m(x)
    assert set(out) == {"personal", "global", "local"}
    expect = 0.25 * out["local"] + 0.75 * out["global"]
    assert torch.allclose(out["personal"], expect, atol=1e-6)


def test_apfl_update_alpha_moves_and_clamps():
    from fl4health_amd.model_bases.apfl_base import ApflModule

    torch.manual_seed(0)
    m = ApflModule(tiny_net(), alpha=0.5, alpha_lr=0.1)
    x, y = torch.randn(8, 12), torch.randint(0, 3, (8,))
    nn.functional.cross_entropy(m(x)["personal"], y).backward()
    before = m.alpha
    m.update_alpha()
    assert m.alpha != before
    assert 0.0 <= m.alpha <= 1.0
    # huge lr must clamp into [0, 1]
    m2 = ApflModule(tiny_net(), alpha=0.5, alpha_lr=1e6)
    nn.functional.cross_entropy(m2(x)["personal"], y).backward()
    m2.update_alpha()
    assert m2.alpha in (0.0, 1.0)


# ---------------------------------------------------------------------------
# FeatureExtractorBuffer
# ---------------------------------------------------------------------------

def test_feature_extractor_buffer_accumulation_and_flatten():
    from fl4health_amd.model_bases.feature_extractor_buffer import FeatureExtractorBuffer

    model = nn.Sequential(nn.Flatten(), nn.Linear(12, 8), nn.ReLU(), nn.Linear(8, 3))
    buf = FeatureExtractorBuffer(model, {"1": True})
    buf._maybe_register_hooks()
    buf.enable_accumulating_features()
    model(torch.randn(4, 12))
    model(torch.randn(4, 12))
    assert len(buf.extracted_features_buffers["1"]) == 2
    feats = buf.get_extracted_features()["1"]
    assert feats.shape == (8, 8)  # two batches of 4, flattened dim 8
    # without accumulation the buffer holds only the latest batch
    buf.disable_accumulating_features()
    model(torch.randn(4, 12))
    assert len(buf.extracted_features_buffers["1"]) == 1
    buf.clear_buffers()
    assert all(len(v) == 0 for v in buf.extracted_features_buffers.values())
    buf.remove_hooks()
    model(torch.randn(4, 12))
    assert all(len(v) == 0 for v in buf.extracted_features_buffers.values())


# ---------------------------------------------------------------------------
# FedSimCLR
# ---------------------------------------------------------------------------

def test_fedsimclr_pretrain_vs_finetune_paths(tmp_path):
    from fl4health_amd.model_bases.fedsimclr_base import FedSimClrModel

    enc = nn.Sequential(nn.Flatten(), nn.Linear(12, 8))
    proj = nn.Linear(8, 4)
    pred = nn.Linear(8, 3)
    m = FedSimClrModel(enc, projection_head=proj, prediction_head=pred, pretrain=True)
    x = torch.randn(5, 12)
    assert m(x).shape == (5, 4)  # projection output during pretraining
    m.pretrain = False
    assert m(x).shape == (5, 3)  # prediction head during fine-tuning
    path = tmp_path / "simclr.pt"
    m.pretrain = True
    torch.save(m, path)
    loaded = FedSimClrModel.load_pretrained_model(str(path))
    assert loaded.pretrain is False  # loading flips to fine-tune mode
    assert loaded(x).shape == (5, 3)


def test_fedsimclr_without_prediction_head_requires_pretrain():
    from fl4health_amd.model_bases.fedsimclr_base import FedSimClrModel

    m = FedSimClrModel(nn.Sequential(nn.Flatten(), nn.Linear(12, 8)), pretrain=False)
    with pytest.raises(AssertionError):
        m(torch.randn(2, 12))


# ---------------------------------------------------------------------------
# Ensemble
# ---------------------------------------------------------------------------

def test_ensemble_average_and_vote():
    from fl4health_amd.model_bases.ensemble_base import EnsembleAggregationMode, EnsembleModel

    torch.manual_seed(1)
    models = {f"model_{i}": tiny_net() for i in range(3)}
    avg = EnsembleModel(dict(models), EnsembleAggregationMode.AVERAGE)
    x = torch.randn(4, 12)
    out = avg(x)
    assert set(out) == {"model_0", "model_1", "model_2", "ensemble-pred"}
    manual = torch.stack([out["model_0"], out["model_1"], out["model_2"]]).mean(0)
    assert torch.allclose(out["ensemble-pred"], manual, atol=1e-6)
    vote = EnsembleModel(dict(models), EnsembleAggregationMode.VOTE)
    vout = vote(x)["ensemble-pred"]
    assert vout.shape == (4, 3)
    assert torch.allclose(vout.sum(dim=-1), torch.ones(4))  # normalized vote shares


# ---------------------------------------------------------------------------
# GPFL pieces
# ---------------------------------------------------------------------------

def test_gpfl_gce_lookup_and_loss_scalar():
    from fl4health_amd.model_bases.gpfl_base import CoV, Gce

    gce = Gce(feature_dim=6, num_classes=4)
    labels = torch.tensor([0, 2, 3])
    emb = gce.lookup(labels)
    assert emb.shape == (3, 6)
    loss = gce(torch.randn(3, 6), labels)
    assert loss.dim() == 0 and torch.isfinite(loss)
    cov = CoV(feature_dim=6)
    out = cov(torch.randn(3, 6), torch.randn(3, 6))
    assert out.shape == (3, 6)


# ---------------------------------------------------------------------------
# FENDA / parallel-split
# ---------------------------------------------------------------------------

def test_fenda_model_joins_local_and_global_features():
    from fl4health_amd.model_bases.fenda_base import FendaModel
    from fl4health_amd.model_bases.parallel_split_models import (
        ParallelFeatureJoinMode,
        ParallelSplitHeadModule,
    )

    class Head(ParallelSplitHeadModule):
        def __init__(self):
            super().__init__(ParallelFeatureJoinMode.CONCATENATE)
            self.fc = nn.Linear(10, 3)

        def parallel_output_join(self, local_tensor, global_tensor):
            return torch.cat([local_tensor, global_tensor], dim=1)

        def head_forward(self, input_tensor):
            return self.fc(input_tensor)

    local = nn.Sequential(nn.Flatten(), nn.Linear(12, 5))
    glob = nn.Sequential(nn.Flatten(), nn.Linear(12, 5))
    m = FendaModel(local, glob, Head())
    x = torch.randn(4, 12)
    out = m(x)
    pred = out[0] if isinstance(out, tuple) else out
    if isinstance(pred, dict):
        pred = next(iter(pred.values()))
    assert pred.shape == (4, 3)
    # only the global branch (second extractor) is exchanged
    names = m.layers_to_exchange()
    assert names and all(n.startswith("second_feature_extractor.") for n in names)
    assert m.global_module is m.second_feature_extractor
