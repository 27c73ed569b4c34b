"""GraphedPredictor: eager fallback on CPU, graph parity on GPU."""

import pytest
import torch

from gradient_accumulation_tf_estimator_amd.serving import GraphedPredictor


def test_cpu_fallback_matches_model():
    from gradient_accumulation_tf_estimator_amd.models.housing import HousingMLP

    torch.manual_seed(0)
    m = HousingMLP(hidden=(8, 4))
    x = torch.randn(16, 14)
    pred = GraphedPredictor(m, x)
    with torch.no_grad():
        ref = m(x)
    torch.testing.assert_close(pred(x), ref)


@pytest.mark.gpu
def test_gpu_graph_parity_and_shape_guard():
    from gradient_accumulation_tf_estimator_amd.models.bert import (
        BertConfig, BertForSequenceClassification)

    cfg = BertConfig(vocab_size=256, hidden_size=512, num_layers=2,
                     num_heads=8, intermediate_size=2048,
                     max_position_embeddings=128)
    torch.manual_seed(0)
    m = BertForSequenceClassification(cfg).to("cuda", torch.bfloat16)
    ids = torch.randint(0, 256, (8, 64), device="cuda")
    pred = GraphedPredictor(m, ids)
    ids2 = torch.randint(0, 256, (8, 64), device="cuda")
    with torch.no_grad():
        ref = m(ids2)
    got = pred(ids2)
    torch.testing.assert_close(got.float(), ref.float(), rtol=1e-3, atol=1e-3)
    with pytest.raises(ValueError):
        pred(torch.randint(0, 256, (4, 64), device="cuda"))
