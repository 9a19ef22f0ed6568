"""Model/data factory coverage: every name the launcher and bench accept
constructs (tiny variants train a step on CPU) and the synthetic data is
shaped/typed for its model (SURVEY.md §4 'fixtures')."""

import pytest
import torch

from baton_amd.models.factory import create_model, make_data

SMALL = ["linreg10", "tinymlp", "bert-tiny", "llama-tiny"]


@pytest.mark.parametrize("name", SMALL)
def test_create_and_one_round(name):
    torch.manual_seed(0)
    model = create_model(name)
    data, n = make_data(name, 16, seed=1, seq_len=32)
    assert n == 16
    losses = model.train_round(*data, n_epoch=1)
    assert len(losses) == 1 and all(l == l for l in losses)  # no NaN


@pytest.mark.parametrize("name", ["resnet18", "resnet50"])
def test_resnet_factory_shapes(name):
    # constructing the full ResNets is cheap; just check data shape/dtype
    model = create_model(name)
    (x, y), n = make_data(name, 8, dtype=torch.float32)
    assert x.shape == (8, 32, 32, 3) and x.dtype == torch.float32  # NHWC
    assert y.shape == (8,) and y.dtype == torch.long
    assert sum(p.numel() for p in model.parameters()) > 1e6


def test_bert_tiny_seq_clamped():
    # tiny config max_positions = 64: longer requests are clamped, the
    # model must accept the produced batch (regression for the factory
    # IndexError on seq 128 > max_positions)
    model = create_model("bert-tiny")
    (ids, labels), _ = make_data("bert-tiny", 4, seq_len=128)
    assert ids.shape[1] <= 64
    h = model(ids)
    assert h.shape[:2] == ids.shape


def test_unknown_name_raises():
    with pytest.raises(ValueError):
        create_model("nope")
    with pytest.raises(ValueError):
        make_data("nope", 4)


def test_mlm_labels_convention():
    (ids, labels), _ = make_data("bert-tiny", 32, seq_len=32)
    masked = labels != -100
    assert masked.any(dim=1).all()          # every sample has >=1 mask
    assert (ids[masked] == 0).all()         # masked ids replaced by [MASK]=0
    assert (labels[~masked] == -100).all()  # HF ignore-index elsewhere
