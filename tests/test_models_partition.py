"""Partition-correctness tests (CPU).

For every model: full forward == composition of partition forwards at several cut
points, and state_dict keys partition the full model's keys exactly — the
property the server's partial-key slicing (reference src/Server.py:241-254) and
the .pth format compatibility depend on.
"""

import pytest
import torch

from split_learning_amd.models import (build_partition, get_model_class)

CASES = [
    # (model, data, input factory, cuts to try)
    ("VGG16", "CIFAR10", lambda: torch.randn(2, 3, 32, 32), [1, 7, 14, 24, 45]),
    ("VGG16", "MNIST", lambda: torch.randn(2, 1, 28, 28), [7, 34]),
    ("MobileNetv1", "CIFAR10", lambda: torch.randn(2, 3, 32, 32), [3, 40]),
    ("MobileNetv1", "MNIST", lambda: torch.randn(2, 1, 28, 28), [12]),
    ("ViT", "CIFAR10", lambda: torch.randn(2, 3, 32, 32), [2, 6]),
    ("ViT", "MNIST", lambda: torch.randn(2, 1, 28, 28), [6]),
    ("KWT", "SPEECHCOMMANDS", lambda: torch.randn(2, 40, 98), [2, 7]),
    ("BERT", "AGNEWS", lambda: torch.randint(0, 1000, (2, 16)), [2, 13]),
    ("BERT", "EMOTION", lambda: torch.randint(0, 1000, (2, 16)), [1, 13]),
]


@pytest.mark.parametrize("model_name,data_name,make_x,cuts",
                         CASES, ids=[f"{m}_{d}" for m, d, _, _ in CASES])
def test_partition_composition(model_name, data_name, make_x, cuts):
    torch.manual_seed(0)
    klass = get_model_class(model_name, data_name)
    full = klass().eval()
    x = make_x()
    with torch.no_grad():
        y_full = full(x)

    sd = full.state_dict()
    for cut in cuts:
        first = build_partition(model_name, data_name, [0, cut]).eval()
        last = build_partition(model_name, data_name, [cut, -1]).eval()
        # load partition weights from the full state dict (server-side slicing)
        for part in (first, last):
            psd = part.state_dict()
            part.load_state_dict({k: sd[k] for k in psd.keys()})
        with torch.no_grad():
            y_split = last(first(x))
        assert torch.allclose(y_full, y_split, atol=1e-5), \
            f"{model_name}/{data_name} cut={cut}: split forward != full forward"


@pytest.mark.parametrize("model_name,data_name,make_x,cuts",
                         CASES, ids=[f"{m}_{d}" for m, d, _, _ in CASES])
def test_state_dict_partition_of_keys(model_name, data_name, make_x, cuts):
    klass = get_model_class(model_name, data_name)
    full = klass()
    full_keys = set(full.state_dict().keys())
    for cut in cuts:
        first = build_partition(model_name, data_name, [0, cut])
        last = build_partition(model_name, data_name, [cut, -1])
        k1 = set(first.state_dict().keys())
        k2 = set(last.state_dict().keys())
        assert k1.isdisjoint(k2)
        assert k1 | k2 == full_keys, (
            f"cut={cut}: missing={full_keys - (k1 | k2)}, extra={(k1 | k2) - full_keys}")


def test_three_stage_composition_vgg16():
    torch.manual_seed(0)
    full = get_model_class("VGG16", "CIFAR10")().eval()
    sd = full.state_dict()
    x = torch.randn(2, 3, 32, 32)
    s1 = build_partition("VGG16", "CIFAR10", [0, 7]).eval()
    s2 = build_partition("VGG16", "CIFAR10", [7, 14]).eval()
    s3 = build_partition("VGG16", "CIFAR10", [14, -1]).eval()
    for part in (s1, s2, s3):
        part.load_state_dict({k: sd[k] for k in part.state_dict().keys()})
    with torch.no_grad():
        assert torch.allclose(full(x), s3(s2(s1(x))), atol=1e-5)


def test_expected_vgg16_state_dict_names():
    """Spot-check the exact key names the reference .pth format uses."""
    full = get_model_class("VGG16", "CIFAR10")()
    keys = set(full.state_dict().keys())
    for expect in ["layer1.weight", "layer1.bias", "layer2.weight", "layer2.bias",
                   "layer2.running_mean", "layer2.running_var",
                   "layer2.num_batches_tracked", "layer47.weight", "layer52.bias"]:
        assert expect in keys, f"missing {expect}"
    # ReLU/pool/flatten/dropout units carry no params
    assert not any(k.startswith("layer3.") for k in keys)
    assert not any(k.startswith("layer7.") for k in keys)


def test_expected_kwt_bare_params():
    full = get_model_class("KWT", "SPEECHCOMMANDS")()
    keys = set(full.state_dict().keys())
    assert "cls_token" in keys and "pos_embed" in keys
    assert "layer4.mha.in_proj_weight" in keys
    assert "layer4.mlp.0.weight" in keys


def test_expected_bert_agnews_names():
    full = get_model_class("BERT", "AGNEWS")()
    keys = set(full.state_dict().keys())
    for expect in ["layer1.word_embeddings.weight", "layer1.LayerNorm.weight",
                   "layer2.attention.self.query.weight",
                   "layer2.intermediate.dense.weight", "layer2.output.LayerNorm.bias",
                   "layer14.dense.weight", "layer15.classifier.weight"]:
        assert expect in keys, f"missing {expect}"


def test_expected_bert_emotion_names():
    full = get_model_class("BERT", "EMOTION")()
    keys = set(full.state_dict().keys())
    for expect in ["layer2.0.query.weight", "layer2.1.dense.weight",
                   "layer3.0.dense.weight", "layer3.1.LayerNorm.weight",
                   "layer26.dense.weight", "layer27.classifier.weight"]:
        assert expect in keys, f"missing {expect}"


def test_pth_roundtrip(tmp_path):
    full = get_model_class("VGG16", "CIFAR10")()
    p = tmp_path / "VGG16_CIFAR10.pth"
    torch.save(full.state_dict(), p)
    loaded = torch.load(p, weights_only=True)
    part = build_partition("VGG16", "CIFAR10", [0, 7])
    part.load_state_dict({k: loaded[k] for k in part.state_dict().keys()})


def test_end_layer_minus_one_means_total():
    m = build_partition("VGG16", "CIFAR10", [14, -1])
    assert m.end_layer == 52
    m = build_partition("BERT", "AGNEWS", [2, -1])
    assert m.end_layer == 15
