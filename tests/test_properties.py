"""Property-based tests (hypothesis) for the pure-logic core.

These pin invariants rather than examples: partition stage ranges must tile
the unit axis exactly (reference src/Server.py:221-228 semantics), and FedAvg
must be a true weighted mean with the reference's NaN->0 and integer
round-back rules (src/Utils.py:35-66)."""

import torch
from hypothesis import given, settings, strategies as st

from split_learning_amd.config import stage_ranges
from split_learning_amd.models import build_partition
from split_learning_amd.parallel.fedavg import fedavg_state_dicts


@given(st.integers(2, 6), st.data())
@settings(max_examples=50, deadline=None)
def test_stage_ranges_tile_the_unit_axis(n_stages, data):
    total = data.draw(st.integers(n_stages, 60))
    cuts = sorted(data.draw(st.lists(st.integers(1, total - 1),
                                     min_size=n_stages - 1,
                                     max_size=n_stages - 1, unique=True)))
    ranges = stage_ranges(cuts, n_stages, total)
    assert len(ranges) == n_stages
    assert ranges[0][0] == 0 and ranges[-1][1] == -1
    # consecutive ranges chain: stage k ends where stage k+1 begins, so the
    # active-unit predicate start < i <= end covers 1..total exactly once
    covered = []
    for (a, b) in ranges:
        end = total if b == -1 else b
        covered.extend(range(a + 1, end + 1))
    assert covered == list(range(1, total + 1))


@given(st.integers(1, 4), st.integers(1, 5))
@settings(max_examples=30, deadline=None)
def test_vgg_partition_active_units_disjoint(seed, n_stages_minus1):
    """build_partition honors the ranges: each of VGG16's 52 units has a
    parameter owner in exactly one stage."""
    n_stages = n_stages_minus1 + 1
    torch.manual_seed(seed)
    total = 52
    cuts = sorted(torch.randperm(total - 2)[:n_stages - 1].add(1).tolist())
    parts = [build_partition("VGG16", "CIFAR10", r)
             for r in stage_ranges(cuts, n_stages, total)]
    full = build_partition("VGG16", "CIFAR10", [0, 0])
    part_keys = [set(p.state_dict().keys()) for p in parts]
    for i in range(len(part_keys)):
        for j in range(i + 1, len(part_keys)):
            assert not (part_keys[i] & part_keys[j]), (cuts, i, j)
    assert set().union(*part_keys) == set(full.state_dict().keys()), cuts


@given(st.integers(1, 4), st.lists(st.floats(0.5, 100.0), min_size=2,
                                   max_size=4))
@settings(max_examples=40, deadline=None)
def test_fedavg_is_weighted_mean(seed, weights):
    torch.manual_seed(seed)
    n = len(weights)
    sds = [{"w": torch.randn(7, 3), "b": torch.randn(5)} for _ in range(n)]
    out = fedavg_state_dicts(sds, weights)
    tw = sum(weights)
    for key in ("w", "b"):
        expect = sum(sd[key] * w for sd, w in zip(sds, weights)) / tw
        assert torch.allclose(out[key], expect, atol=1e-5), key


def test_fedavg_nan_and_int_rules():
    a = {"w": torch.tensor([1.0, float("nan")]), "steps": torch.tensor([3])}
    b = {"w": torch.tensor([3.0, 2.0]), "steps": torch.tensor([4])}
    out = fedavg_state_dicts([a, b])
    # NaN -> 0 BEFORE averaging (reference src/Utils.py:51-52)
    assert torch.allclose(out["w"], torch.tensor([2.0, 1.0]))
    # integer tensors round back to the original dtype
    assert out["steps"].dtype == torch.int64
    assert out["steps"].item() == 4  # (3+4)/2 = 3.5 -> round -> 4


def test_fedavg_key_union():
    """Keys present in only some clients still appear (union semantics)."""
    a = {"w": torch.ones(2)}
    b = {"w": torch.ones(2) * 3.0, "extra": torch.ones(1) * 6.0}
    out = fedavg_state_dicts([a, b])
    assert torch.allclose(out["w"], torch.ones(2) * 2.0)
    # "extra" averaged over TOTAL weight (reference divides by all-client sum)
    assert torch.allclose(out["extra"], torch.ones(1) * 3.0)
