"""GPU convergence/accuracy tests (round-1 VERDICT missing #2): the engine
must LEARN (accuracy far above chance on a learnable synthetic task of the
headline shape) and the cut=7 split pipeline must reach the same accuracy as
monolithic training with identical seeds and data order."""

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")


@needs_gpu
@pytest.mark.timeout(600)
def test_split_learns_and_matches_monolithic():
    import sys
    import os
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from tools.convergence_acc import make_data, train_mono, train_split

    # calibration (MI355X, gpurun_out/acc_*.log): 800 steps -> acc ~0.17-0.20,
    # 1500 -> ~0.27, 5000 -> mono 0.569 / split 0.581 / stock-torch 0.531;
    # 2000 steps lands mid-trajectory in a few seconds per arm
    steps = 2000
    train, test = make_data(device="cuda")
    acc_mono, tr_mono = train_mono(train, test, steps, "cuda",
                                   log_every=steps // 5)
    acc_split, tr_split = train_split(train, test, steps, "cuda",
                                      log_every=steps // 5)
    # learns: far above 10% chance
    assert acc_mono > 0.22, f"monolithic failed to learn: {acc_mono}"
    assert acc_split > 0.22, f"split failed to learn: {acc_split}"
    # split == mono within run tolerance (same math at control-count 1;
    # kernel-order nondeterminism from atomics allows small drift)
    assert abs(acc_mono - acc_split) < 0.08, (acc_mono, acc_split)
    # loss decreased materially in both arms
    assert tr_mono[-1] < tr_mono[0] * 0.75, tr_mono
    assert tr_split[-1] < tr_split[0] * 0.75, tr_split
