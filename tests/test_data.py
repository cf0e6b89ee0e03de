"""Data layer tests: MFCC frontend parity properties, synthetic distribution
honouring, loader determinism."""

import numpy as np
import torch

from split_learning_amd.data import data_loader
from split_learning_amd.data.mfcc import compute_mfcc
from split_learning_amd.data.synthetic import synthetic_tensors


def test_mfcc_shape_and_dynamics():
    rng = np.random.default_rng(0)
    wav = rng.standard_normal(16000).astype(np.float32)
    m = compute_mfcc(wav)
    # reference geometry: 40 coefficients x 98 frames (KWT input,
    # src/model/KWT_SPEECHCOMMANDS.py N_MFCC=40 TIME_STEPS=98)
    assert m.shape == (40, 98)
    assert np.isfinite(m).all()
    # a pure tone concentrates energy in few mel bands -> different first
    # coefficients than white noise
    t = np.arange(16000) / 16000.0
    tone = np.sin(2 * np.pi * 440 * t).astype(np.float32)
    m2 = compute_mfcc(tone)
    assert m2.shape == (40, 98)
    assert abs(m2[0].mean() - m[0].mean()) > 1.0


def test_mfcc_matches_reference_algorithm():
    """Independent per-frame recomputation (straight from the published
    algorithm) must match the vectorised implementation."""
    from scipy.fftpack import dct
    rng = np.random.default_rng(1)
    wav = rng.standard_normal(16000)
    ours = compute_mfcc(wav)

    emphasized = np.append(wav[0], wav[1:] - 0.97 * wav[:-1])
    n_fft, hop, n_mels = 480, 160, 40
    num_frames = 1 + (len(emphasized) - n_fft) // hop
    ham = np.hamming(n_fft)
    high = 2595 * np.log10(1 + 8000 / 700)
    mel_pts = np.linspace(0, high, n_mels + 2)
    hz = 700 * (10 ** (mel_pts / 2595) - 1)
    bins = np.floor((n_fft + 1) * hz / 16000).astype(int)
    fbank = np.zeros((n_mels, n_fft // 2 + 1))
    for m in range(1, n_mels + 1):
        for k in range(bins[m - 1], bins[m]):
            fbank[m - 1, k] = (k - bins[m - 1]) / (bins[m] - bins[m - 1])
        for k in range(bins[m], bins[m + 1]):
            fbank[m - 1, k] = (bins[m + 1] - k) / (bins[m + 1] - bins[m])
    rows = []
    for i in range(num_frames):
        fr = emphasized[i * hop: i * hop + n_fft] * ham
        p = np.abs(np.fft.rfft(fr, n_fft)) ** 2 / n_fft
        fb = p @ fbank.T
        fb = np.where(fb == 0, np.finfo(float).eps, fb)
        rows.append(20 * np.log10(fb))
    ref = dct(np.stack(rows), type=2, axis=1, norm="ortho")[:, :40].T
    np.testing.assert_allclose(ours, ref, rtol=1e-9, atol=1e-9)


def test_synthetic_distribution_counts():
    dist = [5, 0, 3, 0, 0, 7, 0, 0, 0, 1]
    x, y = synthetic_tensors("CIFAR10", dist)
    assert x.shape[0] == 16
    counts = torch.bincount(y, minlength=10).tolist()
    assert counts == dist


def test_loader_determinism():
    l1 = data_loader("CIFAR10", 8, [8] * 10, train=True, seed=3)
    l2 = data_loader("CIFAR10", 8, [8] * 10, train=True, seed=3)
    for (x1, y1), (x2, y2) in zip(l1, l2):
        assert torch.equal(x1, x2) and torch.equal(y1, y2)
