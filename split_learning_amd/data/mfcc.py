"""Numpy MFCC frontend for SpeechCommands (40 coeffs x 98 frames from 1 s @
16 kHz) — behavioural parity with reference src/dataset/SPEECHCOMMANDS.py:11-47:
pre-emphasis (0.97), framed Hamming window (n_fft 480, hop 160), power
spectrum, 40-band mel filterbank (point-slope triangles on floor'd FFT bins),
log-power in dB, ortho DCT-II, first 40 coefficients, transposed to
[n_mfcc, frames].

Vectorised re-implementation (stride-tricks framing + precomputed filterbank)
rather than the reference's per-frame python loops; numerically equivalent.
"""

from __future__ import annotations

import numpy as np

_FBANK_CACHE: dict = {}


def _mel_filterbank(sample_rate: int, n_fft: int, n_mels: int) -> np.ndarray:
    key = (sample_rate, n_fft, n_mels)
    if key in _FBANK_CACHE:
        return _FBANK_CACHE[key]
    high_mel = 2595.0 * np.log10(1.0 + (sample_rate / 2.0) / 700.0)
    mel_pts = np.linspace(0.0, high_mel, n_mels + 2)
    hz_pts = 700.0 * (10.0 ** (mel_pts / 2595.0) - 1.0)
    bins = np.floor((n_fft + 1) * hz_pts / sample_rate).astype(int)
    fbank = np.zeros((n_mels, n_fft // 2 + 1))
    for m in range(1, n_mels + 1):
        lo, mid, hi = bins[m - 1], bins[m], bins[m + 1]
        for k in range(lo, mid):
            fbank[m - 1, k] = (k - lo) / (mid - lo)
        for k in range(mid, hi):
            fbank[m - 1, k] = (hi - k) / (hi - mid)
    _FBANK_CACHE[key] = fbank
    return fbank


def compute_mfcc(waveform: np.ndarray, sample_rate: int = 16000, n_mfcc: int = 40,
                 n_fft: int = 480, hop_length: int = 160,
                 n_mels: int = 40) -> np.ndarray:
    """waveform [T] float -> mfcc [n_mfcc, 1 + (T - n_fft)//hop]."""
    from scipy.fftpack import dct

    wav = np.asarray(waveform, dtype=np.float64)
    emphasized = np.concatenate(([wav[0]], wav[1:] - 0.97 * wav[:-1]))

    num_frames = 1 + (len(emphasized) - n_fft) // hop_length
    idx = (np.arange(num_frames)[:, None] * hop_length + np.arange(n_fft)[None, :])
    frames = emphasized[idx] * np.hamming(n_fft)[None, :]

    power = np.abs(np.fft.rfft(frames, n_fft)) ** 2 / n_fft
    fb = _mel_filterbank(sample_rate, n_fft, n_mels)
    mel = power @ fb.T
    mel = np.where(mel == 0, np.finfo(float).eps, mel)
    log_mel = 20.0 * np.log10(mel)
    mfcc = dct(log_mel, type=2, axis=1, norm="ortho")[:, :n_mfcc]
    return mfcc.T
