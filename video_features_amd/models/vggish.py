"""Native VGGish audio embedding model + log-mel frontend.

The reference ships two copies of this (TF1 slim and torch — reference
models/vggish/vggish_src/* and models/vggish_torch/vggish_src/*) plus a
pure-numpy STFT frontend (reference mel_features.py).  Here there is ONE
implementation serving both feature-type strings; the frontend is
torch-native (batched rFFT) so it runs on GPU and the same code is the CPU
reference.

Frontend parameters (reference vggish_params.py:22-35): 16 kHz mono, 25 ms
window / 10 ms hop, periodic Hann, |rFFT| at 512 points, 64 HTK-mel bands
125–7500 Hz, log(mel + 0.01), framed into 0.96 s examples of 96 frames.

Network (reference vggish.py:108-118): VGG stack
[64, M, 128, M, 256, 256, M, 512, 512, M] on (N, 1, 96, 64) → TF-order
flatten → FC 12288→4096→4096→128.  ``Postprocessor`` applies the released
PCA whitening + [-2, 2] clip + 8-bit quantization (reference
vggish.py:34-105); with random-init weights the PCA defaults to identity.
"""
from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch
import torch.nn.functional as F
from torch import nn

# ---- frontend parameters
SAMPLE_RATE = 16000
STFT_WINDOW_SEC = 0.025
STFT_HOP_SEC = 0.010
NUM_MEL_BINS = 64
MEL_MIN_HZ = 125.0
MEL_MAX_HZ = 7500.0
LOG_OFFSET = 0.01
EXAMPLE_WINDOW_SEC = 0.96
EXAMPLE_HOP_SEC = 0.96
EMBEDDING_SIZE = 128


def _hz_to_mel(f):
    return 1127.0 * np.log(1.0 + np.asarray(f, dtype=np.float64) / 700.0)


def mel_filterbank(num_spectrogram_bins: int, sample_rate: int = SAMPLE_RATE,
                   num_mel_bins: int = NUM_MEL_BINS,
                   lower_hz: float = MEL_MIN_HZ,
                   upper_hz: float = MEL_MAX_HZ) -> np.ndarray:
    """(num_spectrogram_bins, num_mel_bins) triangular filters on the HTK mel
    scale (reference mel_features.py:114-189 semantics)."""
    nyquist = sample_rate / 2.0
    spectrogram_hz = np.linspace(0.0, nyquist, num_spectrogram_bins)
    spectrogram_mel = _hz_to_mel(spectrogram_hz)
    band_edges_mel = np.linspace(_hz_to_mel(lower_hz), _hz_to_mel(upper_hz),
                                 num_mel_bins + 2)
    weights = np.zeros((num_spectrogram_bins, num_mel_bins))
    for i in range(num_mel_bins):
        lo, center, hi = band_edges_mel[i:i + 3]
        lower_slope = (spectrogram_mel - lo) / (center - lo)
        upper_slope = (hi - spectrogram_mel) / (hi - center)
        weights[:, i] = np.maximum(0.0, np.minimum(lower_slope, upper_slope))
    weights[0, :] = 0.0   # DC bin excluded (reference mel_features.py:186)
    return weights


def waveform_to_examples(samples: torch.Tensor,
                         device: Optional[torch.device] = None) -> torch.Tensor:
    """16 kHz mono float waveform (T,) → (N, 96, 64) log-mel examples.

    Torch-native equivalent of the reference's numpy pipeline
    (reference mel_features.py:192-223 + vggish_input.py).
    """
    if device is not None:
        samples = samples.to(device)
    samples = samples.float()
    win = int(round(SAMPLE_RATE * STFT_WINDOW_SEC))      # 400
    hop = int(round(SAMPLE_RATE * STFT_HOP_SEC))         # 160
    fft_len = 2 ** int(math.ceil(math.log2(win)))        # 512
    if samples.numel() < win:
        samples = F.pad(samples, (0, win - samples.numel()))
    n_frames = 1 + (samples.numel() - win) // hop
    idx = (torch.arange(n_frames, device=samples.device)[:, None] * hop
           + torch.arange(win, device=samples.device)[None, :])
    frames = samples[idx]                                 # (F, win)
    # periodic Hann (reference mel_features.py:48-68)
    window = 0.5 - 0.5 * torch.cos(
        2 * math.pi * torch.arange(win, device=samples.device) / win)
    spec = torch.fft.rfft(frames * window, n=fft_len).abs()   # (F, 257)
    mel = torch.from_numpy(
        mel_filterbank(fft_len // 2 + 1)).to(samples.device, torch.float32)
    log_mel = torch.log(spec @ mel + LOG_OFFSET)          # (F, 64)
    # frame into 0.96 s examples
    frames_per_example = int(round(EXAMPLE_WINDOW_SEC / STFT_HOP_SEC))   # 96
    n_ex = log_mel.shape[0] // frames_per_example
    if n_ex == 0:
        raise ValueError('audio shorter than one 0.96 s VGGish example')
    return log_mel[:n_ex * frames_per_example].reshape(
        n_ex, frames_per_example, NUM_MEL_BINS)


class VGGishNet(nn.Module):
    def __init__(self):
        super().__init__()
        cfg = [64, 'M', 128, 'M', 256, 256, 'M', 512, 512, 'M']
        layers, in_ch = [], 1
        for v in cfg:
            if v == 'M':
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(in_ch, v, 3, 1, 1), nn.ReLU(inplace=True)]
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.embeddings = nn.Sequential(
            nn.Linear(512 * 4 * 6, 4096), nn.ReLU(inplace=True),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True),
            nn.Linear(4096, EMBEDDING_SIZE), nn.ReLU(inplace=True))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """(N, 96, 64) log-mel examples → (N, 128) embeddings."""
        x = x[:, None]
        if (x.is_cuda and x.dtype == torch.bfloat16
            and __import__('video_features_amd.ops', fromlist=['o'])
                .hip_available()):
            from .. import ops
            # conv+ReLU pairs through the in-tree implicit-GEMM kernel
            # (first conv has C=1 → eager fallback inside conv2d_mod)
            x = x.contiguous(memory_format=torch.channels_last)
            mods = list(self.features)
            i = 0
            while i < len(mods):
                if isinstance(mods[i], nn.Conv2d):
                    x = ops.conv2d_mod(mods[i], x, 'relu')
                    i += 2                       # skip the fused ReLU
                else:
                    x = mods[i](x)
                    i += 1
        else:
            x = self.features(x)                 # (N, 512, 6, 4)
        # TF-order flatten (reference vggish.py:22-29)
        x = x.permute(0, 2, 3, 1).contiguous().flatten(1)
        return self.embeddings(x)


class Postprocessor(nn.Module):
    """PCA whitening + clip + 8-bit quantization
    (reference vggish.py:34-105).  Identity PCA when no released params."""

    def __init__(self):
        super().__init__()
        self.register_buffer('pca_matrix', torch.eye(EMBEDDING_SIZE))
        self.register_buffer('pca_means', torch.zeros(EMBEDDING_SIZE, 1))
        self.quant_min, self.quant_max = -2.0, 2.0

    def forward(self, embeddings: torch.Tensor) -> torch.Tensor:
        x = self.pca_matrix @ (embeddings.t() - self.pca_means)
        x = x.t().clamp(self.quant_min, self.quant_max)
        x = (x - self.quant_min) * (255.0 / (self.quant_max - self.quant_min))
        return x.round()


class VGGish(nn.Module):
    def __init__(self, postprocess: bool = False):
        super().__init__()
        self.net = VGGishNet()
        self.postprocess = postprocess
        self.pproc = Postprocessor() if postprocess else None

    def forward(self, examples: torch.Tensor) -> torch.Tensor:
        out = self.net(examples)
        if self.postprocess:
            out = self.pproc(out.float())
        return out
