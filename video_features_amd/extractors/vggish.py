"""VGGish audio extractor: (Ta, 128), one row per 0.96 s of audio.

Capability parity with BOTH reference variants — the TF1 ``vggish``
(reference models/vggish/extract_vggish.py) and torch ``vggish_torch``
(reference models/vggish_torch/extract_vggish.py) — served by the single
native implementation in models/vggish.py.  Audio arrives from a .wav
input, a sidecar .wav, or ffmpeg extraction (mp4→aac→wav, reference
utils/utils.py:247-276); tmp files are removed unless ``--keep_tmp_files``.
"""
from __future__ import annotations

import os
from typing import Dict

import numpy as np
import torch

from ..io.audio import load_audio_for_video, resample
from ..models.vggish import SAMPLE_RATE, VGGish, waveform_to_examples
from .base import BaseExtractor


class ExtractVGGish(BaseExtractor):
    def build_models(self, device: torch.device, dtype: torch.dtype):
        model = VGGish(postprocess=False)
        if self.cfg.weights_path:
            self.load_weights(model, self.cfg.weights_path)
        return model.to(device=device, dtype=dtype).eval()

    def prepare(self, video_path):
        # audio demux + resample + log-mel framing: the CPU-heavy half of
        # the VGGish pipeline, run one video ahead on the decode thread
        samples, sr, tmp_files = load_audio_for_video(
            video_path, self.tmp_path, self.keep_tmp_files)
        samples = resample(samples, sr, SAMPLE_RATE)
        wav = torch.from_numpy(np.ascontiguousarray(samples))
        return waveform_to_examples(wav, None), tmp_files

    def discard_prepared(self, prepared) -> None:
        # a prefetched-then-skipped video's tmp wavs must not leak
        _, tmp_files = prepared
        if not self.keep_tmp_files:
            for f in tmp_files:
                if os.path.exists(f):
                    os.remove(f)

    def extract(self, device: torch.device, model,
                video_path, prepared=None) -> Dict[str, np.ndarray]:
        examples, tmp_files = (prepared if prepared is not None
                               else self.prepare(video_path))
        try:
            examples = examples.to(device)
            dtype = self.compute_dtype(device)
            feats = model(examples.to(dtype)).float().cpu().numpy()
        finally:
            if not self.keep_tmp_files:
                for f in tmp_files:
                    if os.path.exists(f):
                        os.remove(f)
        n = feats.shape[0]
        return {
            self.feature_type: feats,
            'fps': np.array(float(SAMPLE_RATE)),
            'timestamps_ms': np.array([i * 960.0 for i in range(n)]),
        }
