"""R(2+1)D-18 clip features: (num_windows, 512), 16-frame stacks, step 16.

Capability parity with the reference's ``ExtractR21D``
(reference models/r21d/extract_r21d.py): whole-video decode, sliding
windows via ``form_slices``, Kinetics preprocessing to 112×112,
``--show_pred`` top-5 over Kinetics-400.
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import transforms as T
from ..io.sampling import form_slices
from ..io.video import open_video
from ..models.r21d import R2Plus1D18
from .base import BaseExtractor

STACK_SIZE = 16
STEP_SIZE = 16


class ExtractR21D(BaseExtractor):
    def __init__(self, args, external_call: bool = False):
        super().__init__(args, external_call)
        self.stack_size = self.cfg.stack_size or STACK_SIZE
        self.step_size = self.cfg.step_size or STEP_SIZE

    def build_models(self, device: torch.device, dtype: torch.dtype):
        model = R2Plus1D18()
        if self.cfg.weights_path:
            self.load_weights(model, self.cfg.weights_path)
        return model.to(device=device, dtype=dtype).eval()

    def prepare(self, video_path):
        # whole-video decode (the reference reads the full video too,
        # reference extract_r21d.py:102) on the decode thread
        reader = open_video(video_path, self.tmp_path, None)
        n = reader.frame_count
        frames = torch.from_numpy(reader.read_frames(range(n)))
        return frames, reader.fps, n

    def extract(self, device: torch.device, model,
                video_path, prepared=None) -> Dict[str, np.ndarray]:
        all_frames, fps, n = (prepared if prepared is not None
                              else self.prepare(video_path))
        slices = form_slices(n, self.stack_size, self.step_size)
        if not slices:
            slices = [(0, n)]   # shorter than one stack: use what exists
        # temporal parallelism: this rank owns every tp_world-th window
        slices = slices[self.cfg.tp_rank::self.cfg.tp_world]
        dtype = self.compute_dtype(device)
        feats, ts = [], []
        for (start, end) in slices:
            frames_u8 = all_frames[start:end]
            if device.type == 'cuda':   # preprocess on the GPU (u8 upload)
                frames_u8 = frames_u8.to(device, non_blocking=True)
            clip = T.r21d_preprocess(frames_u8)[None].to(dtype)
            feats.append(model.forward_features(clip).float().cpu())
            ts.append(start / fps * 1000.0)
            if self.show_pred:
                from ..utils.labels import show_predictions_on_dataset
                show_predictions_on_dataset(
                    model.fc(feats[-1].to(device=device, dtype=dtype)), 'kinetics')
        features = torch.cat(feats).numpy()
        return {
            self.feature_type: features,
            'fps': np.array(fps),
            'timestamps_ms': np.array(ts),
        }
