"""ResNet frame-wise features: (num_frames, 512/2048) per video.

Capability parity with the reference's ``ExtractResNet``
(reference models/resnet/extract_resnet.py): streaming decode with
batch-size accumulation, ImageNet eval preprocessing, per-frame
``timestamps_ms``, ``--show_pred`` top-5 over ImageNet.
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import transforms as T
from ..io.sampling import timestamps_ms
from ..io.video import open_video
from ..models.resnet import build_resnet
from .base import BaseExtractor, decode_ahead


class ExtractResNet(BaseExtractor):
    def __init__(self, args, external_call: bool = False):
        super().__init__(args, external_call)
        self.batch_size = max(1, self.cfg.batch_size)

    def build_models(self, device: torch.device, dtype: torch.dtype):
        model = build_resnet(self.feature_type)
        if self.cfg.weights_path:
            self.load_weights(model, self.cfg.weights_path)
        return model.to(device=device, dtype=dtype).eval()

    def extract(self, device: torch.device, model,
                video_path, prepared=None) -> Dict[str, np.ndarray]:
        reader = open_video(video_path, self.tmp_path, self.extraction_fps)
        fps = reader.fps
        dtype = self.compute_dtype(device)
        feats, idx_done = [], []
        n = reader.frame_count
        pin = device.type == 'cuda'

        def read_chunk(idxs):
            t = torch.from_numpy(reader.read_frames(idxs))
            return t.pin_memory() if pin else t

        chunks = [list(range(s, min(s + self.batch_size, n)))
                  for s in range(0, n, self.batch_size)]
        for idxs, frames_u8 in decode_ahead(read_chunk, chunks):
            if device.type == 'cuda':   # preprocess on the GPU (u8 upload)
                frames_u8 = frames_u8.to(device, non_blocking=True)
            batch = T.imagenet_preprocess(frames_u8).to(dtype)
            feats.append(model.forward_features(batch).float().cpu())
            idx_done.extend(idxs)
            if self.show_pred:
                from ..utils.labels import show_predictions_on_dataset
                show_predictions_on_dataset(model.fc(feats[-1].to(device=device,
                                                                  dtype=dtype)),
                                            'imagenet')
        features = torch.cat(feats).numpy()
        return {
            self.feature_type: features,
            'fps': np.array(fps),
            'timestamps_ms': np.array(timestamps_ms(np.array(idx_done), fps)),
        }
