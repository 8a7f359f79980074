"""PWC-Net optical-flow extractor: (T-1, 2, H, W) flow frames per video.

Capability parity with the reference's ``ExtractPWC``
(reference models/pwc/extract_pwc.py): same streaming loop as RAFT minus the
/8 padder (PWC pads to /64 internally via bilinear resize).
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import transforms as T
from ..io.video import open_video
from ..models.pwc import PWCNet
from .base import BaseExtractor, decode_ahead


class ExtractPWC(BaseExtractor):
    def __init__(self, args, external_call: bool = False):
        super().__init__(args, external_call)
        self.batch_size = max(2, self.cfg.batch_size + 1)
        self.side_size = self.cfg.side_size
        self.resize_smaller = self.cfg.resize_to_smaller_edge

    def build_models(self, device: torch.device, dtype: torch.dtype):
        model = PWCNet()
        if self.cfg.weights_path:
            self.load_weights(model, self.cfg.weights_path)
        return model.to(device=device, dtype=dtype).eval()

    def _prep(self, frames_u8: torch.Tensor) -> torch.Tensor:
        x = frames_u8.permute(0, 3, 1, 2).float()
        if self.side_size:
            x = T.resize_improved(x, self.side_size, self.resize_smaller)
        return x

    def extract(self, device: torch.device, model,
                video_path, prepared=None) -> Dict[str, np.ndarray]:
        reader = open_video(video_path, self.tmp_path, self.extraction_fps)
        fps = reader.fps
        n = reader.frame_count
        dtype = self.compute_dtype(device)
        flows = []
        # (start, stop) windows with 1-frame carry-over; next window decodes
        # ahead on a worker thread (see raft extractor)
        spans, start = [], 0
        while start < n - 1:
            stop = min(start + self.batch_size, n)
            spans.append((start, stop))
            start = stop - 1
        pin = device.type == 'cuda'

        def read_chunk(span):
            t = torch.from_numpy(reader.read_frames(range(*span)))
            return t.pin_memory() if pin else t

        for (start, stop), frames in decode_ahead(read_chunk, spans):
            batch = self._prep(frames).to(device=device, dtype=dtype,
                                          non_blocking=True)
            flow = model(batch[:-1], batch[1:])
            flows.append(flow.float().cpu())
            if self.show_pred:
                # headless flow visualization (see raft extractor note)
                from ..utils.flow_viz import flow_to_image, save_ppm
                import os as _os
                mag = flow.norm(dim=1)
                print(f'flow frames {start}-{stop - 1}: '
                      f'|flow| mean {mag.mean():.3f} max {mag.max():.3f}')
                _os.makedirs(self.tmp_path, exist_ok=True)
                img = flow_to_image(
                    flow[0].float().cpu().numpy().transpose(1, 2, 0))
                save_ppm(_os.path.join(
                    self.tmp_path, f'flow_vis_{start:05d}.ppm'), img)
        features = torch.cat(flows).numpy() if flows else np.zeros((0, 2, 0, 0))
        return {
            self.feature_type: features,
            'fps': np.array(fps),
            'timestamps_ms': np.array([i * 1000.0 / fps for i in range(1, n)]),
        }
