"""RAFT optical-flow extractor: (T-1, 2, H, W) flow frames per video.

Capability parity with the reference's ``ExtractRAFT``
(reference models/raft/extract_raft.py): streaming batches with a 1-frame
carry-over for continuity, pad-to-/8 via InputPadder with unpad before
save, optional ``--side_size`` / ``--resize_to_larger_edge`` resize,
``--show_pred`` flow-statistics print (the reference pops an OpenCV GUI
window; headless here).
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import transforms as T
from ..io.video import open_video
from ..models.raft import RAFT, InputPadder
from .base import BaseExtractor, decode_ahead


class ExtractRAFT(BaseExtractor):
    def __init__(self, args, external_call: bool = False):
        super().__init__(args, external_call)
        self.batch_size = max(2, self.cfg.batch_size + 1)
        self.side_size = self.cfg.side_size
        self.resize_smaller = self.cfg.resize_to_smaller_edge

    def build_models(self, device: torch.device, dtype: torch.dtype):
        model = RAFT()
        if self.cfg.weights_path:
            self.load_weights(model, self.cfg.weights_path)
        return model.to(device=device, dtype=dtype).eval()

    def _prep(self, frames_u8: torch.Tensor) -> torch.Tensor:
        x = frames_u8.permute(0, 3, 1, 2).float()   # uint8-range (T,3,H,W)
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
        # (start, stop) windows with the reference's 1-frame carry-over
        # (extract_raft.py:143-146); the next window's decode runs ahead on
        # a worker thread while this window's flow runs on the GPU
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
            padder = InputPadder(batch.shape)
            im1, im2 = padder.pad(batch[:-1], batch[1:])
            flow = model(im1, im2, test_mode=True)
            flows.append(padder.unpad(flow).float().cpu())
            if self.show_pred:
                # the reference pops a cv2 window of the Middlebury flow
                # visualization (reference extract_raft.py:165-178); this
                # environment is headless, so save the image + print stats
                from ..utils.flow_viz import flow_to_image, save_ppm
                import os as _os
                mag = flow.norm(dim=1)
                print(f'flow frames {start}-{stop - 1}: '
                      f'|flow| mean {mag.mean():.3f} max {mag.max():.3f}')
                _os.makedirs(self.tmp_path, exist_ok=True)
                img = flow_to_image(
                    padder.unpad(flow)[0].float().cpu().numpy()
                    .transpose(1, 2, 0))
                save_ppm(_os.path.join(
                    self.tmp_path, f'flow_vis_{start:05d}.ppm'), img)
        features = torch.cat(flows).numpy() if flows else np.zeros((0, 2, 0, 0))
        return {
            self.feature_type: features,
            'fps': np.array(fps),
            'timestamps_ms': np.array([i * 1000.0 / fps for i in range(1, n)]),
        }
