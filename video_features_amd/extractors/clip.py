"""CLIP feature extractor: 512-d embedding per sampled frame.

Capability parity with the reference's ``ExtractCLIP``
(reference models/CLIP/extract_clip.py): ``uni_N``/``fix_N`` frame sampling,
feature shape (N, 512), meta keys fps/timestamps_ms, ``external_call`` API,
feature types CLIP-ViT-B/32, CLIP-ViT-B/16, CLIP4CLIP-ViT-B-32 (the
CLIP4CLIP variant loads task-tuned ViT-B/32 weights from ``weights_path``).
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import transforms as T
from ..io.sampling import sample_indices, timestamps_ms
from ..io.video import open_video
from ..models.clip_resnet import build_clip_resnet
from ..models.clip_vit import build_clip_vit
from .base import BaseExtractor


class ExtractCLIP(BaseExtractor):
    def __init__(self, args, external_call: bool = False):
        super().__init__(args, external_call)
        self.extract_method = self.cfg.extract_method or 'uni_12'
        self.batch_size = max(1, self.cfg.batch_size)

    def build_models(self, device: torch.device, dtype: torch.dtype):
        if self.feature_type.startswith('CLIP-RN'):
            model = build_clip_resnet(self.feature_type)
        else:
            model = build_clip_vit(self.feature_type)
        wp = self.cfg.weights_path
        if not wp and self.feature_type == 'CLIP4CLIP-ViT-B-32':
            # reference convenience: the CLIP4CLIP variant auto-loads its
            # local checkpoint (reference models/CLIP/extract_clip.py:58-63
            # uses models/CLIP/checkpoints/CLIP4CLIP-ViT-B-32.pth); here the
            # conventional spots are ./checkpoints/ and the weights cache
            from ..utils.weights import cache_dir
            import os
            for cand in (os.path.join('checkpoints',
                                      'CLIP4CLIP-ViT-B-32.pth'),
                         os.path.join(cache_dir(),
                                      'CLIP4CLIP-ViT-B-32.pth')):
                if os.path.exists(cand):
                    wp = cand
                    break
        if wp:
            self.load_weights(model, wp)
        model = model.to(device=device, dtype=dtype).eval()
        return model

    def prepare(self, video_path):
        """CPU decode + frame sampling — runs one video ahead on the
        decode thread (overlaps the previous video's GPU forward)."""
        reader = open_video(video_path, self.tmp_path, self.extraction_fps)
        fps = reader.fps
        idxs = sample_indices(self.extract_method, reader.frame_count, fps)
        frames = torch.from_numpy(reader.read_frames(idxs))
        if torch.cuda.is_available():
            frames = frames.pin_memory()     # async-capable H2D upload
        return frames, fps, idxs

    def extract(self, device: torch.device, model,
                video_path, prepared=None) -> Dict[str, np.ndarray]:
        with self._prof('decode'):
            frames_u8, fps, idxs = (prepared if prepared is not None
                                    else self.prepare(video_path))
        with self._prof('preprocess'):
            # the transforms are pure torch — run them ON the GPU (the u8
            # upload is 4-12x smaller than uploading preprocessed floats,
            # and CPU-side bicubic was 87% of end-to-end extractor wall)
            if device.type == 'cuda':
                frames_u8 = frames_u8.to(device, non_blocking=True)
            res = getattr(model, 'input_resolution', None) \
                or model.cfg.input_resolution
            batch = T.clip_preprocess(frames_u8, res)
        dtype = self.compute_dtype(device)
        feats = []
        # reference semantics (extract_clip.py:125-128): the whole sampled
        # stack goes through encode_image in ONE forward; batch_size is an
        # additive knob here — the default (1) keeps whole-stack behavior,
        # an explicit --batch_size B is honored exactly (e.g. to fit memory)
        bs = self.batch_size if self.batch_size > 1 else batch.shape[0]
        with self._prof('infer'):
            for s in range(0, batch.shape[0], bs):
                chunk = batch[s:s + bs]
                chunk = chunk.to(device=device, dtype=dtype, non_blocking=True)
                feats.append(model.encode_image(chunk).float().cpu())
            features = torch.cat(feats).numpy()
        return {
            self.feature_type: features,
            'fps': np.array(fps),
            'timestamps_ms': np.array(timestamps_ms(idxs, fps)),
        }
