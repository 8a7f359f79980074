"""BaseExtractor — the shared pipeline controller every feature family uses.

The reference has *no* shared engine: each of its seven extractors
re-implements the decode→transform→infer→sink loop with copy-pasted
variations (see reference models/CLIP/extract_clip.py:22-135 and its six
siblings).  Here the loop lives once, and each family implements only
``build_models(device, dtype)`` and ``extract(device, path)``.

Preserved reference contracts:
- constructor ``ExtractX(args, external_call=False)`` where ``args`` is any
  namespace-like object (reference README.md:39-51);
- ``forward(indices: LongTensor)`` — indices into the resolved path list,
  ``indices.device`` selects the GPU (reference extract_clip.py:38-66);
- per-video try/except that logs and continues, KeyboardInterrupt re-raised
  (reference extract_clip.py:70-84);
- ``external_call=True`` returns the feats list instead of sinking;
- every feats dict carries ``fps`` and ``timestamps_ms`` meta keys.
"""
from __future__ import annotations

import traceback
from typing import Any, Dict, List

import numpy as np
import torch

from ..config import Config
from ..io.listing import form_list_from_user_input
from ..runtime.sinks import (action_on_extraction, make_output_path,
                             outputs_exist)
from ..runtime.progress import make_progress


def decode_ahead(read_chunk, chunk_specs, depth: int = None):
    """Iterate ``(spec, read_chunk(spec))`` with the next ``depth`` chunks'
    decodes running on worker threads while the caller runs the current
    chunk on the GPU — intra-video decode/compute overlap for extractors
    that stream a video in batches (ResNet/RAFT/PWC; the per-video
    ``prepare`` hook covers the whole-video-decode families)."""
    import os
    from concurrent.futures import ThreadPoolExecutor
    if depth is None:
        depth = max(1, int(os.environ.get('VFA_DECODE_AHEAD', '2')))
    chunk_specs = list(chunk_specs)
    if len(chunk_specs) <= 1:
        for spec in chunk_specs:
            yield spec, read_chunk(spec)
        return
    with ThreadPoolExecutor(max_workers=depth,
                            thread_name_prefix='vfa-chunk-decode') as pool:
        futs = {p: pool.submit(read_chunk, chunk_specs[p])
                for p in range(min(depth, len(chunk_specs)))}
        for i, spec in enumerate(chunk_specs):
            cur = futs.pop(i).result()
            nxt = i + depth
            if nxt < len(chunk_specs):
                futs[nxt] = pool.submit(read_chunk, chunk_specs[nxt])
            yield spec, cur


class BaseExtractor(torch.nn.Module):
    feature_type: str = ''

    prof = None

    def __init__(self, args: Any, external_call: bool = False):
        super().__init__()
        cfg = Config.coerce(args)
        self.cfg = cfg
        self.feature_type = cfg.feature_type
        self.path_list = form_list_from_user_input(cfg)
        self.extraction_fps = cfg.extraction_fps
        self.extract_method = cfg.extract_method
        self.on_extraction = cfg.on_extraction
        self.external_call = external_call
        self.output_direct = cfg.output_direct
        self.output_path = make_output_path(cfg.output_path, self.feature_type,
                                            cfg.output_direct)
        self.tmp_path = cfg.tmp_path
        self.keep_tmp_files = cfg.keep_tmp_files
        self.show_pred = cfg.show_pred
        self.progress = make_progress(total=len(self.path_list))
        self._models_cache: Dict[str, Any] = {}

    # -------------------------------------------------------------- hooks
    def build_models(self, device: torch.device, dtype: torch.dtype) -> Any:
        """Construct (and move to device) everything ``extract`` needs."""
        raise NotImplementedError

    def extract(self, device: torch.device, models: Any,
                video_path, prepared: Any = None) -> Dict[str, np.ndarray]:
        raise NotImplementedError

    # optional hook: CPU-side decode/sampling for ``video_path``, run one
    # video AHEAD on a worker thread so host decode overlaps GPU compute
    # (the decode pipeline of SURVEY §7 hard-part (d)).  Extractors that
    # override it receive the result as ``prepared``.
    prepare = None

    @staticmethod
    def load_weights(model: torch.nn.Module, path: str) -> None:
        """Load a state-dict file or URL (http/https/file — downloaded
        once into the weights cache, utils/weights.py), accepting the
        published reference checkpoint schemes (OpenAI CLIP / reference
        i3d / torchvision r2plus1d / torchvggish / sniklaus PWC /
        DataParallel 'module.' prefixes) via convert_auto."""
        from ..utils.convert_checkpoints import convert_auto
        from ..utils.weights import resolve_weights_path
        sd = torch.load(resolve_weights_path(path), map_location='cpu',
                        weights_only=True)
        if isinstance(sd, dict) and 'state_dict' in sd \
                and isinstance(sd['state_dict'], dict):
            sd = sd['state_dict']
        model.load_state_dict(convert_auto(sd))

    def _prof(self, stage: str):
        from ..runtime.profiler import StageProfiler
        if self.prof is None:
            self.prof = StageProfiler(False)
        return self.prof(stage)

    # ------------------------------------------------------------ helpers
    def compute_dtype(self, device: torch.device) -> torch.dtype:
        d = self.cfg.dtype
        if d == 'fp32':
            return torch.float32
        if d == 'bf16':
            return torch.bfloat16
        return torch.bfloat16 if device.type == 'cuda' else torch.float32

    def models_for(self, device: torch.device) -> Any:
        key = str(device)
        if key not in self._models_cache:
            dtype = self.compute_dtype(device)
            torch.manual_seed(self.cfg.seed)
            built = self.build_models(device, dtype)
            self._fold_bn_tree(built)
            self._models_cache[key] = built
        return self._models_cache[key]

    @staticmethod
    def _fold_bn_tree(obj: Any) -> None:
        """Extraction is inference-only: fold BatchNorm affine maps into the
        preceding convs of every built model (see utils/fold_bn.py)."""
        from ..utils.fold_bn import fold_batchnorms
        if isinstance(obj, torch.nn.Module):
            if not obj.training:
                fold_batchnorms(obj)
        elif isinstance(obj, dict):
            for v in obj.values():
                BaseExtractor._fold_bn_tree(v)
        elif isinstance(obj, (list, tuple)):
            for v in obj:
                BaseExtractor._fold_bn_tree(v)

    # ------------------------------------------------------------ forward
    @torch.no_grad()
    def forward(self, indices: torch.LongTensor) -> List[Dict[str, np.ndarray]]:
        device = indices.device
        from ..runtime.profiler import StageProfiler
        self.prof = StageProfiler(bool(getattr(self.cfg, 'profile', False)),
                                  device)
        models = self.models_for(device)
        feats_list: List[Dict[str, np.ndarray]] = []
        idx_list = indices.tolist()
        # multi-process sharding: this worker only sees its own indices, so
        # the bar total is the shard size, not the whole video list
        if getattr(self.progress, 'total', None) not in (None, len(idx_list)):
            try:
                self.progress.total = len(idx_list)
                self.progress.refresh()
            except Exception:
                pass
        pipeline = type(self).prepare is not None and len(idx_list) > 1
        pool = None
        futures = {}                   # position -> decode future
        if pipeline:
            import os as _os
            from concurrent.futures import ThreadPoolExecutor
            # decode-ahead depth (VFA_DECODE_AHEAD, default 2): the
            # extractor path is file-decode-bound, so overlapping more
            # than one video's decode with GPU compute pays; memory grows
            # by one decoded video per extra worker
            depth = max(1, int(_os.environ.get('VFA_DECODE_AHEAD', '2')))
            pool = ThreadPoolExecutor(max_workers=depth,
                                      thread_name_prefix='vfa-decode')

            def fill(from_pos):
                for p in range(from_pos,
                               min(from_pos + depth, len(idx_list))):
                    if p not in futures:
                        futures[p] = pool.submit(
                            self.prepare, self.path_list[idx_list[p]])
        for pos, idx in enumerate(idx_list):
            video_path = self.path_list[idx]
            # the future at THIS position is always for THIS video: take
            # ownership first, so a resume-skip or a raise can never hand
            # video i's decode to video i+1
            pending = futures.pop(pos, None)
            try:
                if (self.cfg.resume and not self.external_call
                        and self._already_done(video_path)):
                    if pending is not None:
                        # consume + discard this video's prefetched decode;
                        # let the extractor release any side artifacts
                        # (e.g. vggish tmp wavs)
                        try:
                            stale = pending.result()
                        except Exception:
                            stale = None
                        pending = None
                        hook = getattr(self, 'discard_prepared', None)
                        if stale is not None and hook is not None:
                            hook(stale)
                    self.progress.update()
                    continue
                prepared = None
                if pipeline:
                    fill(pos + 1)      # keep the decode pipeline full
                    with self._prof('decode'):
                        prepared = (pending.result() if pending is not None
                                    else self.prepare(video_path))
                feats_dict = self.extract(device, models, video_path,
                                          prepared)
                if self.external_call:
                    feats_list.append(feats_dict)
                else:
                    with self.prof('sink'):
                        action_on_extraction(
                            feats_dict, self._stem_path(video_path),
                            self.output_path, self.on_extraction,
                            self.output_direct, self.feature_type)
            except KeyboardInterrupt:
                raise
            except Exception as e:
                print(e)
                print(f'Extraction failed at: {video_path} with error (^). '
                      'Continuing extraction')
                traceback.print_exc()
            finally:
                self._cleanup_tmp(video_path)
            self.progress.update()
        if pool is not None:
            pool.shutdown(wait=False, cancel_futures=True)
        rep = self.prof.report(f'{self.feature_type} shard on {device}')
        if rep:
            print(rep)
        return feats_list

    def _stem_path(self, video_path) -> str:
        # pre-computed-flow inputs are (video, flow_dir) tuples
        return video_path[0] if isinstance(video_path, tuple) else video_path

    def _cleanup_tmp(self, video_path) -> None:
        """Remove this video's transcode artifact (tmp_path/{stem}.y4m,
        written by io.ffmpeg.decode_to_y4m for non-native codecs) unless
        ``--keep_tmp_files`` (reference extract_resnet.py:159-160
        semantics; audio tmp wavs are cleaned by the vggish extractor
        itself)."""
        if self.keep_tmp_files:
            return
        import os
        from pathlib import Path
        src = str(self._stem_path(video_path))
        stem = Path(src).stem
        p = os.path.join(self.tmp_path, stem + '.y4m')
        if os.path.abspath(p) == os.path.abspath(src):
            return          # the SOURCE lives in tmp_path — never delete it
        if os.path.exists(p):
            try:
                os.remove(p)
            except OSError:
                pass

    def _already_done(self, video_path) -> bool:
        keys = getattr(self, 'output_feat_keys', [self.feature_type])
        return outputs_exist(keys, self._stem_path(video_path), self.output_path,
                             self.on_extraction, self.output_direct,
                             self.feature_type)
