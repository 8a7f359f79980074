"""HTTP feature-extraction service (FastAPI).

The reference is batch-only; this serves the same extractors behind an
endpoint for production use: models load once per process and stay resident
on the GPU, requests stream video bytes in and get an ``.npz`` of features
back (the same arrays ``--on_extraction save_numpy`` would write).

Run:  python -m video_features_amd.serve --feature_type CLIP-ViT-B/32 \
          [--port 8080] [--cpu]
Then: curl --data-binary @clip.mp4 \
          'localhost:8080/extract?filename=clip.mp4' -o feats.npz

Scaling note: run one server process per GPU (HIP_VISIBLE_DEVICES=K) behind
any HTTP load balancer — extraction is embarrassingly parallel across
videos, so no cross-process state is needed.
"""
import asyncio
import io
import os
import tempfile
from typing import Optional

import numpy as np
import torch

from .config import Config, sanity_check
from .models.registry import get_extractor_class


def create_app(cfg: Config):
    from fastapi import FastAPI, HTTPException, Request
    from fastapi.responses import Response

    cfg = Config.coerce(cfg).replace(video_paths=['__server__'])
    device = torch.device('cpu' if cfg.cpu or not torch.cuda.is_available()
                          else 'cuda:0')
    app = FastAPI(title='video-features-mi355x',
                  description=f'{cfg.feature_type} feature extraction')
    # the extractor is shared mutable state (ex.path_list is swapped per
    # request): the lock serializes extractions, and run_in_executor keeps
    # the blocking GPU work OFF the event loop so /health stays responsive
    state = {'extractor': None, 'lock': asyncio.Lock()}

    def extractor(path: str):
        # built on first request (the input lister validates paths exist,
        # so construction needs a real file); models stay resident after
        if state['extractor'] is None:
            ex = get_extractor_class(cfg.feature_type)(
                cfg.replace(video_paths=[path]), external_call=True)
            ex.models_for(device)      # build + fold once, stay resident
            state['extractor'] = ex
        ex = state['extractor']
        ex.path_list = [path]
        return ex

    @app.get('/health')
    def health():
        return {'status': 'ok', 'feature_type': cfg.feature_type,
                'device': str(device),
                'hip_kernels': __import__(
                    'video_features_amd.ops', fromlist=['ops']
                ).hip_available()}

    @app.post('/extract')
    async def extract(request: Request, filename: str = 'video.mp4'):
        # raw-body upload (no multipart dependency):
        #   curl --data-binary @clip.mp4 '/extract?filename=clip.mp4'
        suffix = os.path.splitext(filename)[1] or '.mp4'
        data = await request.body()
        if not data:
            raise HTTPException(400, 'empty request body')
        with tempfile.NamedTemporaryFile(suffix=suffix, delete=False) as f:
            f.write(data)
            path = f.name
        try:
            def run():
                ex = extractor(path)
                return ex(torch.zeros(1, dtype=torch.long, device=device))
            async with state['lock']:
                out = await asyncio.get_event_loop().run_in_executor(
                    None, run)
            if not out:
                raise HTTPException(422, 'extraction failed for this file')
            buf = io.BytesIO()
            np.savez(buf, **{k.replace('/', '_'): np.asarray(v)
                             for k, v in out[0].items()})
            return Response(content=buf.getvalue(),
                            media_type='application/octet-stream',
                            headers={'Content-Disposition':
                                     'attachment; filename=features.npz'})
        finally:
            os.unlink(path)

    return app


def main(argv: Optional[list] = None):
    import argparse

    import uvicorn

    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument('--feature_type', required=True)
    p.add_argument('--cpu', action='store_true')
    p.add_argument('--host', default='127.0.0.1')
    p.add_argument('--port', type=int, default=8080)
    p.add_argument('--extract_method', default=None)
    p.add_argument('--flow_type', default='pwc')
    p.add_argument('--batch_size', type=int, default=16)
    p.add_argument('--weights_path', default=None)
    args = p.parse_args(argv)
    cfg = Config(feature_type=args.feature_type, cpu=args.cpu,
                 extract_method=args.extract_method,
                 flow_type=args.flow_type, batch_size=args.batch_size,
                 weights_path=args.weights_path,
                 video_paths=['__server__'])
    sanity_check(cfg.replace(video_paths=None))
    uvicorn.run(create_app(cfg), host=args.host, port=args.port)


if __name__ == '__main__':
    main()
