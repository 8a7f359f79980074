"""HTTP serving layer: upload a video, get the feature .npz back."""
import io

import numpy as np
import pytest
import torch

fastapi = pytest.importorskip('fastapi')
from fastapi.testclient import TestClient

from tests.conftest import synthetic_frames
from video_features_amd.config import Config
from video_features_amd.io.y4m import write_y4m
from video_features_amd.serve import create_app


def test_serve_extract_roundtrip(tmp_path):
    cfg = Config(feature_type='CLIP-ViT-B/32', cpu=True,
                 extract_method='uni_4', video_paths=['__server__'],
                 tmp_path=str(tmp_path / 'tmp'))
    app = create_app(cfg)
    client = TestClient(app)

    r = client.get('/health')
    assert r.status_code == 200 and r.json()['status'] == 'ok'

    vid = tmp_path / 'v.y4m'
    write_y4m(str(vid), synthetic_frames(t=12, h=64, w=64), fps=25.0)
    r = client.post('/extract?filename=v.y4m', content=vid.read_bytes())
    assert r.status_code == 200, r.text
    npz = np.load(io.BytesIO(r.content))
    assert npz['CLIP-ViT-B_32'].shape == (4, 512)
    assert float(npz['fps']) == 25.0

    # second request reuses the resident models
    r2 = client.post('/extract?filename=v.y4m', content=vid.read_bytes())
    assert r2.status_code == 200


def test_serve_error_paths(tmp_path):
    cfg = Config(feature_type='CLIP-ViT-B/32', cpu=True,
                 extract_method='uni_2', video_paths=['__server__'],
                 tmp_path=str(tmp_path / 'tmp'))
    client = TestClient(create_app(cfg))
    r = client.post('/extract?filename=v.y4m', content=b'')
    assert r.status_code == 400
    r = client.post('/extract?filename=v.y4m', content=b'garbage' * 100)
    assert r.status_code == 422
