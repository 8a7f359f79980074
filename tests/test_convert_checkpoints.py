"""Checkpoint converters: build a reference-scheme state dict from our own
models (renamed to the published conventions), convert back, and load —
round-trip must be exact."""
import re

import pytest
import torch

from video_features_amd.utils import convert_checkpoints as cc


def test_i3d_reference_scheme_roundtrip():
    from video_features_amd.models.i3d import I3D
    torch.manual_seed(0)
    m = I3D(modality='rgb')
    sd = m.state_dict()
    legacy = {}
    for k, v in sd.items():
        k = k.replace('.conv.', '.conv3d.').replace('.bn.', '.batch3d.')
        for i in range(4):
            k = k.replace(f'.b{i}.', f'.branch_{i}.')
        legacy[k] = v.clone()
    back = cc.convert_auto(legacy)
    m2 = I3D(modality='rgb')
    m2.load_state_dict(back)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_clip_visual_scheme_roundtrip():
    from video_features_amd.models.clip_vit import VisionTransformer
    torch.manual_seed(0)
    m = VisionTransformer()
    sd = m.state_dict()
    legacy = {}
    for k, v in sd.items():
        k = 'visual.' + k.replace('blocks.', 'transformer.resblocks.')
        k = k.replace('.attn.qkv.weight', '.attn.in_proj_weight')
        k = k.replace('.attn.qkv.bias', '.attn.in_proj_bias')
        k = k.replace('.attn.proj.', '.attn.out_proj.')
        legacy[k] = v.clone()
    # text-tower keys that must be dropped
    legacy['token_embedding.weight'] = torch.zeros(1)
    legacy['ln_final.weight'] = torch.zeros(1)
    legacy['logit_scale'] = torch.zeros(())
    back = cc.convert_auto(legacy)
    m2 = VisionTransformer()
    m2.load_state_dict(back)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_clip_full_model_dict_with_text_resblocks():
    """A FULL OpenAI CLIP dict carries text-tower 'transformer.resblocks.*'
    keys at top level; they must not collide with the visual tower's renamed
    blocks (reference loads full model dicts, extract_clip.py:46)."""
    from video_features_amd.models.clip_vit import VisionTransformer
    torch.manual_seed(0)
    m = VisionTransformer()
    sd = m.state_dict()
    legacy = {}
    for k, v in sd.items():
        k = 'visual.' + k.replace('blocks.', 'transformer.resblocks.')
        k = k.replace('.attn.qkv.weight', '.attn.in_proj_weight')
        k = k.replace('.attn.qkv.bias', '.attn.in_proj_bias')
        k = k.replace('.attn.proj.', '.attn.out_proj.')
        legacy[k] = v.clone()
    # text tower of a full model: resblocks with DIFFERENT (wrong if loaded)
    # shapes, plus the other text keys
    for i in range(12):
        legacy[f'transformer.resblocks.{i}.attn.in_proj_weight'] = \
            torch.zeros(3 * 512, 512)
        legacy[f'transformer.resblocks.{i}.mlp.c_fc.weight'] = \
            torch.zeros(2048, 512)
    legacy['positional_embedding'] = torch.zeros(77, 512)
    legacy['token_embedding.weight'] = torch.zeros(49408, 512)
    legacy['text_projection'] = torch.zeros(512, 512)
    legacy['ln_final.weight'] = torch.zeros(512)
    legacy['logit_scale'] = torch.zeros(())
    back = cc.convert_auto(legacy)
    m2 = VisionTransformer()
    m2.load_state_dict(back)  # strict: text keys must all be gone
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_clip_bare_visual_tower_dict():
    """A visual-tower-only dict (no 'visual.' prefix) keeps its
    positional_embedding — it belongs to the image encoder there."""
    from video_features_amd.models.clip_vit import VisionTransformer
    torch.manual_seed(0)
    m = VisionTransformer()
    sd = m.state_dict()
    legacy = {}
    for k, v in sd.items():
        k = k.replace('blocks.', 'transformer.resblocks.')
        k = k.replace('.attn.qkv.weight', '.attn.in_proj_weight')
        k = k.replace('.attn.qkv.bias', '.attn.in_proj_bias')
        k = k.replace('.attn.proj.', '.attn.out_proj.')
        legacy[k] = v.clone()
    back = cc.convert_clip_visual(legacy)
    m2 = VisionTransformer()
    m2.load_state_dict(back)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_r21d_torchvision_scheme_roundtrip():
    from video_features_amd.models.r21d import R2Plus1D18
    torch.manual_seed(0)
    m = R2Plus1D18()
    sd = m.state_dict()
    legacy = {}
    for k, v in sd.items():
        mm = re.match(r'(layer\d\.\d\.)(.*)', k)
        if mm:
            pre, rest = mm.groups()
            rest = re.sub(r'conv(\d)\.spatial\.', r'conv\1.0.0.', rest)
            rest = re.sub(r'conv(\d)\.bn\.', r'conv\1.0.1.', rest)
            rest = re.sub(r'conv(\d)\.temporal\.', r'conv\1.0.3.', rest)
            rest = re.sub(r'^bn(\d)\.', r'conv\1.1.', rest)
            k = pre + rest
        legacy[k] = v.clone()
    back = cc.convert_auto(legacy)
    m2 = R2Plus1D18()
    m2.load_state_dict(back)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_vggish_scheme_roundtrip():
    from video_features_amd.models.vggish import VGGish
    torch.manual_seed(0)
    m = VGGish()
    sd = m.state_dict()
    legacy = {k.removeprefix('net.'): v.clone() for k, v in sd.items()}
    back = cc.convert_auto(legacy)
    m2 = VGGish()
    m2.load_state_dict(back, strict=False)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_weights_url_fetch_and_load(tmp_path, monkeypatch):
    """--weights_path accepts a URL: fetched once into the cache, then
    loaded through the usual converter path (file:// here; http in prod)."""
    from video_features_amd.models.clip_vit import VisionTransformer
    from video_features_amd.utils import weights as W
    from video_features_amd.extractors.base import BaseExtractor
    torch.manual_seed(0)
    m = VisionTransformer()
    ck = tmp_path / 'visual.pt'
    torch.save(m.state_dict(), str(ck))
    monkeypatch.setenv('VFA_WEIGHTS_CACHE', str(tmp_path / 'cache'))
    url = ck.as_uri()
    assert W.is_url(url) and not W.is_url(str(ck))
    m2 = VisionTransformer()
    BaseExtractor.load_weights(m2, url)
    for k, v in m2.state_dict().items():
        assert torch.equal(v, m.state_dict()[k]), k
    # second resolve hits the cache (delete the source to prove it)
    cached = W.resolve_weights_path(url)
    ck.unlink()
    assert W.resolve_weights_path(url) == cached
