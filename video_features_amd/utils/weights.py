"""Checkpoint acquisition: URL-aware weight loading with a local cache.

The reference auto-downloads its checkpoints (torch hub for vggish_torch,
reference models/vggish_torch/extract_vggish.py:22-27; the `clip` package
for CLIP, models/CLIP/extract_clip.py:46-47).  Here the capability is
unified: ``--weights_path`` accepts a filesystem path OR a URL
(http/https/file); URLs are downloaded once into
``$VFA_WEIGHTS_CACHE`` (default ``~/.cache/video_features_amd``) and
re-used — the converters in ``convert_checkpoints`` then accept the
published key schemes as usual.
"""
from __future__ import annotations

import hashlib
import os
import urllib.parse
import urllib.request


def is_url(path: str) -> bool:
    scheme = urllib.parse.urlparse(str(path)).scheme
    return scheme in ('http', 'https', 'file')


def cache_dir() -> str:
    return os.environ.get(
        'VFA_WEIGHTS_CACHE',
        os.path.join(os.path.expanduser('~'), '.cache',
                     'video_features_amd'))


def fetch_weights(url: str) -> str:
    """Return a local path for ``url``, downloading into the cache on the
    first use.  The cache key includes a URL hash so same-named files from
    different sources do not collide."""
    name = os.path.basename(urllib.parse.urlparse(url).path) or 'weights'
    digest = hashlib.sha256(url.encode()).hexdigest()[:12]
    target = os.path.join(cache_dir(), f'{digest}_{name}')
    if os.path.exists(target):
        return target
    os.makedirs(cache_dir(), exist_ok=True)
    tmp = target + '.part'
    urllib.request.urlretrieve(url, tmp)
    os.replace(tmp, target)
    return target


def resolve_weights_path(path: str) -> str:
    """Filesystem path -> itself; URL -> cached local file."""
    return fetch_weights(path) if is_url(path) else path
