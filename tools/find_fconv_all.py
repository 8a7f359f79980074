"""Debug: spy every F.conv2d/conv3d/linear during forwards of all families."""
import os, sys; sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch, torch.nn.functional as F
from video_features_amd.utils.fold_bn import fold_batchnorms

leaks = []
for name in ('conv2d', 'conv3d'):
    orig = getattr(F, name)
    def spy(x, w, *a, _n=name, _o=orig, **k):
        leaks.append((_n, tuple(x.shape), tuple(w.shape)))
        return _o(x, w, *a, **k)
    setattr(F, name, spy)
    setattr(torch.nn.functional, name, spy)

dev, dt = 'cuda:0', torch.bfloat16

def run(tag, fn):
    leaks.clear()
    with torch.no_grad():
        fn()
    torch.cuda.synchronize()
    print(tag, 'leaks:', len(leaks))
    for l in leaks[:40]:
        print('  ', l)

# PWC at bench shape
from video_features_amd.models.pwc import PWCNet
m = PWCNet().eval().to(dev, dt)
fold_batchnorms(m)
m = m.to(memory_format=torch.channels_last)
a = torch.rand(8, 3, 256, 256, device=dev).to(dt)
run('pwc', lambda: m(a, a))

# VGGish
from video_features_amd.models.vggish import VGGishNet
v = VGGishNet().eval().to(dev, dt); fold_batchnorms(v)
v = v.to(memory_format=torch.channels_last)
xe = torch.randn(32, 96, 64, device=dev).to(dt)
run("vggish", lambda: v(xe))

# R21D-34
from video_features_amd.models.r21d import R2Plus1D34
r = R2Plus1D34().eval().to(dev, dt); fold_batchnorms(r)
xc = torch.rand(8, 3, 16, 112, 112, device=dev).to(dt)
run('r21d34', lambda: r.forward_features(xc))

# I3D rgb
from video_features_amd.models.i3d import I3D
i3 = I3D(modality='rgb').eval().to(dev, dt); fold_batchnorms(i3)
xi = torch.rand(2, 3, 64, 224, 224, device=dev).to(dt)
run('i3d', lambda: i3(xi))

# RAFT — feed channels_last as the extractor/bench do (use_channels_last
# flips the module; an NCHW input here reports every conv as a false leak)
from video_features_amd.models.raft import RAFT
rf = RAFT().eval().to(dev, dt); fold_batchnorms(rf)
if hasattr(rf, 'use_channels_last'):
    rf = rf.use_channels_last()
f1 = (torch.rand(8, 3, 224, 224, device=dev).to(dt) * 255).contiguous(
    memory_format=torch.channels_last)
run('raft', lambda: rf(f1, f1, iters=3))
print('done')
