"""Debug: print every F.conv2d call during one ResNet-50 bench-shaped step."""
import os, sys; sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch, torch.nn.functional as F
from video_features_amd.models.resnet import build_resnet
from video_features_amd.utils.fold_bn import fold_batchnorms

orig = F.conv2d
def spy(x, w, b=None, stride=1, padding=0, dilation=1, groups=1):
    print('F.conv2d:', tuple(x.shape), tuple(w.shape), 'stride', stride,
          'pad', padding, 'CL', x.is_contiguous(memory_format=torch.channels_last))
    return orig(x, w, b, stride, padding, dilation, groups)
F.conv2d = spy
torch.nn.functional.conv2d = spy

m = build_resnet('resnet50').eval().to('cuda:0', torch.bfloat16)
fold_batchnorms(m)
m = m.to(memory_format=torch.channels_last)
x = torch.randn(16, 3, 224, 224, device='cuda:0').to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
with torch.no_grad():
    m.forward_features(x)
torch.cuda.synchronize()
print('done')
