import torch, time, sys
sys.path.insert(0, '.')
from video_features_amd import ops
dev='cuda:0'
def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6
m,n,k = 1204224,256,64
x=(torch.randn(m,k,device=dev)/2).to(torch.bfloat16)
w=(torch.randn(n,k,device=dev)/2).to(torch.bfloat16)
b=torch.randn(n,device=dev).to(torch.bfloat16)
r=torch.randn(m,n,device=dev).to(torch.bfloat16)
print('no-res :', t(lambda: ops.linear_act(x,w,b,'relu')), 'us')
print('res    :', t(lambda: ops.linear_act(x,w,b,'relu',r)), 'us')
# graph-captured variant
g = torch.cuda.CUDAGraph()
s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    ops.linear_act(x,w,b,'relu',r)
torch.cuda.current_stream().wait_stream(s)
with torch.cuda.graph(g):
    ops.linear_act(x,w,b,'relu',r)
print('graph  :', t(lambda: g.replay()), 'us')
# conv1x1_act route (as the model calls it)
xc = x.reshape(768,56,56,64).permute(0,3,1,2)
rc = r.reshape(768,56,56,256).permute(0,3,1,2)
print('conv1x1:', t(lambda: ops.conv1x1_act(xc, w.reshape(n,k,1,1), b, 'relu', rc)), 'us')
