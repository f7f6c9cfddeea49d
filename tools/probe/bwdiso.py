import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import torch, time, sys
from cilfw import _hip_ops as H
def t(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.time()-t0)/iters*1e6
for name,(B,hw,C,K,R,s) in {"l1":(128,32,64,64,3,1),"l2":(128,16,128,128,3,1),
                            "l3":(128,8,256,256,3,1),"l4":(128,4,512,512,3,1),
                            "l2s":(128,32,64,128,3,2)}.items():
    pad=R//2; Ho=(hw+2*pad-R)//s+1
    dy = torch.randn(B,Ho,Ho,K,device="cuda").bfloat16().contiguous()
    w = torch.randn(R,R,C,K,device="cuda").bfloat16().contiguous()
    us = t(lambda: H.conv2d_bwd_data(dy, w, s, pad, hw, hw))
    tf = 2.0*B*Ho*Ho*K*C*R*R/(us*1e6)
    print(f"{name}: {us:.1f}us {tf:.1f}TF")
