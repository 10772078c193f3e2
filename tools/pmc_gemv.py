import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lfmq.ops import hip_ext
DEV="cuda:0"
ext = hip_ext()
def mk(n,k,g=128,bits=4):
    per=32//bits; G=k//g
    qw = torch.randint(-2**31,2**31-1,(n,k//per),dtype=torch.int32,device=DEV)
    sc = (torch.rand(n,G,device=DEV)*0.01).to(torch.bfloat16)
    qz = torch.randint(-2**31,2**31-1,(n,(G+per-1)//per),dtype=torch.int32,device=DEV)
    return qw,sc,qz
x4 = torch.randn(1,4096,device=DEV).to(torch.bfloat16)
x14 = torch.randn(1,14336,device=DEV).to(torch.bfloat16)
qw_gu, sc_gu, qz_gu = mk(28672,4096)
qw_dn, sc_dn, qz_dn = mk(4096,14336)
for _ in range(20):
    ext.dq_gemv_gateup(x4, qw_gu, sc_gu, qz_gu, 4, 128, -1.0, 4)
    ext.dq_linear(x14, qw_dn, sc_dn, qz_dn, 4, 128, None, None, -1.0, 2)
torch.cuda.synchronize()
