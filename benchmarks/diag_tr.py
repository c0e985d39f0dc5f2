import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tepdist_amd.ops import _tepdist_hip as ext
from tepdist_amd.ops import hip

pat = torch.zeros(256, device="cuda"); uni = torch.zeros(256, device="cuda")
ext.tr16_probe(pat.data_ptr(), uni.data_ptr(), torch.cuda.current_stream().cuda_stream)
torch.cuda.synchronize()
print("PATTERN (lane: 4 elems), addr_l=(l&15)+(l>>4)*64 elems:")
for l in range(0, 64):
    print(l, [int(x) for x in pat.reshape(64,4)[l].tolist()])
print("UNIFORM (all lanes addr=0):")
for l in range(0, 64):
    print(l, [int(x) for x in uni.reshape(64,4)[l].tolist()])

# identity GEMM k-trace: A=I (32x32), B[k][n]=k*100+n scaled small
K, N = 32, 32
A = torch.eye(K, dtype=torch.bfloat16, device="cuda")       # [M=32,K] kc
Bm = (torch.arange(K).reshape(K,1)*0.125 + torch.arange(N).reshape(1,N)*0.001).to(torch.bfloat16).cuda()  # [K,N] KO
C = hip.matmul(A, Bm)   # should equal Bm
torch.cuda.synchronize()
ok = torch.allclose(C.float(), Bm.float(), atol=1e-2)
print("identity NN test pass:", ok)
if not ok:
    # print mapping: C[m][0] tells which k-row landed at m
    print("C[:,0]/0.125 (which k row ended at m):", [round(x/0.125) for x in C[:,0].float().tolist()])
    print("C[0,:] (row 0):", [round(v,3) for v in C[0,:8].float().tolist()])
