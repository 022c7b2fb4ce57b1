#!/usr/bin/env python3
"""A/B the GEMM variants at key shapes (run on a GPU box)."""
import os, sys, time, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from vantage6_amd.ops import hip_ops
ops = hip_ops()

def bench(fn, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

shapes = [(4096,4096,4096), (8192,8192,8192), (4096,3072,768), (4096,768,3072)]
for (m,n,k) in shapes:
    a = torch.randn(m,k,dtype=torch.bfloat16,device="cuda")
    b = torch.randn(n,k,dtype=torch.bfloat16,device="cuda")
    ms = bench(lambda: ops.gemm_bf16(a,b,False,True))
    tf = 2*m*n*k/ms/1e12
    ref = torch.matmul(a, b.t())
    c = ops.gemm_bf16(a,b,False,True)
    rel = (c.float()-ref.float()).norm().item()/max(ref.float().norm().item(),1e-9)
    print(json.dumps({"nt": [m,n,k], "ms": round(ms*1e3,3), "tflops": round(tf,1), "rel_vs_blas": round(rel,6)}))
# wgrad / dgrad forms
for (m,na,nb) in [(4096,768,3072),(4096,3072,768),(8192,1024,1024)]:
    a = torch.randn(m,na,dtype=torch.bfloat16,device="cuda")
    b = torch.randn(m,nb,dtype=torch.bfloat16,device="cuda")
    ms = bench(lambda: ops.gemm_bf16(a,b,True,False))
    tf = 2*m*na*nb/ms/1e12
    print(json.dumps({"tn": [m,na,nb], "ms": round(ms*1e3,3), "tflops": round(tf,1)}))
for (m,r,k) in [(4096,768,3072),(4096,3072,768)]:
    a = torch.randn(m,r,dtype=torch.bfloat16,device="cuda")
    b = torch.randn(r,k,dtype=torch.bfloat16,device="cuda")
    ms = bench(lambda: ops.gemm_bf16(a,b,False,False))
    tf = 2*m*r*k/ms/1e12
    print(json.dumps({"nn": [m,r,k], "ms": round(ms*1e3,3), "tflops": round(tf,1)}))
