import sys, torch
sys.path.insert(0, "/root/repo")
from torch.utils.cpp_extension import load
m = load(name="trprobe", sources=["/root/repo/tools/tr_probe_all.hip"],
         extra_cuda_cflags=["--offload-arch=gfx950", "-O2"], verbose=False)
u, l = m.run()
u = u.reshape(64, 4).int().cpu()
l = l.reshape(64, 4).int().cpu()
print("uniform-addr:")
for i in [0,1,2,3,4,8,15,16,17,31,32,48]: print(i, u[i].tolist())
print("per-lane-addr (base + lane*8B):")
for i in [0,1,2,3,4,8,15,16,17,31,32,48]: print(i, l[i].tolist())
