import ctypes, subprocess, sys
import torch
subprocess.run(["hipcc", "--offload-arch=gfx950", "-O2", "--genco",
                "/root/repo/tools/tr_probe.hip", "-o", "/tmp/tr_probe.hsaco"], check=True)
from hip import hip  # may not exist; fallback to torch utils
