import os, sys
keys = [k for k in os.environ if any(s in k.upper() for s in
        ("VISIBLE", "HIP", "ROCR", "CUDA", "RANK", "WORLD", "MASTER", "OMP"))]
print("RANK", os.environ.get("RANK"), {k: os.environ[k] for k in sorted(keys)}, flush=True)
import torch
print("RANK", os.environ.get("RANK"), "count:", torch.cuda.device_count(), flush=True)
try:
    torch.cuda.init()
    print("RANK", os.environ.get("RANK"), "init ok", flush=True)
except Exception as e:
    print("RANK", os.environ.get("RANK"), "init fail:", e, flush=True)
