"""Dispatch-level copy spy: python source lines of big aten copies."""
import collections
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.utils._python_dispatch import TorchDispatchMode

counter = collections.Counter()
bytes_c = collections.Counter()


class CopySpy(TorchDispatchMode):
    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if any(k in name for k in ("copy_", "_to_copy", "clone", "cat")):
            t = args[0]
            if isinstance(t, (list, tuple)):
                t = t[0]
            if hasattr(t, "numel"):
                nb = t.numel() * t.element_size()
                if nb > (4 << 20):
                    tb = traceback.extract_stack()
                    fr = [f for f in tb if "gan_deeplearning4j_amd" in
                          f.filename and "trace_copies" not in f.filename]
                    key = tuple(f"{f.filename.split('/')[-1]}:{f.lineno}"
                                for f in fr[-3:]) or ("<outside>",)
                    counter[key] += 1
                    bytes_c[key] += nb
        return func(*args, **(kwargs or {}))


from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer

cfg = preset("dcgan64")
gen, dis = build_dcgan(cfg)
tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                dtype=torch.bfloat16, capture=False)
real = (torch.rand(4096, 3, 64, 64, device="cuda:0", dtype=torch.bfloat16)
        * 2 - 1).contiguous(memory_format=torch.channels_last)
for _ in range(2):
    tr.step(real)
torch.cuda.synchronize()
with CopySpy():
    tr.step(real)
torch.cuda.synchronize()
for key, nb in bytes_c.most_common(14):
    print(f"{nb / 2**20:8.0f} MiB x{counter[key]:<3} {' <- '.join(key)}")
