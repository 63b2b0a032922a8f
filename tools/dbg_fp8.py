import sys, math, torch
sys.path.insert(0, "/root/repo")
from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.ops import gpu_ops
from gan_deeplearning4j_amd.train import GanTrainer

for label, env in (("bwd_on", None), ("bwd_off", "0")):
    import os
    if env is None:
        os.environ.pop("GDLJ_FP8_BWD", None)
    else:
        os.environ["GDLJ_FP8_BWD"] = env
    cfg = preset("dcgan128")
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                    dtype=torch.bfloat16, capture=False)
    torch.manual_seed(0)
    real = (torch.rand(16, 3, 128, 128, device="cuda:0",
                       dtype=torch.bfloat16) * 2 - 1)
    for i in range(3):
        out = tr.step(real)
        torch.cuda.synchronize()
        print(label, i, {k: float(v) for k, v in out.items()})
    # scan params for nan
    bad = [n for n, p in list(tr.dis.named_parameters()) +
           list(tr.gen.named_parameters())
           if not torch.isfinite(p.float()).all()]
    print(label, "nonfinite params:", bad[:5])
    gpu_ops.set_fp8_conv(False)
