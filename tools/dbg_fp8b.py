import sys, os, torch
sys.path.insert(0, "/root/repo")
combos = [
    ("all_on", {}),
    ("no_convt", {"GDLJ_FP8_CONVT": "0"}),
    ("no_delayed", {"GDLJ_FP8_DELAYED": "0"}),
    ("no_bwd", {"GDLJ_FP8_BWD": "0"}),
    ("no_convt_no_delayed_no_bwd", {"GDLJ_FP8_CONVT": "0",
                                    "GDLJ_FP8_DELAYED": "0",
                                    "GDLJ_FP8_BWD": "0"}),
]
for label, env in combos:
    for k in ("GDLJ_FP8_CONVT", "GDLJ_FP8_DELAYED", "GDLJ_FP8_BWD"):
        os.environ.pop(k, None)
    os.environ.update(env)
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.ops import gpu_ops
    from gan_deeplearning4j_amd.train import GanTrainer
    cfg = preset("dcgan128")
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                    dtype=torch.bfloat16, capture=False)
    torch.manual_seed(0)
    real = (torch.rand(16, 3, 128, 128, device="cuda:0",
                       dtype=torch.bfloat16) * 2 - 1)
    outs = []
    for i in range(2):
        out = tr.step(real)
        torch.cuda.synchronize()
        outs.append((round(float(out["loss_d"]), 4),
                     round(float(out["loss_g"]), 4)))
    print(label, outs, flush=True)
    gpu_ops.set_fp8_conv(False)
