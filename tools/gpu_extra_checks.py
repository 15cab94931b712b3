"""One-off GPU checks: large-batch HBM sizing + Sintel volume step."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def big_batch(batch=512):
    from deepof_amd.models import build_model
    from deepof_amd.losses import MultiScaleUnsupLoss, preprocess_images
    from deepof_amd.engine.optim import FusedAdam
    dev = "cuda:0"
    torch.backends.cudnn.benchmark = True
    model, scales, w = build_model("flownets")
    model.to(dev).to(memory_format=torch.channels_last)
    loss_fn = MultiScaleUnsupLoss(scales, w)
    opt = FusedAdam(model.parameters(), lr=1e-5)
    img1 = torch.rand(batch, 3, 384, 512, device=dev) * 255
    img2 = torch.rand(batch, 3, 384, 512, device=dev) * 255
    x = torch.cat([preprocess_images(img1, loss_fn.mean_bgr),
                   preprocess_images(img2, loss_fn.mean_bgr)], 1
                  ).to(memory_format=torch.channels_last)
    for i in range(4):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            flows = model(x)
        res = loss_fn(flows, img1, img2)
        res["total"].backward()
        opt.step(); opt.zero_grad(set_to_none=False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(6):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            flows = model(x)
        res = loss_fn(flows, img1, img2)
        res["total"].backward()
        opt.step(); opt.zero_grad(set_to_none=False)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 6
    mem = torch.cuda.max_memory_allocated() / 2**30
    print(f"BIGBATCH batch={batch}: {batch/dt:.0f} imgs/sec, "
          f"{dt*1000:.1f} ms/step, peak HBM {mem:.1f} GiB")

def volume_step():
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer
    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(256, 512), batch_size=4,
        num_workers=0, model="inception_v3", precision="bf16",
        device="cuda", channels_last=True, time_step=10,
        log_dir="/tmp/l", run_name="v", log_interval=1))
    tr = Trainer(cfg)
    vol = torch.rand(4, 30, 256, 512) * 255
    t0 = time.perf_counter()
    parts = tr.train_step({"volume": vol})
    torch.cuda.synchronize()
    print(f"VOLUME T=10 step ok: total={parts['total']:.3f} "
          f"({time.perf_counter()-t0:.1f}s incl. autotune)")

if __name__ == "__main__":
    if sys.argv[1] == "big":
        big_batch(int(sys.argv[2]) if len(sys.argv) > 2 else 512)
    else:
        volume_step()
