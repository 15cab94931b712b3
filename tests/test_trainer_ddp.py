"""2-process gloo end-to-end Trainer.fit (init_distributed + DDP +
sampler sharding + checkpointing under a real process group)."""

import os

import pytest
import torch.multiprocessing as mp


def _worker(rank, world, port, log_dir, ok):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        from deepof_amd.config import Config
        from deepof_amd.engine import Trainer

        cfg = Config.from_dict(dict(
            dataset="synthetic", image_size=(48, 64), batch_size=2,
            num_workers=0, model="flownets", precision="fp32",
            device="cpu", log_dir=log_dir, run_name="ddp2",
            log_interval=1,
        ))
        tr = Trainer(cfg)
        assert tr.world == 2
        tr.fit(max_steps=2)
        ok[rank] = tr.global_step
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_trainer_two_process_gloo(tmp_path):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        ok = mgr.dict()
        procs = [ctx.Process(target=_worker,
                             args=(r, 2, 29523, str(tmp_path), ok))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(240)
            assert p.exitcode == 0
        assert ok[0] == 2 and ok[1] == 2
    assert os.path.exists(os.path.join(str(tmp_path), "ddp2",
                                       "ckpt_last.pt"))
