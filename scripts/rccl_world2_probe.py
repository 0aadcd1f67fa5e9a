"""Probe: can RCCL host TWO ranks on ONE MI355X? (NCCL traditionally
refuses duplicate GPUs; if this build allows it, the engine's whole
distributed surface can be RCCL-validated on a single-GPU box.)
Each worker is bounded; failure to init prints PROBE_UNSUPPORTED and
exits 0 so the caller treats it as a clean skip."""
import os
import sys
import time

import torch
import torch.multiprocessing as mp


def worker(rank: int, world: int) -> None:
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT="29873",
                      LOCAL_RANK="0",  # both ranks -> cuda:0
                      HSA_ENABLE_IPC_MODE_LEGACY="0")
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch.distributed as dist
    from datetime import timedelta
    torch.cuda.set_device(0)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=world,
                                timeout=timedelta(seconds=60))
        t = torch.ones(4, device="cuda") * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert t[0].item() == 3.0, t
        if rank == 0:
            print("PROBE_OK: all_reduce works with 2 ranks on 1 GPU",
                  flush=True)
        # engine pipeline world-2 over real RCCL
        import matrel_amd as ma
        ma.set_config(ma.MatrelConfig(block_size=256))
        s = ma.MatrelSession()
        A = s.randn(1024, 768, seed=1)
        B = s.randn(768, 512, seed=2)
        C = (A @ B).compute()
        ref = A.to_tensor() @ B.to_tensor()
        err = (C.to_tensor() - ref).abs().max() / ref.abs().max()
        assert err < 2e-2, err.item()
        E = s.sparse_random(1024, 768, 0.02, seed=3)
        D = (E @ B).compute()
        refd = E.to_tensor() @ B.to_tensor()
        errd = (D.to_tensor() - refd).abs().max().item()
        assert errd < 0.5, errd
        if rank == 0:
            print("PROBE_ENGINE_OK: SUMMA + CSR wire over real RCCL "
                  f"(relerr {err.item():.2e})", flush=True)
        dist.destroy_process_group()
    except Exception as e:
        print(f"PROBE_UNSUPPORTED rank{rank}: {type(e).__name__}: "
              f"{str(e)[:200]}", flush=True)
        os._exit(0)


if __name__ == "__main__":
    t0 = time.time()
    mp.spawn(worker, args=(2,), nprocs=2, join=True)
    print(f"probe done in {time.time()-t0:.1f}s")
