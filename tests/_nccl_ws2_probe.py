"""Two-rank RCCL probe, launched under torch.distributed.run by
tests/test_gpu_ddp.py on a single-GPU box (both ranks share cuda:0).

Exercises the REAL GradAllReducer code path (parallel/ddp.py) over the
nccl(=RCCL) backend.  RCCL, like NCCL, may refuse two ranks on one device
("Duplicate GPU detected"); that outcome exits 77 so the caller can skip —
any other failure is a real bug.
"""
import os
import sys

# torchrun children get sys.path[0]=tests/ — put the repo root back
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main() -> int:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    device = torch.device("cuda:0")
    torch.cuda.set_device(device)
    import torch.distributed as dist

    try:
        dist.init_process_group(backend="nccl")
        from gymfx_amd.parallel.ddp import GradAllReducer, allreduce_mean_

        red = GradAllReducer(world, dist.group.WORLD, timeout_s=60.0)
        grads = torch.full((107_000,), float(rank + 1), device=device)
        red.start(grads)
        # overlap window: unrelated compute on the current stream
        filler = torch.ones(256, 256, device=device) @ torch.ones(256, 256, device=device)
        red.finish()
        torch.cuda.synchronize()
        expect = (1.0 + world) / 2.0  # mean over ranks of (rank+1)
        assert torch.allclose(grads, torch.full_like(grads, expect)), (
            f"rank {rank}: all-reduce mean wrong: {grads[0].item()} != {expect}")
        m = torch.tensor([float(rank)], device=device)
        allreduce_mean_(m, world)
        assert abs(m.item() - (world - 1) / 2.0) < 1e-6
        assert filler[0, 0].item() == 256.0
        dist.destroy_process_group()
        print(f"rank {rank}: WS2_PROBE_OK")
        return 0
    except Exception as exc:  # noqa: BLE001
        import traceback

        msg = str(exc)
        tb = traceback.format_exc()
        # torchrun swallows child tracebacks: persist them for the caller
        try:
            os.makedirs("gpurun_out", exist_ok=True)
            with open(f"gpurun_out/ws2_probe_rank{rank}.log", "w") as f:
                f.write(tb)
        except OSError:
            pass
        print(tb, flush=True)
        low = msg.lower()
        if ("duplicate gpu" in low or "invalid usage" in low
                or "invalid argument" in low or "unhandled system error" in low):
            print(f"rank {rank}: RCCL refuses shared device: {msg}", flush=True)
            return 77
        raise


if __name__ == "__main__":
    sys.exit(main())
