"""One-shot xGMI all-reduce (csrc/allreduce.hip) validated with two
processes sharing ONE GPU: hipIpc handle exchange, the stage/flag/reduce
protocol, parity double-buffering and the init self-test all run exactly
as they would across GPUs — only the transport differs (local HBM vs
xGMI links)."""
import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _rank_main(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from agentfield_amd.parallel.oneshot import OneShotAllReduce
        dev = "cuda:0"  # both ranks on one device: IPC still cross-process
        torch.cuda.set_device(dev)
        osr = OneShotAllReduce(dist.group.WORLD, dev)
        assert osr.ok, "one-shot init/self-test failed"
        results = []
        for trial in range(4):  # exercise both parity slots twice
            n = [4096, 64 * 4096, 1000, 4096][trial]
            x = (torch.arange(n, device=dev, dtype=torch.float32)
                 * 1e-4 + (rank + 1) * (trial + 1)).bfloat16()
            # reference reduced on CPU (bit-exact ranks sum)
            wc = x.float().cpu()
            dist.all_reduce(wc)
            got = osr.allreduce(x)
            torch.cuda.synchronize()
            ok = torch.allclose(got.float().cpu(), wc, atol=5e-2, rtol=5e-2)
            results.append(bool(ok))
        if rank == 0:
            q.put(("ok", results))
    except Exception as e:
        import traceback
        if rank == 0:
            q.put(("err", f"{e}\n{traceback.format_exc()}"))
        raise
    finally:
        dist.destroy_process_group()


def test_oneshot_allreduce_two_procs_one_gpu():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, 29731, q))
             for r in range(2)]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=240)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert all(payload), f"mismatch in trials: {payload}"
