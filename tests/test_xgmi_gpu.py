"""xGMI custom all-reduce (reference counterpart: csrc/custom_all_reduce).

The 2-process test shares ONE GPU between two ranks via HIP IPC (dmabuf
mode) — it validates the full handle-exchange + generation-barrier +
reduce protocol; cross-device xGMI runs on the 8-GPU node at round end.
"""
import multiprocessing as mp
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_xgmi_world1_identity():
    import realhf_amd._C as C

    h = C.xgmi_create(0, 1, 1 << 20)
    hs = [bytes(b) for b in C.xgmi_handles(h)]
    C.xgmi_connect(h, [hs[0]], [hs[1]])
    for dtype in (torch.bfloat16, torch.float32):
        x = torch.randn(4096, device="cuda").to(dtype)
        out = C.xgmi_all_reduce(h, x)
        torch.cuda.synchronize()
        torch.testing.assert_close(out.float(), x.float(), atol=1e-2,
                                   rtol=1e-2)
    # repeated calls exercise the monotonic generation counter
    x = torch.ones(8192, device="cuda", dtype=torch.float32)
    for _ in range(5):
        out = C.xgmi_all_reduce(h, x)
    torch.cuda.synchronize()
    assert out.sum().item() == 8192
    assert C.xgmi_status(h) == 0
    C.xgmi_destroy(h)


def _xgmi_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    # force the 2-stage path for the "big" message below (the C++ reads
    # this once, at the first all_reduce of the process)
    os.environ["REALHF_AMD_XGMI_2STAGE_BYTES"] = "65536" 
    import torch
    import torch.distributed as dist

    try:
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)  # both ranks share the single test GPU
        from realhf_amd.parallel.xgmi import XgmiAllReduce

        ar = XgmiAllReduce(capacity_bytes=1 << 20)
        x = torch.full((4096,), float(rank + 1), device="cuda",
                       dtype=torch.float32)
        out = ar.all_reduce(x)
        torch.cuda.synchronize()
        expect = sum(range(1, world + 1))
        ok1 = bool((out == expect).all().item())
        # second round: bf16, different values
        y = (torch.randn(2048, device="cuda") * 0.1).to(torch.bfloat16)
        y_cpu = y.float().cpu()  # gloo gathers fp32 on cpu
        ys = [torch.empty_like(y_cpu) for _ in range(world)]
        dist.all_gather(ys, y_cpu)
        out2 = ar.all_reduce(y)
        torch.cuda.synchronize()
        ref = sum(ys)
        ok2 = bool(torch.allclose(out2.float().cpu(), ref, atol=0.1,
                                  rtol=0.05))
        # large message -> 2-stage (reduce-scatter + all-gather) path
        big = torch.full((65536,), float(rank + 1), device="cuda",
                         dtype=torch.float32)
        out3 = ar.all_reduce(big)
        torch.cuda.synchronize()
        ok2b = bool((out3 == expect).all().item())
        okst = ar.status_ok()
        detail = (f"ok1={ok1} ok2={ok2} ok2b={ok2b} status={okst} "
                  f"big_uniq={out3.unique()[:4].tolist()}")
        ar.close()
        dist.barrier()
        q.put((rank, ok1 and ok2 and ok2b and okst, detail))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def _run_two_process_once():
    from realhf_amd.base.testing import find_free_port

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = find_free_port()
    procs = [ctx.Process(target=_xgmi_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, ok, err = q.get(timeout=180)
        results[rank] = (ok, err)
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    return results


def test_xgmi_two_process_one_gpu():
    results = _run_two_process_once()
    if not all(ok for ok, _ in results.values()):
        # multiprocess + IPC: allow one retry before declaring failure
        results = _run_two_process_once()
    assert all(ok for ok, _ in results.values()), results
