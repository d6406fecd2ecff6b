"""One-shot fused xGMI all-reduce: 2 processes on ONE GPU.

hipIpc handles work across processes on the same device, so the IPC
plumbing, flag protocol and fused reduce+add+rmsnorm numerics are all
validated on a 1-GPU box (XgmiAllReduce's own init self-check compares
against a gloo all-reduce + the reference epilogue). Multi-device xGMI
bandwidth is exercised by the driver's 8-GPU bench (bench.py --tp 8).
"""
import multiprocessing as mp
import socket

import pytest
import torch

pytestmark = pytest.mark.gpu


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, port, q):
    import torch.distributed as dist

    try:
        dist.init_process_group(
            "gloo",
            init_method=f"tcp://127.0.0.1:{port}",
            rank=rank,
            world_size=world,
        )
        dev = torch.device("cuda:0")
        torch.cuda.set_device(dev)
        from kubeai_amd.parallel.comm import XgmiAllReduce

        comm = XgmiAllReduce(None, dev)  # init runs the self-check
        # a few extra rounds with fresh tensors (exercises both slots +
        # the monotonic seq protocol)
        from kubeai_amd import ops

        for round_i in range(5):
            torch.manual_seed(100 + round_i)
            T, H = (7, 1024) if round_i % 2 else (40, 4096)
            base = torch.randn(world, T, H, dtype=torch.bfloat16, device=dev)
            res0 = torch.randn(T, H, dtype=torch.bfloat16, device=dev)
            w = torch.randn(H, dtype=torch.bfloat16, device=dev)
            x, res = base[rank].clone(), res0.clone()
            x, res = comm.fused_allreduce_add_rmsnorm(x, res, w, 1e-5)
            want_sum = base.float().sum(0) + res0.float()
            rrms = torch.rsqrt(want_sum.pow(2).mean(-1, keepdim=True) + 1e-5)
            want_x = want_sum * rrms * w.float()
            torch.cuda.synchronize()
            assert torch.allclose(
                res.float(), want_sum, atol=3e-2, rtol=3e-2
            ), f"residual mismatch round {round_i}"
            assert torch.allclose(
                x.float(), want_x, atol=5e-2, rtol=5e-2
            ), f"norm mismatch round {round_i}"
        assert comm.error_count() == 0
        if rank == 0:
            q.put(("ok", None))
    except Exception:  # noqa: BLE001
        import traceback

        q.put(("err", traceback.format_exc()))
        raise
    finally:
        try:
            from kubeai_amd.ops import _C

            _C.xgmi_shutdown()
        except Exception:
            pass


def test_one_shot_allreduce_two_procs_one_gpu():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
    assert status == "ok", payload
