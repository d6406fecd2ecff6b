"""One-shot fused all-reduce over xGMI (csrc/allreduce.hip wrapper).

Each TP rank shares an 8 MiB-per-slot device buffer via HIP IPC
(dmabuf mode — HSA_ENABLE_IPC_MODE_LEGACY=0); the fused kernel pair
replaces every decode-path (all_reduce -> add -> rmsnorm) sequence with
two launches and zero host round trips (SURVEY hard part #2: at TP=8
small-tensor all-reduce LATENCY bounds the decode step; a ring also
pays ~2*(N-1)/N of the bytes over one 153 GB/s link — one-shot reads
each peer's partial exactly once, directly over the p2p mesh).

Safety: a randomized self-check at init compares the fused kernel
against RCCL all_reduce + the reference epilogue; any mismatch or IPC
failure falls the TPGroup back to the RCCL path. Bounded device
spin-waits surface as xgmi_error_count() instead of wedging the GPU.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from kubeai_amd import ops


class XgmiAllReduce:
    def __init__(self, group, device: torch.device):
        from kubeai_amd.ops import _C  # raises if the extension is absent

        self._C = _C
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        self.max_elems = int(_C.xgmi_max_elems())
        handle = _C.xgmi_alloc()
        gathered: list = [None] * self.world
        dist.all_gather_object(gathered, bytes(handle), group=group)
        _C.xgmi_connect(self.rank, self.world, list(gathered))
        self._self_check(device)

    def fits(self, x: torch.Tensor) -> bool:
        return (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and x.is_contiguous()
            and x.numel() <= self.max_elems
        )

    def fused_allreduce_add_rmsnorm(self, x, residual, weight, eps):
        self._C.xgmi_fused_allreduce_add_rmsnorm(x, residual, weight, eps)
        return x, residual

    def error_count(self) -> int:
        return int(self._C.xgmi_error_count())

    def _self_check(self, device) -> None:
        """Fused kernel must agree with RCCL + reference epilogue."""
        torch.manual_seed(1234)  # identical on every rank
        for T, H in ((3, 512), (40, 4096)):
            base = torch.randn(
                self.world, T, H, dtype=torch.bfloat16, device=device
            )
            res0 = torch.randn(T, H, dtype=torch.bfloat16, device=device)
            w = torch.randn(H, dtype=torch.bfloat16, device=device)
            x = base[self.rank].clone()
            res = res0.clone()
            x, res = self.fused_allreduce_add_rmsnorm(x, res, w, 1e-5)
            # reference: RCCL all-reduce + fused_add_rmsnorm
            xr = base[self.rank].clone()
            rr = res0.clone()
            dist.all_reduce(xr, group=self.group)
            xr, rr = ops.fused_add_rmsnorm(xr, rr, w, 1e-5)
            torch.cuda.synchronize(device)
            if not torch.allclose(
                x.float(), xr.float(), atol=2e-2, rtol=2e-2
            ) or not torch.allclose(
                res.float(), rr.float(), atol=2e-2, rtol=2e-2
            ):
                raise RuntimeError(
                    "xgmi one-shot all-reduce self-check mismatch "
                    f"(max err {(x.float()-xr.float()).abs().max().item():.4f})"
                )
        if self.error_count():
            raise RuntimeError("xgmi comm reported spin-wait timeouts")
