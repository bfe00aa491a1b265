"""One-shot all-reduce over xGMI peer buffers (csrc/allreduce.hip).

Latency-optimized path for the small bf16 [B, 4096] partial sums a TP
decode step produces: every rank stages into its own
hipIpc-shared buffer and reads all peers directly — one xGMI hop instead
of a 2(W-1)-hop ring.  IPC handles are exchanged once at init; an init
self-test reduces a known pattern and compares against RCCL before the
path is ever used (falls back to dist.all_reduce on any failure).
"""
from __future__ import annotations

import ctypes

import torch
import torch.distributed as dist

from ..ops import _lib

HDR_ELEMS = 64                 # 128-B header as bf16 elements
DEFAULT_SLOT_ELEMS = 128 * 4096  # 1 MiB/slot: covers decode batches


class OneShotAllReduce:
    def __init__(self, group, device, slot_elems: int = DEFAULT_SLOT_ELEMS):
        self.group = group
        self.device = torch.device(device)
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        self.slot_elems = slot_elems
        self.seq = 0
        self.ok = False
        if self.world < 2 or self.world > 8 or self.device.type != "cuda":
            return
        try:
            lib = _lib.lib()
            # header + two parity slots
            self.buf = torch.zeros(HDR_ELEMS + 2 * slot_elems,
                                   dtype=torch.bfloat16, device=self.device)
            h = (ctypes.c_char * 64)()
            rc = lib.af_ipc_get_handle(
                ctypes.c_void_p(self.buf.data_ptr()), h)
            if rc != 0:
                return
            handles: list = [None] * self.world
            dist.all_gather_object(handles, bytes(h), group=group)
            ptrs = (ctypes.c_void_p * self.world)()
            self._opened = []
            for r in range(self.world):
                if r == self.rank:
                    ptrs[r] = self.buf.data_ptr()
                else:
                    p = ctypes.c_void_p()
                    rc = lib.af_ipc_open_handle(
                        (ctypes.c_char * 64).from_buffer_copy(handles[r]),
                        ctypes.byref(p))
                    if rc != 0:
                        return
                    ptrs[r] = p.value
                    self._opened.append(p.value)
            self.ptrs = ptrs
            self.ok = self._selftest()
        except Exception:
            self.ok = False

    def _selftest(self) -> bool:
        """Reduce a rank-dependent pattern twice (both parity slots) and
        compare against RCCL's answer on every rank."""
        n = 4096
        x = (torch.arange(n, device=self.device, dtype=torch.float32)
             * 1e-3 + self.rank).bfloat16()
        want = x.clone()
        dist.all_reduce(want, group=self.group)
        good = True
        for _ in range(2):
            got = self.allreduce(x.clone())
            torch.cuda.synchronize()
            if not torch.allclose(got.float(), want.float(),
                                  atol=2e-2, rtol=2e-2):
                good = False
        v = torch.tensor([int(good)], dtype=torch.int32, device=self.device)
        dist.all_reduce(v, op=dist.ReduceOp.MIN, group=self.group)
        return bool(int(v.item()))

    def eligible(self, t: torch.Tensor) -> bool:
        return (self.ok and t.dtype == torch.bfloat16 and t.is_contiguous()
                and t.numel() <= self.slot_elems
                and not torch.cuda.is_current_stream_capturing())

    def allreduce(self, t: torch.Tensor) -> torch.Tensor:
        """In-place-semantics all-reduce (returns the reduced tensor).
        Caller must invoke collectively, same order on every rank."""
        self.seq += 1
        out = torch.empty_like(t)
        rc = _lib.lib().af_oneshot_allreduce(
            ctypes.c_void_p(out.data_ptr()), ctypes.c_void_p(t.data_ptr()),
            self.ptrs, self.world, self.rank,
            ctypes.c_long(t.numel()), ctypes.c_long(self.slot_elems),
            ctypes.c_ulonglong(self.seq), _lib.cur_stream())
        _lib.check(rc, "af_oneshot_allreduce")
        t.copy_(out)
        return t
