"""Stage-to-stage P2P with cached tensor metadata
(reference: colossalai/pipeline/p2p.py:364,539).

Protocol: the first exchange in each (peer, direction) sends a fixed-size
int64 header describing the payload STRUCTURE — n_tensors and per-tensor
(ndim, dtype, dims) — after which the header is cached and only payloads
travel (metadata is constant across microbatches). Stage IO may be a single
tensor or a list/tuple of tensors (multi-tensor models: encoder state +
mask, MoE aux losses, ...); multi-tensor payloads go out as one batched
isend group. Payload recv is blocking dist.recv on the default group — on
ROCm this is RCCL point-to-point over xGMI.
"""

from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

from .stage_manager import PipelineStageManager

__all__ = ["PipelineP2PCommunication"]

_MAX_TENSORS = 8
_MAX_DIMS = 6
_SLOT = 2 + _MAX_DIMS
_HEADER_LEN = 1 + _MAX_TENSORS * _SLOT
_DTYPE_CODES = {
    torch.float32: 0,
    torch.float16: 1,
    torch.bfloat16: 2,
    torch.int64: 3,
    torch.int32: 4,
}
_CODE_DTYPES = {v: k for k, v in _DTYPE_CODES.items()}


class PipelineP2PCommunication:
    def __init__(self, stage_manager: PipelineStageManager, overlap_p2p: bool = False):
        self.stage_manager = stage_manager
        self.overlap_p2p = overlap_p2p
        self._send_meta: Dict[int, bool] = {}  # peer -> header already sent
        self._recv_meta: Dict[int, Tuple[torch.Size, torch.dtype]] = {}
        self._device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        self._pending = []  # outstanding (work, tensor) isends

    # ------------------------------------------------------------- internals
    def _isend(self, tensor: torch.Tensor, peer: int) -> None:
        work = dist.isend(tensor, peer)
        self._pending.append((work, tensor))
        if len(self._pending) > 32:
            w, _ = self._pending.pop(0)
            w.wait()

    def flush_sends(self) -> None:
        for w, _ in self._pending:
            w.wait()
        self._pending.clear()

    def _send_tensor(self, obj, peer: int) -> None:
        # sends are NON-BLOCKING: interleaved/1F1B warmups legitimately have
        # both neighbors sending before anyone receives (rendezvous sends
        # deadlock there). Tensors are kept alive until the work completes.
        tensors = [obj] if isinstance(obj, torch.Tensor) else list(obj)
        assert 1 <= len(tensors) <= _MAX_TENSORS, "stage IO: 1..8 tensors"
        tensors = [t.contiguous() for t in tensors]
        struct = (-1 if isinstance(obj, torch.Tensor) else len(tensors))
        prev = self._send_meta.get(peer)
        assert prev is None or prev == struct, (
            "stage IO structure changed mid-stream (metadata is cached after the "
            "first microbatch); call clear_meta_cache() on both sides first"
        )
        if prev is None:
            header = torch.zeros(_HEADER_LEN, dtype=torch.int64, device=self._device)
            header[0] = len(tensors) if not isinstance(obj, torch.Tensor) else -1
            for k, t in enumerate(tensors):
                assert t.dim() <= _MAX_DIMS
                base = 1 + k * _SLOT
                header[base] = t.dim()
                header[base + 1] = _DTYPE_CODES[t.dtype]
                for i, d in enumerate(t.shape):
                    header[base + 2 + i] = d
            self._isend(header, peer)
            self._send_meta[peer] = struct
        for t in tensors:
            self._isend(t, peer)

    def _recv_tensor(self, peer: int):
        if peer not in self._recv_meta:
            header = torch.zeros(_HEADER_LEN, dtype=torch.int64, device=self._device)
            dist.recv(header, peer)
            header = header.cpu()
            n = int(header[0])
            single = n == -1
            n = 1 if single else n
            metas = []
            for k in range(n):
                base = 1 + k * _SLOT
                ndim = int(header[base])
                dtype = _CODE_DTYPES[int(header[base + 1])]
                shape = torch.Size(int(header[base + 2 + i]) for i in range(ndim))
                metas.append((shape, dtype))
            self._recv_meta[peer] = (single, metas)
        single, metas = self._recv_meta[peer]
        bufs = []
        for shape, dtype in metas:
            buf = torch.empty(shape, dtype=dtype, device=self._device)
            dist.recv(buf, peer)
            bufs.append(buf)
        return bufs[0] if single else bufs

    def clear_meta_cache(self) -> None:
        self._send_meta.clear()
        self._recv_meta.clear()

    # ------------------------------------------------------------------- api
    def send_forward(self, output: torch.Tensor, peer: Optional[int] = None) -> None:
        peer = self.stage_manager.get_next_rank() if peer is None else peer
        self._send_tensor(output, peer)

    def recv_forward(self, peer: Optional[int] = None) -> torch.Tensor:
        peer = self.stage_manager.get_prev_rank() if peer is None else peer
        return self._recv_tensor(peer)

    def send_backward(self, grad: torch.Tensor, peer: Optional[int] = None) -> None:
        peer = self.stage_manager.get_prev_rank() if peer is None else peer
        self._send_tensor(grad, peer)

    def recv_backward(self, peer: Optional[int] = None) -> torch.Tensor:
        peer = self.stage_manager.get_next_rank() if peer is None else peer
        return self._recv_tensor(peer)

    def send_forward_recv_backward(self, output: torch.Tensor) -> torch.Tensor:
        # even/odd stage ordering avoids send/send deadlock on blocking links
        if self.stage_manager.stage % 2 == 0:
            self.send_forward(output)
            return self.recv_backward()
        grad = self.recv_backward()
        self.send_forward(output)
        return grad

    def send_backward_recv_forward(self, grad: torch.Tensor) -> torch.Tensor:
        if self.stage_manager.stage % 2 == 0:
            self.send_backward(grad)
            return self.recv_forward()
        x = self.recv_forward()
        self.send_backward(grad)
        return x
