"""Stage-to-stage P2P with cached tensor metadata
(reference: colossalai/pipeline/p2p.py:539).

Protocol: the first exchange in each (peer, tag-direction) sends a small
int64 header [ndim | dtype-code | dims...]; after both sides have seen one
message the header is cached and only payloads travel (metadata is constant
across microbatches). Payloads use blocking dist.send/recv on the default
group — on ROCm this is RCCL point-to-point over xGMI.
"""

from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

from .stage_manager import PipelineStageManager

__all__ = ["PipelineP2PCommunication"]

_HEADER_LEN = 10
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

    def _send_tensor(self, tensor: torch.Tensor, peer: int) -> None:
        # sends are NON-BLOCKING: interleaved/1F1B warmups legitimately have
        # both neighbors sending before anyone receives (rendezvous sends
        # deadlock there). Tensors are kept alive until the work completes.
        tensor = tensor.contiguous()
        if not self._send_meta.get(peer, False):
            header = torch.zeros(_HEADER_LEN, dtype=torch.int64, device=self._device)
            header[0] = tensor.dim()
            header[1] = _DTYPE_CODES[tensor.dtype]
            for i, d in enumerate(tensor.shape):
                header[2 + i] = d
            self._isend(header, peer)
            self._send_meta[peer] = True
        self._isend(tensor, peer)

    def _recv_tensor(self, peer: int) -> torch.Tensor:
        if peer not in self._recv_meta:
            header = torch.zeros(_HEADER_LEN, dtype=torch.int64, device=self._device)
            dist.recv(header, peer)
            header = header.cpu()
            ndim = int(header[0])
            dtype = _CODE_DTYPES[int(header[1])]
            shape = torch.Size(int(header[2 + i]) for i in range(ndim))
            self._recv_meta[peer] = (shape, dtype)
        shape, dtype = self._recv_meta[peer]
        buf = torch.empty(shape, dtype=dtype, device=self._device)
        dist.recv(buf, peer)
        return buf

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
