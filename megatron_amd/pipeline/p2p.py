"""Point-to-point pipeline communication.

Capability analog of reference megatron/core/pipeline_parallel/
p2p_communication.py (P2PCommunicator :140, _p2p_ops :55, _batched_p2p_ops
:17).  All four directions can be fused into one ``batch_isend_irecv`` so
concurrent sends/recvs match without ordering deadlocks — on one xGMI node
every PP hop is a single direct link.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


class P2PCommunicator:
    def __init__(self, config, seq_length: int, micro_batch_size: int):
        self.config = config
        grid = G.get_grid()
        self.grid = grid
        self.prev_rank = grid.pipeline_prev_rank()
        self.next_rank = grid.pipeline_next_rank()
        self.group = grid.group("pp")          # forward direction (activations)
        try:
            self.group_bwd = grid.group("pp_bwd")  # backward direction (grads)
        except KeyError:
            self.group_bwd = self.group
        s = seq_length
        if config.sequence_parallel:
            s //= max(1, config.tensor_parallel_size)
        s //= max(1, config.context_parallel_size)
        self.shape = (s, micro_batch_size, config.hidden_size)
        self.dtype = config.pipeline_dtype

    def _buf(self, device) -> torch.Tensor:
        return torch.empty(self.shape, dtype=self.dtype, device=device, requires_grad=True)

    def _device(self):
        return torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")

    def _exchange_shapes(self, tensor_send_next, tensor_send_prev,
                         recv_next: bool, recv_prev: bool, dev):
        """Variable-seq-len shape pre-exchange (reference
        p2p_communication.py:186 _communicate_shapes): a tiny [3] int64
        message per direction carries the payload shape so recv buffers can
        be sized exactly."""
        to_t = lambda t: torch.tensor(t.shape, dtype=torch.int64, device=dev)
        recv_prev_shape = torch.empty(3, dtype=torch.int64, device=dev) if recv_prev else None
        recv_next_shape = torch.empty(3, dtype=torch.int64, device=dev) if recv_next else None
        ops = []
        even = self.grid.pp_rank % 2 == 0

        def sends():
            if tensor_send_prev is not None:
                ops.append(dist.P2POp(dist.isend, to_t(tensor_send_prev), self.prev_rank, group=self.group_bwd))
            if tensor_send_next is not None:
                ops.append(dist.P2POp(dist.isend, to_t(tensor_send_next), self.next_rank, group=self.group))

        def recvs():
            if recv_prev_shape is not None:
                ops.append(dist.P2POp(dist.irecv, recv_prev_shape, self.prev_rank, group=self.group))
            if recv_next_shape is not None:
                ops.append(dist.P2POp(dist.irecv, recv_next_shape, self.next_rank, group=self.group_bwd))

        if even:
            sends()
            recvs()
        else:
            recvs()
            sends()
        reqs = []
        groups = [self.group] + ([self.group_bwd] if self.group_bwd is not self.group else [])
        for grp in groups:
            grp_ops = [o for o in ops if o.group is grp]
            if grp_ops:
                reqs.extend(dist.batch_isend_irecv(grp_ops))
        for r in reqs:
            r.wait()
        return (tuple(recv_prev_shape.tolist()) if recv_prev_shape is not None else None,
                tuple(recv_next_shape.tolist()) if recv_next_shape is not None else None)

    def communicate(
        self,
        tensor_send_next: Optional[torch.Tensor] = None,
        tensor_send_prev: Optional[torch.Tensor] = None,
        recv_next: bool = False,
        recv_prev: bool = False,
    ):
        """Fused bidirectional exchange; returns (tensor_recv_prev, tensor_recv_next)."""
        dev = self._device()
        if self.config.variable_seq_lengths:
            prev_shape, next_shape = self._exchange_shapes(
                tensor_send_next, tensor_send_prev, recv_next, recv_prev, dev)
            tensor_recv_prev = (torch.empty(prev_shape, dtype=self.dtype, device=dev,
                                            requires_grad=True) if recv_prev else None)
            tensor_recv_next = (torch.empty(next_shape, dtype=self.dtype, device=dev,
                                            requires_grad=True) if recv_next else None)
        else:
            tensor_recv_prev = self._buf(dev) if recv_prev else None
            tensor_recv_next = self._buf(dev) if recv_next else None
        ops = []
        # rank-parity ordering keeps pairwise matching deterministic even if
        # the backend serializes (reference p2p_communication.py:67)
        even = self.grid.pp_rank % 2 == 0
        def add_sends():
            if tensor_send_prev is not None:
                ops.append(dist.P2POp(dist.isend, tensor_send_prev.contiguous(), self.prev_rank, group=self.group_bwd))
            if tensor_send_next is not None:
                ops.append(dist.P2POp(dist.isend, tensor_send_next.contiguous(), self.next_rank, group=self.group))
        def add_recvs():
            if tensor_recv_prev is not None:
                ops.append(dist.P2POp(dist.irecv, tensor_recv_prev, self.prev_rank, group=self.group))
            if tensor_recv_next is not None:
                ops.append(dist.P2POp(dist.irecv, tensor_recv_next, self.next_rank, group=self.group_bwd))
        if even:
            add_sends()
            add_recvs()
        else:
            add_recvs()
            add_sends()
        # batch_isend_irecv requires a single group per batch: issue one
        # batch per direction-communicator, wait on both
        reqs = []
        groups = [self.group] + ([self.group_bwd] if self.group_bwd is not self.group else [])
        for grp in groups:
            grp_ops = [o for o in ops if o.group is grp]
            if grp_ops:
                reqs.extend(dist.batch_isend_irecv(grp_ops))
        for r in reqs:
            r.wait()
        return tensor_recv_prev, tensor_recv_next

    # -- convenience wrappers (reference API shape) --------------------------

    def recv_forward(self, is_first_stage: bool):
        if is_first_stage:
            return None
        t, _ = self.communicate(recv_prev=True)
        return t

    def recv_backward(self, is_last_stage: bool):
        if is_last_stage:
            return None
        _, t = self.communicate(recv_next=True)
        return t

    def send_forward(self, output, is_last_stage: bool):
        if not is_last_stage:
            self.communicate(tensor_send_next=output)

    def send_backward(self, input_grad, is_first_stage: bool):
        if not is_first_stage:
            self.communicate(tensor_send_prev=input_grad)

    def send_forward_recv_backward(self, output, is_last_stage: bool):
        if is_last_stage:
            return None
        _, grad = self.communicate(tensor_send_next=output, recv_next=True)
        return grad

    def send_backward_recv_forward(self, input_grad, is_first_stage: bool):
        if is_first_stage:
            return None
        t, _ = self.communicate(tensor_send_prev=input_grad, recv_prev=True)
        return t
