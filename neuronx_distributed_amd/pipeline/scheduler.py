"""Pipeline schedules as explicit task streams.

Parity with reference ``pipeline/scheduler.py``: ``InferenceSchedule``
(:144), ``Train1F1BSchedule`` (:157-253), ``TrainInterleavedSchedule``
(:256-541).  Each schedule yields :class:`PipelineTask` objects the engine
executes via an instruction map (reference model.py:1716-1743).

With real RCCL P2P the bidirectional steady-state exchanges are batched
(SendFwdRecvBwd / SendBwdRecvFwd), which removes the reference's
deadlock-ordering constraints (scheduler.py:226-233).  The reference's
odd/even rank scheduler for num_microbatches == pp_size (model.py:288-295)
exists only to break its all-gather-P2P ordering cycles — with true
bidirectional P2P the plain 1F1B stream is already cycle-free for every
mb/pp combination, so no parity-special schedule is needed here.
"""

from dataclasses import dataclass
from typing import Iterator


@dataclass(frozen=True)
class PipelineTask:
    mb: int  # microbatch index
    chunk: int = 0  # model chunk (interleaved schedule)
    # second (mb, chunk) pair for FUSED bidirectional tasks: the RECV side
    # of SendBackwardRecvForward (reference scheduler.py:281-293 fused
    # send-recv in the interleaved schedule)
    mb2: int = -1
    chunk2: int = -1


class RecvForward(PipelineTask):
    pass


class ForwardStep(PipelineTask):
    pass


class SendForward(PipelineTask):
    pass


class RecvBackward(PipelineTask):
    pass


class BackwardStep(PipelineTask):
    pass


class SendBackward(PipelineTask):
    pass


class SendForwardRecvBackward(PipelineTask):
    pass


class SendBackwardRecvForward(PipelineTask):
    pass


class ReduceGrads(PipelineTask):
    pass


class BaseSchedule:
    def __init__(self, num_microbatches: int, stage_id: int, num_stages: int):
        self.num_microbatches = num_microbatches
        self.stage_id = stage_id
        self.num_stages = num_stages

    @property
    def is_first(self):
        return self.stage_id == 0

    @property
    def is_last(self):
        return self.stage_id == self.num_stages - 1

    def steps(self) -> Iterator[PipelineTask]:
        raise NotImplementedError


class InferenceSchedule(BaseSchedule):
    """Forward-only (reference scheduler.py:144)."""

    def steps(self):
        for mb in range(self.num_microbatches):
            if not self.is_first:
                yield RecvForward(mb)
            yield ForwardStep(mb)
            if not self.is_last:
                yield SendForward(mb)


class Train1F1BSchedule(BaseSchedule):
    """One-forward-one-backward (reference scheduler.py:157-253)."""

    def steps(self):
        n = self.num_microbatches
        warmup = min(self.num_stages - self.stage_id - 1, n)
        steady = n - warmup

        # warmup forwards
        for mb in range(warmup):
            if not self.is_first:
                yield RecvForward(mb)
            yield ForwardStep(mb)
            if not self.is_last:
                yield SendForward(mb)

        # steady state: 1 forward + 1 backward per iteration.  The
        # send-forward/recv-backward pair is BATCHED (both directions in one
        # batch_isend_irecv) so neighbours can't deadlock regardless of
        # their own op order.
        for i in range(steady):
            fwd_mb = warmup + i
            bwd_mb = i
            if not self.is_first:
                yield RecvForward(fwd_mb)
            yield ForwardStep(fwd_mb)
            if self.is_last:
                yield BackwardStep(bwd_mb)
            else:
                yield SendForwardRecvBackward(fwd_mb)
                yield BackwardStep(bwd_mb)
            if not self.is_first:
                yield SendBackward(bwd_mb)

        # cooldown backwards
        for i in range(steady, n):
            if not self.is_last:
                yield RecvBackward(i)
            yield BackwardStep(i)
            if not self.is_first:
                yield SendBackward(i)

        yield ReduceGrads(0)


class TrainInterleavedSchedule(BaseSchedule):
    """Interleaved/virtual-pipeline schedule (reference scheduler.py:256-541,
    Megatron-style): each rank owns ``num_chunks`` model chunks; microbatch
    work is issued in chunk-major order during warmup and 1F1B afterwards."""

    def __init__(self, num_microbatches, stage_id, num_stages, num_chunks=1):
        super().__init__(num_microbatches, stage_id, num_stages)
        self.num_chunks = num_chunks

    def steps(self):
        n = self.num_microbatches
        P = self.num_stages
        C = self.num_chunks
        total = n * C
        if C == 1:
            yield from Train1F1BSchedule(n, self.stage_id, P).steps()
            return

        # virtual microbatch k runs chunk (k // n_per_round) per round-robin
        # Megatron interleave: groups of P microbatches cycle chunks
        warmup = min((P - self.stage_id - 1) * 2 + (C - 1) * P, total)

        def chunk_of(step, forward):
            k = step // P
            c = k % C
            return c if forward else (C - 1 - c)

        def mb_of(step):
            k = step // P
            return (k // C) * P + step % P

        fwd_i = 0
        bwd_i = 0
        for _ in range(warmup):
            c = chunk_of(fwd_i, True)
            mb = mb_of(fwd_i)
            if not (self.is_first and c == 0):
                yield RecvForward(mb, c)
            yield ForwardStep(mb, c)
            if not (self.is_last and c == C - 1):
                yield SendForward(mb, c)
            fwd_i += 1
        # steady state with FUSED bidirectional exchanges (reference
        # scheduler.py:281-293): send-fwd + recv-bwd batched toward the
        # next rank, send-bwd + recv-next-fwd batched toward the previous
        # rank — each pair flies in one batched isend/irecv.
        for _ in range(total - warmup):
            c = chunk_of(fwd_i, True)
            mb = mb_of(fwd_i)
            if not (self.is_first and c == 0):
                yield RecvForward(mb, c)
            yield ForwardStep(mb, c)
            cb = chunk_of(bwd_i, False)
            mbb = mb_of(bwd_i)
            send_f = not (self.is_last and c == C - 1)
            recv_b = not (self.is_last and cb == C - 1)
            if send_f and recv_b:
                yield SendForwardRecvBackward(mb, c)
            elif send_f:
                yield SendForward(mb, c)
            elif recv_b:
                yield RecvBackward(mbb, cb)
            yield BackwardStep(mbb, cb)
            if not (self.is_first and cb == 0):
                yield SendBackward(mbb, cb)
            bwd_i += 1
            fwd_i += 1
        for _ in range(total - bwd_i):
            cb = chunk_of(bwd_i, False)
            mbb = mb_of(bwd_i)
            if not (self.is_last and cb == C - 1):
                yield RecvBackward(mbb, cb)
            yield BackwardStep(mbb, cb)
            if not (self.is_first and cb == 0):
                yield SendBackward(mbb, cb)
            bwd_i += 1
        yield ReduceGrads(0)
