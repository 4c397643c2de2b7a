"""Pipeline P2P communication over RCCL/gloo.

The reference implements PP sends as 2-rank all-gathers because XLA lacked
real P2P (reference pipeline/comm.py:30-72); on MI355X we use genuine
``torch.distributed`` send/recv over xGMI, batched with
``batch_isend_irecv`` where both directions fly together (deadlock-free by
construction; the reference's ordering rules scheduler.py:226-233 become
unnecessary).

Shape metadata travels as a small fixed-size header tensor before each
payload (replacing the reference's TCPStore handshake, pipeline/comm.py:
114-211) — eager P2P makes the out-of-band store unnecessary.
"""

import pickle
from typing import List, Optional

import torch
import torch.distributed as dist

from ..parallel import parallel_state as ps

_MAX_DIMS = 8
_DTYPES = [torch.float32, torch.float16, torch.bfloat16, torch.int64,
           torch.int32, torch.bool, torch.float64, torch.uint8]
_OBJ_TAG = -1  # header dtype tag: payload is a pickled python object


def _header_from(tensor: torch.Tensor) -> torch.Tensor:
    h = torch.zeros(2 + _MAX_DIMS, dtype=torch.int64)
    h[0] = _DTYPES.index(tensor.dtype)
    h[1] = tensor.dim()
    for i, d in enumerate(tensor.shape):
        h[2 + i] = d
    return h


def _encode_item(item, dev):
    """(header, payload) for a tensor OR any picklable python object —
    non-tensor stage IO travels inline in the same message (reference
    partition.py:132-223 pass-through objects + pipeline/comm.py:114-211
    python-object channel)."""
    if isinstance(item, torch.Tensor):
        return (_header_from(item).to(_header_device()),
                item.detach().contiguous().to(dev))
    data = pickle.dumps(item)
    buf = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
    h = torch.zeros(2 + _MAX_DIMS, dtype=torch.int64)
    h[0] = _OBJ_TAG
    h[1] = buf.numel()
    return h.to(_header_device()), buf.to(_header_device())


def _recv_payload_shell(h_cpu: torch.Tensor, device):
    """Empty receive buffer described by a header (tensor or object)."""
    if int(h_cpu[0]) == _OBJ_TAG:
        return torch.empty(int(h_cpu[1]), dtype=torch.uint8, device=device)
    dtype = _DTYPES[int(h_cpu[0])]
    dims = [int(x) for x in h_cpu[2:2 + int(h_cpu[1])]]
    return torch.empty(dims, dtype=dtype, device=device)


def _decode_payload(h_cpu: torch.Tensor, payload: torch.Tensor):
    if int(h_cpu[0]) == _OBJ_TAG:
        return pickle.loads(bytes(payload.cpu().numpy().tobytes()))
    return payload


def _tensor_from_header(h: torch.Tensor, device) -> torch.Tensor:
    return _recv_payload_shell(h, device)


def _p2p_device():
    if torch.cuda.is_available() and dist.get_backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _header_device():
    # nccl needs device tensors even for the header
    return _p2p_device()


def send(items: List, dst: int):
    """Send a list of tensors / python objects (shapes inline)."""
    dev = _p2p_device()
    n = torch.tensor([len(items)], dtype=torch.int64, device=_header_device())
    dist.send(n, dst=dst)
    for t in items:
        h, p = _encode_item(t, dev)
        dist.send(h, dst=dst)
        dist.send(p, dst=dst)


def send_async(items: List, dst: int):
    """Post the whole message (count, headers, payloads) as isends and
    return (works, refs).  The caller must keep ``refs`` alive and wait the
    works before reusing/freeing — the engine drains them at schedule end.
    Blocking sends can mutually deadlock in the interleaved schedule where
    two neighbours send to each other concurrently (fwd one way, bwd the
    other); isends progress on the transport's own streams."""
    dev = _p2p_device()
    n = torch.tensor([len(items)], dtype=torch.int64,
                     device=_header_device())
    works, refs = [], [n]
    works.append(dist.isend(n, dst))
    for t in items:
        h, p = _encode_item(t, dev)
        refs += [h, p]
        works.append(dist.isend(h, dst))
        works.append(dist.isend(p, dst))
    return works, refs


def recv_from(src: int) -> List:
    dev = _p2p_device()
    n = torch.empty(1, dtype=torch.int64, device=_header_device())
    dist.recv(n, src=src)
    out = []
    for _ in range(int(n.item())):
        h = torch.empty(2 + _MAX_DIMS, dtype=torch.int64, device=_header_device())
        dist.recv(h, src=src)
        hc = h.cpu()
        t = _recv_payload_shell(hc, dev)
        dist.recv(t, src=src)
        out.append(_decode_payload(hc, t))
    return out


def send_recv(send_tensors: Optional[List[torch.Tensor]], dst: Optional[int],
              recv_src: Optional[int], defer_sends: bool = False):
    """Bidirectional exchange used by 1F1B steady state: both directions
    batched so neither side blocks (send_forward_recv_backward etc.).

    ``defer_sends=True`` returns ``(recv_out, works, refs)`` WITHOUT
    waiting the posted isends — the interleaved ring schedule can form a
    wait cycle where each rank's isend completes only after the peer's
    NEXT recv, so completion must be drained later by the engine's
    pending-send pool."""
    recv_out = None
    if send_tensors is not None and recv_src is not None:
        # Post sends async and NEVER wait on them before the matching
        # peer's receives can be posted (waiting an isend whose completion
        # needs the peer to progress past ITS sends deadlocks on ordered
        # transports).  Only irecvs gate each phase.
        dev = _p2p_device()
        pending_sends = []
        n = torch.tensor([len(send_tensors)], dtype=torch.int64,
                         device=_header_device())
        nr = torch.empty(1, dtype=torch.int64, device=_header_device())
        pending_sends.append(dist.isend(n, dst))
        dist.irecv(nr, recv_src).wait()

        # message order MUST match send()/send_async()/recv_from():
        # [count][h1][p1][h2][p2]... — the peer of one side of a fused
        # exchange may be a plain send/recv
        encoded = [_encode_item(t, dev) for t in send_tensors]
        for h, p in encoded:
            pending_sends.append(dist.isend(h, dst))
            pending_sends.append(dist.isend(p, dst))

        n_recv = int(nr.item())
        recv_out = []
        for _ in range(n_recv):
            h = torch.empty(2 + _MAX_DIMS, dtype=torch.int64,
                            device=_header_device())
            dist.irecv(h, recv_src).wait()
            hc = h.cpu()
            t = _recv_payload_shell(hc, dev)
            dist.irecv(t, recv_src).wait()
            recv_out.append(_decode_payload(hc, t))
        if defer_sends:
            refs = [n] + [h for h, _ in encoded] + [p for _, p in encoded]
            return recv_out, pending_sends, refs
        for r in pending_sends:
            r.wait()
    elif send_tensors is not None:
        send(send_tensors, dst)
    elif recv_src is not None:
        recv_out = recv_from(recv_src)
    if defer_sends:
        return recv_out, [], []
    return recv_out


def send_python_object(obj, dst: int):
    """Arbitrary python objects (reference pipeline/comm.py:114-211 metadata
    channel)."""
    data = pickle.dumps(obj)
    buf = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
    n = torch.tensor([buf.numel()], dtype=torch.int64, device=_header_device())
    dist.send(n, dst=dst)
    dist.send(buf.to(_header_device()), dst=dst)


def recv_python_object(src: int):
    n = torch.empty(1, dtype=torch.int64, device=_header_device())
    dist.recv(n, src=src)
    buf = torch.empty(int(n.item()), dtype=torch.uint8,
                      device=_header_device())
    dist.recv(buf, src=src)
    return pickle.loads(bytes(buf.cpu().numpy().tobytes()))
