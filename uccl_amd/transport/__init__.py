"""Python surface of the software multipath reliable transport.

Reference parity: the collective/rdma transport redesigned for this
stack — chunking, N-path spraying, SACK selective repeat,
SACK-hole fast retransmit, RTO with abort threshold, CC mux
(Timely / Swift / receiver-paced EQDS / none), optional pacing,
deterministic loss injection.

    from uccl_amd.transport import TransportEndpoint
    a = TransportEndpoint(num_paths=8, chunk_bytes=8192)
    flow = a.connect(peer_metadata, tag=rank)   # or a.accept()
    a.send(flow, host_tensor); a.recv(flow, host_tensor)
    a.post_send(flow, t)   # async: keep t alive until a.flush(flow)
    a.flush(flow)          # wait until everything posted is acked
    st = a.stats()   # counters + cwnd/srtt + RTT p50/p99

Knobs (env; interface selection additionally honors the
NCCL_/RCCL_SOCKET_IFNAME aliases): UCCL_TP_CC,
UCCL_TP_EQDS_MBPS, UCCL_TP_CWND_MAX, UCCL_TP_RWND_KB, UCCL_TP_RTO_US,
UCCL_TP_RTO_ABORT, UCCL_TP_PACE_MBPS, UCCL_TP_LOSS_PCT,
UCCL_TP_ACK_LOSS_PCT, UCCL_TP_MAX_MSG_MB.
"""
from __future__ import annotations


def __getattr__(name):
    if name in ("TransportEndpoint", "TransportStats"):
        from uccl_amd import _load_native

        return getattr(_load_native(required=True), name)
    raise AttributeError(name)
