"""gloo world-2 worker: exercises the torch.distributed-integrated glue
that runs on CPU (dispatch layout via compat API, elastic world
detection, object transfer of tensors through p2p while dist is up)."""

import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)

    # compat layout helper under an initialized group
    from uccl_amd.ep.deep_ep_compat import Buffer

    topk = torch.tensor([[0, 3], [1, 2], [3, 0], [2, 1]])
    npr, _, npe, in_rank, _ = Buffer.get_dispatch_layout(topk, 4)
    assert npe.sum().item() == topk.numel()
    assert npr.shape[0] == world

    # elastic buffer world detection from the live group
    from uccl_amd.ep.elastic import ElasticBuffer

    eb = ElasticBuffer(None, num_experts=8, topk=2, hidden=16,
                       max_tokens=4, factory=lambda g, **cfg: object())
    assert eb.world == world, eb.world
    assert eb.expert_rank == [0, 0, 0, 0, 1, 1, 1, 1]

    # dist-coordinated p2p rendezvous: rank 0 serves, rank 1 connects,
    # metadata shipped over the gloo store via broadcast_object_list
    from uccl_amd import p2p

    ep = p2p.Endpoint(gpu=0, num_workers=1)
    md = [ep.metadata() if rank == 0 else None]
    dist.broadcast_object_list(md, src=0)
    if rank == 0:
        cid = ep.accept()
        got = p2p.recv_object(ep, cid)
        assert torch.equal(got["w"], torch.arange(1000).float())
        assert got["step"] == 7
    else:
        cid = ep.connect(md[0])
        p2p.send_object(ep, cid, {"w": torch.arange(1000).float(),
                                  "step": 7})
    dist.barrier()
    print(f"RANK{rank} OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
