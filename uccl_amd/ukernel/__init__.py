"""Chunk-graph collective planner + spray executor.

The uccl_amd analog of the reference's experimental "ukernel" layer
(reference: experimental/ukernel/src/ccl/ — chunk-DAG planner, lowering,
SprayExecutor over pluggable backends; tested there with mock backends in
a single process, as here). Re-designed for the xGMI fullmesh: planning is
path selection (direct link vs 1-hop relays) rather than ring
construction, and cross-rank ordering lowers to explicit signal/wait
tasks — the same flag-round discipline as the collective engine.

    from uccl_amd import ukernel as uk
    topo = uk.Topology(8)
    g = uk.lower(uk.plan_allreduce_rsag(topo, nbytes=1 << 20))
    outs, stats = uk.execute_host(g, inputs)   # host mock backend

The host executor validates plans end-to-end on CPU; the production data
path stays with the hand-tuned HIP kernels in csrc/collective (see
docs/DESIGN.md for why a static plan beats a DAG interpreter on a
single-hop fabric for the standard collectives — this layer is for
irregular/sprayed transfers and plan experimentation).
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from uccl_amd import _load_native

_C = _load_native()

Topology = _C.UkTopology
Graph = _C.UkGraph


def plan_sendrecv(topo, src: int, dst: int, nbytes: int,
                  chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_sendrecv(topo, src, dst, nbytes, chunk_bytes)


def plan_allreduce_rsag(topo, nbytes: int, elem_bytes: int = 4,
                        chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_allreduce_rsag(topo, nbytes, elem_bytes, chunk_bytes)


def plan_allreduce_oneshot(topo, nbytes: int, elem_bytes: int = 4) -> Graph:
    return _C.uk_plan_allreduce_oneshot(topo, nbytes, elem_bytes)


def plan_broadcast(topo, root: int, nbytes: int,
                   chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_broadcast(topo, root, nbytes, chunk_bytes)


def plan_allgather(topo, nbytes: int, chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_allgather(topo, nbytes, chunk_bytes)


def plan_reducescatter(topo, shard_bytes: int, elem_bytes: int = 4,
                       chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_reducescatter(topo, shard_bytes, elem_bytes,
                                    chunk_bytes)


def plan_alltoall(topo, seg_bytes: int, chunk_bytes: int = 1 << 20) -> Graph:
    return _C.uk_plan_alltoall(topo, seg_bytes, chunk_bytes)


def lower(graph: Graph) -> Graph:
    return _C.uk_lower(graph)


def estimate_us(graph: Graph, topo, link_gbps: float = 150.0,
                local_gbps: float = 1500.0, overhead_us: float = 4.0):
    """List-scheduling completion-time estimate: per-link serialization
    at link_gbps x weight, per-rank local ops at local_gbps."""
    return _C.uk_estimate_us(graph, topo, link_gbps, local_gbps,
                             overhead_us)


def plan_allreduce_auto(topo, nbytes: int, elem_bytes: int = 4,
                        chunk_bytes: int = 1 << 20) -> Graph:
    """Pick one-shot vs RS+AG by estimated time; returns a LOWERED
    graph ready for execute_host."""
    return _C.uk_plan_allreduce_auto(topo, nbytes, elem_bytes, chunk_bytes)


def execute_host(graph: Graph, inputs: List[torch.Tensor],
                 out_bytes: int | None = None) -> Tuple[list, dict]:
    """Run a LOWERED graph on the host mock backend (one worker thread per
    rank, fp32 buffers). Returns (per-rank outputs, stats) where stats has
    tasks_run / wait_requeues / link_bytes[world*world]."""
    if out_bytes is None:
        out_bytes = inputs[0].numel() * 4
    return _C.uk_execute_host(graph, list(inputs), out_bytes)


def link_matrix(stats: dict, world: int):
    """stats['link_bytes'] as a world x world nested list."""
    lb = stats["link_bytes"]
    return [[lb[s * world + d] for d in range(world)] for s in range(world)]


def to_dot(graph: Graph) -> str:
    """Graphviz text for a (raw or lowered) plan — one node per task,
    colored by op, edges = dependencies. Render with `dot -Tsvg`."""
    colors = {"copy": "lightblue", "reduce": "orange", "put": "palegreen",
              "signal": "gray80", "wait": "gray60"}
    lines = ["digraph plan {", "  rankdir=LR;", "  node [shape=box];"]
    for ln in graph.dump().splitlines():
        idx, rest = ln.split(":", 1)
        op = rest.strip().split()[0]
        lines.append(
            f'  t{idx} [label="{idx}: {rest.strip()[:48]}" '
            f'style=filled fillcolor={colors.get(op, "white")}];')
        if "deps[" in rest:
            deps = rest.rsplit("deps[", 1)[1].rstrip("]").split(",")
            for d in deps:
                if d:
                    lines.append(f"  t{d.strip()} -> t{idx};")
    lines.append("}")
    return "\n".join(lines)
