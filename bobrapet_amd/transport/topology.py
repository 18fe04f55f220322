"""Topology analysis: direct P2P edges vs hub routing.

Role parity with the reference's TopologyAnalyzer
(reference: pkg/transport/topology.go:46-173 and routing.go:26
StepNeedsHubRouting): a streaming step whose `runtime` block carries
per-packet templates must route through the hub stage (the engine-side
evaluator), everything else flows engram→engram directly.  On MI355X the
"hub" is an in-process routing stage and P2P edges are same-GPU stream
queues or RCCL send/recv over xGMI.
"""
from __future__ import annotations

import typing as _t
from dataclasses import dataclass, field

from ..specs import types as T


@dataclass
class Edge:
    src: str  # step name ("" = ingress)
    dst: str
    mode: str = "p2p"  # p2p | hub
    lane: str = "data"
    cross_rank: bool = False


@dataclass
class Topology:
    stages: _t.List[str] = field(default_factory=list)  # topological order
    edges: _t.List[Edge] = field(default_factory=list)
    hub_steps: _t.Set[str] = field(default_factory=set)

    def downstream_of(self, step: str) -> _t.List[Edge]:
        return [e for e in self.edges if e.src == step]

    def upstream_of(self, step: str) -> _t.List[Edge]:
        return [e for e in self.edges if e.dst == step]


def step_needs_hub_routing(step: T.Step) -> bool:
    """Per-packet templates force hub routing (reference: routing.go:26)."""
    return step.runtime is not None


def analyze(story: T.Story) -> Topology:
    """Build the streaming topology from the story's DAG.

    Steps with no upstream engram feed from the ingress; `needs` edges
    become packet edges.  Hub-routed steps are marked so the runtime
    evaluates their `runtime` templates per packet."""
    from ..engine.dag import compile_story

    cs = compile_story(story)
    topo = Topology()
    # topological order via Kahn over the main steps
    indeg = {s.name: len([d for d in cs.deps.get(s.name, ()) ]) for s in story.steps}
    ready = sorted([n for n, d in indeg.items() if d == 0])
    order: _t.List[str] = []
    while ready:
        cur = ready.pop(0)
        order.append(cur)
        for nxt in sorted(cs.dependents.get(cur, ())):
            indeg[nxt] -= 1
            if indeg[nxt] == 0:
                ready.append(nxt)
    topo.stages = order

    by_name = {s.name: s for s in story.steps}
    for name in order:
        step = by_name[name]
        if step_needs_hub_routing(step):
            topo.hub_steps.add(name)
        ups = sorted(cs.deps.get(name, ()))
        if not ups:
            topo.edges.append(Edge(src="", dst=name, lane=_lane_of(step)))
        for up in ups:
            mode = "hub" if (name in topo.hub_steps or up in topo.hub_steps) else "p2p"
            topo.edges.append(Edge(src=up, dst=name, mode=mode, lane=_lane_of(step)))
    return topo


def _lane_of(step: T.Step) -> str:
    if isinstance(step.with_, dict) and step.with_.get("lane"):
        return str(step.with_["lane"])
    return "data"
