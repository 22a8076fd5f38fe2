"""Layer-graph IR shared by the profiler and the partitioner.

Our rebuild of the reference's graph library
(/root/reference/pipedream-fork/graph/graph.py: Node with fwd/bwd
compute_time, activation_size, parameter_size, stage_id; topological sort,
predecessors/successors, antichain DAG, partition_graph, text
serialization — SURVEY.md §2.7). The algorithms are re-implemented from
the problem statement, not translated; the text format is ours (one
node/edge per line, round-trippable).
"""

from __future__ import annotations

import json
from dataclasses import dataclass
from typing import Dict, List, Optional, Set, Tuple


@dataclass
class Node:
    node_id: int
    desc: str = ""
    fwd_time: float = 0.0          # seconds
    bwd_time: float = 0.0          # seconds
    activation_size: float = 0.0   # bytes (output)
    parameter_size: float = 0.0    # bytes
    stage_id: int = -1

    @property
    def compute_time(self) -> float:
        return self.fwd_time + self.bwd_time


class Graph:
    def __init__(self) -> None:
        self.nodes: Dict[int, Node] = {}
        self.edges: Dict[int, List[int]] = {}      # src -> [dst]
        self.in_edges: Dict[int, List[int]] = {}   # dst -> [src]

    # ---- construction --------------------------------------------------
    def add_node(self, node: Node) -> Node:
        self.nodes[node.node_id] = node
        self.edges.setdefault(node.node_id, [])
        self.in_edges.setdefault(node.node_id, [])
        return node

    def add_edge(self, src: int, dst: int) -> None:
        if dst not in self.edges[src]:
            self.edges[src].append(dst)
            self.in_edges[dst].append(src)

    @classmethod
    def chain(cls, nodes: List[Node]) -> "Graph":
        g = cls()
        for n in nodes:
            g.add_node(n)
        for a, b in zip(nodes, nodes[1:]):
            g.add_edge(a.node_id, b.node_id)
        return g

    # ---- queries -------------------------------------------------------
    def sources(self) -> List[Node]:
        return [self.nodes[i] for i in self.nodes if not self.in_edges[i]]

    def sinks(self) -> List[Node]:
        return [self.nodes[i] for i in self.nodes if not self.edges[i]]

    def topological_sort(self) -> List[Node]:
        indeg = {i: len(self.in_edges[i]) for i in self.nodes}
        ready = sorted(i for i, d in indeg.items() if d == 0)
        order = []
        while ready:
            i = ready.pop(0)
            order.append(self.nodes[i])
            for j in self.edges[i]:
                indeg[j] -= 1
                if indeg[j] == 0:
                    ready.append(j)
            ready.sort()
        if len(order) != len(self.nodes):
            raise ValueError("graph has a cycle")
        return order

    def predecessors(self, node_id: int) -> Set[int]:
        """All transitive predecessors (memoization-free DFS)."""
        seen: Set[int] = set()
        stack = list(self.in_edges[node_id])
        while stack:
            i = stack.pop()
            if i not in seen:
                seen.add(i)
                stack.extend(self.in_edges[i])
        return seen

    def successors(self, node_id: int) -> Set[int]:
        seen: Set[int] = set()
        stack = list(self.edges[node_id])
        while stack:
            i = stack.pop()
            if i not in seen:
                seen.add(i)
                stack.extend(self.edges[i])
        return seen

    def is_chain(self) -> bool:
        return all(len(v) <= 1 for v in self.edges.values()) and \
            all(len(v) <= 1 for v in self.in_edges.values())

    # ---- antichains (partition frontiers for non-chain DAGs) -----------
    def antichain_dag(self) -> Tuple[List[frozenset], Dict[frozenset,
                                                           List[frozenset]]]:
        """Enumerate the cut frontiers of the DAG.

        An antichain here is a minimal set of nodes whose removal (with
        all their predecessors) splits the graph — the candidate pipeline
        split points (the reference enumerates the same states,
        graph.py:420-449). Returns (states in topo order of discovery,
        adjacency). Exponential in width; our model graphs are chains or
        near-chains so this stays tiny."""
        start = frozenset(n.node_id for n in self.sources())
        states: List[frozenset] = [start]
        adj: Dict[frozenset, List[frozenset]] = {start: []}
        work = [start]
        while work:
            ac = work.pop(0)
            # set of nodes "covered" = antichain + its predecessors
            covered: Set[int] = set(ac)
            for i in ac:
                covered |= self.predecessors(i)
            # candidates: any uncovered node whose every parent is covered
            candidates = {j for j in self.nodes
                          if j not in covered and self.in_edges[j]
                          and all(p in covered for p in self.in_edges[j])}
            for j in candidates:
                nxt = set(ac) | {j}
                # keep only maximal members (drop predecessors of others)
                nxt_f = frozenset(
                    m for m in nxt
                    if not any(m in self.predecessors(o)
                               for o in nxt if o != m))
                if nxt_f not in adj:
                    adj[nxt_f] = []
                    states.append(nxt_f)
                    work.append(nxt_f)
                if nxt_f not in adj[ac]:
                    adj[ac].append(nxt_f)
        return states, adj

    # ---- partitioning ---------------------------------------------------
    def partition_by_stage(self) -> Dict[int, "Graph"]:
        """Split into per-stage subgraphs by node.stage_id
        (reference partition_graph, graph.py:117-137)."""
        out: Dict[int, Graph] = {}
        for n in self.nodes.values():
            g = out.setdefault(n.stage_id, Graph())
            g.add_node(n)
        for src, dsts in self.edges.items():
            for dst in dsts:
                s = self.nodes[src].stage_id
                if s == self.nodes[dst].stage_id:
                    out[s].add_edge(src, dst)
        return out

    def stage_boundaries(self) -> List[Tuple[int, int]]:
        """(src, dst) edges that cross stages, in topo order."""
        res = []
        for n in self.topological_sort():
            for dst in self.edges[n.node_id]:
                if self.nodes[dst].stage_id != n.stage_id:
                    res.append((n.node_id, dst))
        return res

    def to_dot(self) -> str:
        """Graphviz text (the reference's to_dot, graph.py:482-616) —
        node label = desc + compute ms, cluster color by stage."""
        lines = ["digraph layers {", "  rankdir=TB;"]
        for n in self.topological_sort():
            ms = n.compute_time * 1e3
            label = f"{n.node_id}: {n.desc}\\n{ms:.2f} ms"
            color = f"/set312/{(n.stage_id % 12) + 1}" if n.stage_id >= 0 \
                else "white"
            lines.append(
                f'  n{n.node_id} [label="{label}", style=filled, '
                f'fillcolor="{color}"];')
        for src in sorted(self.edges):
            for dst in self.edges[src]:
                lines.append(f"  n{src} -> n{dst};")
        lines.append("}")
        return "\n".join(lines) + "\n"

    def plot_cdf(self, path: str) -> None:
        """CDF of cumulative compute time over layers in topo order (the
        reference's plot_cdfs, graph.py:482-616). PNG/PDF by extension."""
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        nodes = self.topological_sort()
        total = sum(n.compute_time for n in nodes) or 1.0
        xs = list(range(1, len(nodes) + 1))
        acc, ys = 0.0, []
        for n in nodes:
            acc += n.compute_time
            ys.append(acc / total)
        fig, ax = plt.subplots(figsize=(6, 4))
        ax.step(xs, ys, where="post")
        ax.set_xlabel("layer (topological order)")
        ax.set_ylabel("cumulative compute time fraction")
        ax.set_ylim(0, 1.02)
        ax.grid(True, alpha=0.3)
        fig.tight_layout()
        fig.savefig(path)
        plt.close(fig)

    def plot_bars(self, path: str) -> None:
        """Per-layer fwd/bwd stacked time bars (the reference's
        plot_bar_graph, graph.py:482-616)."""
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        nodes = self.topological_sort()
        xs = list(range(len(nodes)))
        fwd = [n.fwd_time * 1e3 for n in nodes]
        bwd = [n.bwd_time * 1e3 for n in nodes]
        fig, ax = plt.subplots(figsize=(max(6, len(nodes) * 0.12), 4))
        ax.bar(xs, fwd, label="forward")
        ax.bar(xs, bwd, bottom=fwd, label="backward")
        ax.set_xlabel("layer (topological order)")
        ax.set_ylabel("time (ms)")
        ax.legend()
        fig.tight_layout()
        fig.savefig(path)
        plt.close(fig)

    # ---- serialization (our own line format) ----------------------------
    def dumps(self) -> str:
        lines = []
        for n in self.topological_sort():
            lines.append("node " + json.dumps({
                "id": n.node_id, "desc": n.desc, "fwd_time": n.fwd_time,
                "bwd_time": n.bwd_time, "activation_size": n.activation_size,
                "parameter_size": n.parameter_size, "stage_id": n.stage_id}))
        for src in sorted(self.edges):
            for dst in self.edges[src]:
                lines.append(f"edge {src} {dst}")
        return "\n".join(lines) + "\n"

    @classmethod
    def loads(cls, text: str) -> "Graph":
        g = cls()
        edges = []
        for line in text.splitlines():
            line = line.strip()
            if line.startswith("node "):
                d = json.loads(line[5:])
                g.add_node(Node(d["id"], d["desc"], d["fwd_time"],
                                d["bwd_time"], d["activation_size"],
                                d["parameter_size"], d["stage_id"]))
            elif line.startswith("edge "):
                _, s, t = line.split()
                edges.append((int(s), int(t)))
        for s, t in edges:
            g.add_edge(s, t)
        return g

    def save(self, path: str) -> None:
        with open(path, "w") as f:
            f.write(self.dumps())

    @classmethod
    def load(cls, path: str) -> "Graph":
        with open(path) as f:
            return cls.loads(f.read())


def compress_branches(g: "Graph"):
    """Collapse fork..join regions into super-nodes.

    The reference compresses Inception-style parallel branches so the
    antichain enumeration stays tractable
    (/root/reference/pipedream-fork/graph/graph.py:139-227,
    optimizer/scripts/compress_graph_branches.py). Same purpose here:
    for every fork node f whose join j is the unique first common
    descendant of all branches, replace the region (f..j exclusive of
    f, inclusive of j) with ONE node carrying the summed compute/param
    sizes and j's activation size. Applied repeatedly until no region
    is found. Returns (compressed_graph, mapping super_node_id ->
    [original node ids])."""
    mapping = {i: [i] for i in g.nodes}
    cur = g
    while True:
        region = _find_branch_region(cur)
        if region is None:
            return cur, mapping
        f, j, members = region
        new = Graph()
        # keep original ids; the super node takes j's id
        super_members: List[int] = sorted(members | {j})
        for i, n in cur.nodes.items():
            if i in members:
                continue
            if i == j:
                new.add_node(Node(
                    j, desc=f"super[{len(super_members)}]",
                    fwd_time=sum(cur.nodes[m].fwd_time
                                 for m in super_members),
                    bwd_time=sum(cur.nodes[m].bwd_time
                                 for m in super_members),
                    activation_size=cur.nodes[j].activation_size,
                    parameter_size=sum(cur.nodes[m].parameter_size
                                       for m in super_members)))
            else:
                new.add_node(Node(i, desc=n.desc, fwd_time=n.fwd_time,
                                  bwd_time=n.bwd_time,
                                  activation_size=n.activation_size,
                                  parameter_size=n.parameter_size))
        for src, dsts in cur.edges.items():
            for dst in dsts:
                s = j if src in members else src
                d = j if dst in members else dst
                if s == d or (s in members) or (d in members):
                    continue
                if d not in new.edges[s]:
                    new.add_edge(s, d)
        # merge member mappings into the super node's entry
        merged = []
        for m in super_members:
            merged.extend(mapping.pop(m, [m]))
        mapping[j] = merged
        cur = new


def _find_branch_region(g: "Graph"):
    """First (fork f, join j, interior members) region where every path
    from f re-converges at j and the interior has no edges leaving the
    region. Interior = nodes strictly between f and j."""
    for f in (n.node_id for n in g.topological_sort()):
        outs = g.edges[f]
        if len(outs) < 2:
            continue
        # candidate joins: common successors (incl. direct) of all outs
        succ_sets = []
        for o in outs:
            s = g.successors(o) | {o}
            succ_sets.append(s)
        common = set.intersection(*succ_sets)
        if not common:
            continue
        # earliest common node in topo order
        topo = [n.node_id for n in g.topological_sort()]
        pos = {i: k for k, i in enumerate(topo)}
        j = min(common, key=lambda i: pos[i])
        if j == f:
            continue
        # interior = nodes on paths f -> j (descendants of f that are
        # ancestors of j), excluding f and j
        desc_f = g.successors(f)
        anc_j = g.predecessors(j)
        members = (desc_f & anc_j) - {f, j}
        if not members:
            continue
        # validity: no interior node has an edge to outside (other than
        # toward j/members), and nothing outside (other than f) feeds
        # the interior
        ok = True
        for m in members:
            if any(d not in members and d != j for d in g.edges[m]):
                ok = False
                break
            if any(s not in members and s != f for s in g.in_edges[m]):
                ok = False
                break
        # the join must be fed only from the interior or f
        if ok and any(s not in members and s != f
                      for s in g.in_edges[j]):
            ok = False
        if ok:
            return f, j, members
    return None
