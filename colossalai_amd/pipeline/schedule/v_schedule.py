"""ZB-V schedule graph (reference: colossalai/pipeline/schedule/v_schedule.py:46
PipelineGraph.get_v_schedule — re-derived).

The V placement gives rank r virtual stages ``r`` (descending arm) and
``2·pp−1−r`` (ascending arm), so the pipeline turns around at rank pp−1 and
the LAST virtual stage lands back on rank 0 — embeddings, LM head and the
loss are colocated, and every rank holds exactly two chunks.

The reference builds its node list with a cost-weighted heuristic search;
here the node list comes from deterministic list scheduling of the exact
dependency graph under uniform F/B/W durations (the MI355X bench regime —
equal-sized decoder slices): every rank advances the oldest ready B, else
the oldest ready F within the activation bound, and W batches fill the
remaining slots. The simulation also fixes, per directed rank pair, the
ORDER messages cross the wire — consumers buffer out-of-order arrivals so
the untagged P2P channels stay consistent (the reference orders comm the
same way via its communication lists).
"""

from dataclasses import dataclass
from typing import Dict, List, Tuple

__all__ = ["ScheduledNode", "build_zbv_schedule", "owner_of_vstage"]


@dataclass(frozen=True)
class ScheduledNode:
    type: str        # "F" | "B" | "W"
    vstage: int      # 0 .. 2*pp-1
    micro: int


def owner_of_vstage(v: int, pp: int) -> int:
    return v if v < pp else 2 * pp - 1 - v


def build_zbv_schedule(pp: int, n_micro: int, max_live: int = None):
    """-> (per_rank_nodes, channel_orders)

    per_rank_nodes[r]: ordered ScheduledNode list for rank r.
    channel_orders[(src, dst)]: the wire order of (type, vstage, micro)
    messages from src to dst (both endpoints derive recv/send order from it).
    """
    V = 2 * pp
    if max_live is None:
        max_live = V  # ZB-V keeps ~2*pp in-flight activations (1F1B-equal memory)

    def fkey(v, m):
        return ("F", v, m)

    def bkey(v, m):
        return ("B", v, m)

    # dependency map
    deps: Dict[Tuple, List[Tuple]] = {}
    for m in range(n_micro):
        for v in range(V):
            deps[fkey(v, m)] = [fkey(v - 1, m)] if v > 0 else []
            deps[bkey(v, m)] = [bkey(v + 1, m)] if v < V - 1 else [fkey(V - 1, m)]
            deps[("W", v, m)] = [bkey(v, m)]

    done: Dict[Tuple, int] = {}          # node -> completion time
    per_rank: List[List[ScheduledNode]] = [[] for _ in range(pp)]
    live = [0] * pp                      # outstanding activations per rank
    rank_free = [0] * pp                 # next free time slot per rank
    n_nodes = len(deps)
    sends: List[Tuple[int, int, int, Tuple]] = []  # (time, src, dst, key)

    t = 0
    while len(done) < n_nodes:
        progressed = False
        for r in range(pp):
            if rank_free[r] > t:
                continue
            my_vs = [r, 2 * pp - 1 - r]

            def ready(key):
                return key not in done and all(d in done and done[d] <= t for d in deps[key])

            # oldest ready B first (critical path), then F under the memory
            # bound, then one W batch
            pick = None
            for m in range(n_micro):
                for v in my_vs:
                    if ready(bkey(v, m)):
                        pick = bkey(v, m)
                        break
                if pick:
                    break
            if pick is None and live[r] < max_live:
                for m in range(n_micro):
                    for v in my_vs:
                        if ready(fkey(v, m)):
                            pick = fkey(v, m)
                            break
                    if pick:
                        break
            if pick is None:
                for m in range(n_micro):
                    for v in my_vs:
                        if ready(("W", v, m)):
                            pick = ("W", v, m)
                            break
                    if pick:
                        break
            if pick is None:
                continue
            typ, v, m = pick
            done[pick] = t + 1
            rank_free[r] = t + 1
            per_rank[r].append(ScheduledNode(typ, v, m))
            progressed = True
            if typ == "F":
                live[r] += 1
                if v < V - 1 and owner_of_vstage(v + 1, pp) != r:
                    sends.append((t + 1, r, owner_of_vstage(v + 1, pp), pick))
            elif typ == "B":
                live[r] -= 1
                if v > 0 and owner_of_vstage(v - 1, pp) != r:
                    sends.append((t + 1, r, owner_of_vstage(v - 1, pp), pick))
        t += 1

    channel_orders: Dict[Tuple[int, int], List[Tuple]] = {}
    for tm, src, dst, key in sorted(sends, key=lambda x: x[0]):
        channel_orders.setdefault((src, dst), []).append(key)
    return per_rank, channel_orders
