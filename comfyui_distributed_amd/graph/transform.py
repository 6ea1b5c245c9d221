"""Prompt-graph transforms for distributed execution.

Behavioral parity with reference api/orchestration/prompt_transform.py:
worker pruning (:128-162), delegate-master preparation (:165-214),
per-node job ids (:217-225) and per-participant hidden-input overrides
(:228-352). Implemented fresh over :class:`PromptGraph`.
"""

from __future__ import annotations

import json
import secrets
import time
from typing import Iterable

from .prompt import (
    DISTRIBUTED_OUTPUT_CLASSES,
    NODE_CLASS_COLLECTOR,
    NODE_CLASS_EMPTY_IMAGE,
    NODE_CLASS_PREVIEW,
    NODE_CLASS_SEED,
    NODE_CLASS_UPSCALE,
    NODE_CLASS_VALUE,
    PromptGraph,
    is_link,
)


def generate_job_id_map(graph: PromptGraph, prefix: str | None = None) -> dict[str, str]:
    """One job id per distributed output node: ``exec_<ms>_<hex6>_<node_id>``
    (format parity: reference queue_orchestration.py:315-316)."""
    if prefix is None:
        prefix = f"exec_{int(time.time() * 1000)}_{secrets.token_hex(3)}"
    return {
        nid: f"{prefix}_{nid}"
        for nid in graph.nodes_of_class(*DISTRIBUTED_OUTPUT_CLASSES)
    }


def prune_prompt_for_worker(graph: PromptGraph) -> PromptGraph:
    """Worker graph = distributed nodes + their upstream closure.

    If nothing downstream of a distributed output node survives (its results
    used to flow onward into pruned nodes), a PreviewImage sink is appended
    so the executor still treats the graph as producing output
    (reference prompt_transform.py:128-162).
    """
    dist_nodes = graph.nodes_of_class(*DISTRIBUTED_OUTPUT_CLASSES)
    if not dist_nodes:
        return graph.copy()
    keep = graph.upstream_closure(dist_nodes)
    pruned = PromptGraph({nid: node for nid, node in graph.copy().items() if nid in keep})

    # Sink check: does any kept node consume a distributed node's output?
    consumed = set()
    for nid in pruned.node_ids():
        for _name, src in pruned.input_links(nid):
            consumed.add(src)
    for nid in dist_nodes:
        if nid not in consumed:
            sink_id = pruned.next_free_id()
            pruned.add_node(sink_id, {
                "class_type": NODE_CLASS_PREVIEW,
                "inputs": {"images": [nid, 0]},
            })
    return pruned


def prepare_delegate_master_prompt(
    graph: PromptGraph, collector_ids: Iterable[str]
) -> PromptGraph:
    """Delegate-only master graph = collectors + everything downstream.

    Upstream links of the collectors are replaced: the image input points at
    a zero-batch DistributedEmptyImage placeholder, other upstream links are
    dropped (reference prompt_transform.py:165-214).
    """
    collector_ids = [str(c) for c in collector_ids]
    keep = graph.downstream_closure(collector_ids)
    out = PromptGraph({nid: node for nid, node in graph.copy().items() if nid in keep})

    placeholder_id = out.next_free_id()
    placeholder_needed = False
    for cid in collector_ids:
        if cid not in out.raw:
            continue
        inputs = out.inputs(cid)
        for name in list(inputs.keys()):
            if not is_link(inputs[name]):
                continue
            src = str(inputs[name][0])
            if src in keep:
                continue
            if name == "images":
                inputs[name] = [placeholder_id, 0]
                placeholder_needed = True
            else:
                del inputs[name]
        inputs["delegate_only"] = True
    if placeholder_needed:
        out.add_node(placeholder_id, {
            "class_type": NODE_CLASS_EMPTY_IMAGE,
            "inputs": {"width": 64, "height": 64},
        })
    # Drop dangling links of downstream nodes whose sources were pruned.
    for nid in out.node_ids():
        inputs = out.inputs(nid)
        for name in list(inputs.keys()):
            if is_link(inputs[name]) and str(inputs[name][0]) not in out.raw:
                del inputs[name]
    return out


def apply_participant_overrides(
    graph: PromptGraph,
    *,
    is_master: bool,
    participant_id: str,
    enabled_worker_ids: list[str],
    job_id_map: dict[str, str],
    master_url: str = "",
) -> PromptGraph:
    """Fill the hidden inputs of every distributed node for one participant.

    Master: ``is_worker=False``, worker_id cleared. Worker: ``is_worker=True``,
    Seed/Value nodes get positional ``worker_<index>`` ids (index = position
    in ``enabled_worker_ids``; DistributedSeed adds index+1 to the seed),
    Collector/USDU nodes get the config worker id and the master callback URL
    (reference prompt_transform.py:228-352).
    """
    out = graph.copy()
    worker_index = {wid: i for i, wid in enumerate(str(w) for w in enabled_worker_ids)}
    enabled_json = json.dumps([str(w) for w in enabled_worker_ids])
    pid = str(participant_id)

    for nid in out.nodes_of_class(NODE_CLASS_SEED, NODE_CLASS_VALUE):
        inputs = out.inputs(nid)
        inputs["is_worker"] = not is_master
        inputs["worker_id"] = "" if is_master else f"worker_{worker_index.get(pid, 0)}"

    for nid in out.nodes_of_class(NODE_CLASS_COLLECTOR, NODE_CLASS_UPSCALE):
        inputs = out.inputs(nid)
        inputs["multi_job_id"] = job_id_map.get(nid, nid)
        inputs["is_worker"] = not is_master
        inputs["enabled_worker_ids"] = enabled_json
        if is_master:
            inputs.pop("worker_id", None)
            inputs.pop("master_url", None)
        else:
            inputs["worker_id"] = pid
            inputs["master_url"] = master_url
        # A collector fed by a USDU node is a pure pass-through: the USDU
        # master already holds the fully blended canvas
        # (reference prompt_transform.py:258-260).
        if out.class_of(nid) == NODE_CLASS_COLLECTOR and out.has_upstream(
            nid, NODE_CLASS_UPSCALE
        ):
            inputs["pass_through"] = True
    return out
