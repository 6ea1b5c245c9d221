"""Workflow graph ("prompt") representation.

The wire format is ComfyUI's prompt JSON — ``{node_id: {"class_type": str,
"inputs": {name: literal | [src_node_id, output_index]}}}`` — kept verbatim
for API compatibility with the reference (its ``/distributed/queue`` accepts
exactly this shape; reference api/queue_request.py:5-79). This module gives
it a typed wrapper with the link/closure queries the orchestrator needs
(reference builds the same indexes in
api/orchestration/prompt_transform.py:7-53).
"""

from __future__ import annotations

import copy
from typing import Any, Iterator

NODE_CLASS_SEED = "DistributedSeed"
NODE_CLASS_VALUE = "DistributedValue"
NODE_CLASS_MODEL_NAME = "DistributedModelName"
NODE_CLASS_COLLECTOR = "DistributedCollector"
NODE_CLASS_UPSCALE = "UltimateSDUpscaleDistributed"
NODE_CLASS_EMPTY_IMAGE = "DistributedEmptyImage"
NODE_CLASS_PREVIEW = "PreviewImage"

#: Node classes whose presence makes a workflow "distributed".
DISTRIBUTED_OUTPUT_CLASSES = (NODE_CLASS_COLLECTOR, NODE_CLASS_UPSCALE)


def is_link(value: Any) -> bool:
    """True for the ``[node_id, output_index]`` input encoding."""
    return (
        isinstance(value, (list, tuple))
        and len(value) == 2
        and isinstance(value[0], (str, int))
        and isinstance(value[1], int)
    )


class PromptGraph:
    """Wrapper over a prompt dict with cached structural queries."""

    def __init__(self, prompt: dict[str, Any]):
        self.raw = prompt
        self._by_class: dict[str, list[str]] | None = None

    # -- basics ------------------------------------------------------------

    def copy(self) -> "PromptGraph":
        return PromptGraph(copy.deepcopy(self.raw))

    def node_ids(self) -> list[str]:
        return list(self.raw.keys())

    def node(self, node_id: str) -> dict:
        return self.raw[str(node_id)]

    def class_of(self, node_id: str) -> str:
        return self.raw[str(node_id)].get("class_type", "")

    def inputs(self, node_id: str) -> dict:
        return self.raw[str(node_id)].setdefault("inputs", {})

    def items(self) -> Iterator[tuple[str, dict]]:
        return iter(self.raw.items())

    # -- indexes -----------------------------------------------------------

    def nodes_of_class(self, *class_names: str) -> list[str]:
        if self._by_class is None:
            index: dict[str, list[str]] = {}
            for nid, node in self.raw.items():
                index.setdefault(node.get("class_type", ""), []).append(nid)
            self._by_class = index
        out: list[str] = []
        for name in class_names:
            out.extend(self._by_class.get(name, []))
        return out

    def input_links(self, node_id: str) -> list[tuple[str, str]]:
        """(input_name, src_node_id) pairs for linked inputs of a node."""
        out = []
        for name, value in self.raw[str(node_id)].get("inputs", {}).items():
            if is_link(value):
                out.append((name, str(value[0])))
        return out

    def upstream_closure(self, start_ids: list[str]) -> set[str]:
        """All node ids reachable backwards from ``start_ids`` (inclusive)."""
        seen: set[str] = set()
        stack = [str(s) for s in start_ids]
        while stack:
            nid = stack.pop()
            if nid in seen or nid not in self.raw:
                continue
            seen.add(nid)
            for _name, src in self.input_links(nid):
                if src not in seen:
                    stack.append(src)
        return seen

    def downstream_closure(self, start_ids: list[str]) -> set[str]:
        """All node ids reachable forwards from ``start_ids`` (inclusive)."""
        consumers: dict[str, set[str]] = {}
        for nid in self.raw:
            for _name, src in self.input_links(nid):
                consumers.setdefault(src, set()).add(nid)
        seen: set[str] = set()
        stack = [str(s) for s in start_ids]
        while stack:
            nid = stack.pop()
            if nid in seen or nid not in self.raw:
                continue
            seen.add(nid)
            for consumer in consumers.get(nid, ()):
                if consumer not in seen:
                    stack.append(consumer)
        return seen

    def has_upstream(self, node_id: str, target_class: str) -> bool:
        """True when a node of ``target_class`` feeds (transitively) into
        ``node_id`` (reference uses this to flag collectors directly
        downstream of a USDU node, prompt_transform.py:30-53)."""
        closure = self.upstream_closure([node_id])
        closure.discard(str(node_id))
        return any(self.class_of(nid) == target_class for nid in closure)

    def add_node(self, node_id: str, node: dict) -> None:
        """Insert a node and invalidate the class index (direct ``raw``
        writes after a ``nodes_of_class`` call would leave it stale)."""
        self.raw[str(node_id)] = node
        self._by_class = None

    def next_free_id(self) -> str:
        """Smallest unused positive numeric id (prompt ids are numeric
        strings in the ComfyUI wire format)."""
        used = set()
        for nid in self.raw:
            try:
                used.add(int(nid))
            except ValueError:
                pass
        candidate = max(used, default=0) + 1
        return str(candidate)
