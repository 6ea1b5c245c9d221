"""Workflow graph executor.

The reference relies on ComfyUI's executor (L0 in SURVEY.md §1); this
framework executes prompt graphs itself: topological evaluation of the
``{node_id: {class_type, inputs}}`` wire format against a node registry
(the distributed nodes from nodes/ plus the built-in compute nodes below).
"""

from __future__ import annotations

import itertools
from typing import Any

from ..utils.errors import PromptValidationError
from .prompt import PromptGraph, is_link

#: sentinel for "this node must never be cross-run cached" (IS_CHANGED nan)
_VOLATILE = object()
#: monotone source for volatile fingerprints (never repeats across runs)
_volatile_seq = itertools.count()


class NodeRegistry:
    def __init__(self):
        self._classes: dict[str, type] = {}

    def register(self, name: str, cls: type) -> None:
        self._classes[name] = cls

    def register_map(self, mapping: dict[str, type]) -> None:
        self._classes.update(mapping)

    def get(self, name: str) -> type | None:
        return self._classes.get(name)

    def names(self) -> list[str]:
        return list(self._classes)


def default_registry() -> NodeRegistry:
    from ..nodes import NODE_CLASS_MAPPINGS
    from . import builtin_nodes

    reg = NodeRegistry()
    reg.register_map(NODE_CLASS_MAPPINGS)
    reg.register_map(builtin_nodes.BUILTIN_CLASS_MAPPINGS)
    return reg


def validate_prompt(prompt: dict, registry: NodeRegistry) -> None:
    """Structural validation: known classes, resolvable links, acyclic."""
    errors: dict[str, list[str]] = {}
    graph = PromptGraph(prompt)
    for nid, node in graph.items():
        cls_name = node.get("class_type", "")
        cls = registry.get(cls_name)
        if cls is None:
            errors.setdefault(nid, []).append(f"unknown node class {cls_name!r}")
            continue
        inputs = node.get("inputs", {})
        for name, value in inputs.items():
            if not is_link(value):
                continue
            src = str(value[0])
            if src not in prompt:
                errors.setdefault(nid, []).append(
                    f"input {name!r} links to missing node {value[0]!r}"
                )
                continue
            src_cls = registry.get(prompt[src].get("class_type", ""))
            rts = getattr(src_cls, "RETURN_TYPES", None)
            # ByPassTypeTuple marks variable-output nodes (batch dividers):
            # their socket count follows a widget, so no bounds check
            # (same convention as the reference nodes/utilities.py:252)
            if type(rts).__name__ == "ByPassTypeTuple":
                continue
            if rts is not None and int(value[1]) >= len(rts):
                errors.setdefault(nid, []).append(
                    f"input {name!r} links to output {value[1]} of node "
                    f"{src!r} which has only {len(rts)} output(s)"
                )
        # required inputs without a declared default must be present
        # (ComfyUI validates this at queue time; reference workflows rely
        # on the structured node_errors response)
        try:
            spec = cls.INPUT_TYPES().get("required", {})
        except Exception:  # noqa: BLE001 - INPUT_TYPES needing context
            spec = {}
        for name, decl in spec.items():
            has_default = (
                isinstance(decl, (tuple, list)) and len(decl) > 1
                and isinstance(decl[1], dict) and "default" in decl[1]
            )
            if name not in inputs and not has_default:
                errors.setdefault(nid, []).append(
                    f"required input {name!r} missing"
                )
    # cycle check via DFS
    state: dict[str, int] = {}

    def visit(nid: str):
        if state.get(nid) == 1:
            errors.setdefault(nid, []).append("cycle detected")
            return
        if state.get(nid) == 2:
            return
        state[nid] = 1
        for _n, src in graph.input_links(nid):
            if src in prompt:
                visit(src)
        state[nid] = 2

    for nid in prompt:
        visit(nid)
    if errors:
        raise PromptValidationError("prompt validation failed", errors)


class Executor:
    """Evaluates a prompt graph; caches per-node outputs within one run AND
    across consecutive runs (ComfyUI-style execution caching, which the
    reference inherits from its host: a re-submitted workflow only
    re-executes nodes whose input spec changed — see ComfyUI
    execution.py's caching the reference's nodes rely on).

    Cross-run reuse is keyed by a recursive spec fingerprint
    (class_type + widget values + upstream fingerprints), never by tensor
    contents, so it costs O(graph) per run. Output/side-effect nodes
    (OUTPUT_NODE = True: SaveImage, PreviewImage, collectors, USDU) always
    re-run — their upstream work is still reused."""

    def __init__(self, registry: NodeRegistry | None = None,
                 context: dict[str, Any] | None = None):
        self.registry = registry or default_registry()
        #: shared objects node bodies may need (device, model cache, ...)
        self.context = context or {}
        #: previous run's {fingerprint: outputs}; replaced wholesale each
        #: run so memory stays bounded by one workflow's outputs
        self._run_cache: dict[str, tuple] = {}

    def _is_changed_mark(self, class_type: str, widgets: dict):
        """Evaluate a node class's IS_CHANGED hook over its widget values.

        Returns ``None`` (no hook), ``_VOLATILE`` (nan / failing hook =>
        always re-execute, ComfyUI parity), or a hashable content mark to
        fold into the fingerprint (e.g. a file's mtime+size)."""
        cls = self.registry.get(class_type)
        if cls is None or not hasattr(cls, "IS_CHANGED"):
            return None
        try:
            fn = cls.IS_CHANGED
            try:
                mark = fn(context=self.context, **widgets)
            except TypeError:
                mark = fn(**widgets)
        except Exception:
            return _VOLATILE
        if isinstance(mark, float) and mark != mark:  # nan
            return _VOLATILE
        return mark

    def _fingerprints(self, graph: PromptGraph) -> dict[str, str]:
        import hashlib
        import json

        fps: dict[str, str] = {}

        def fp(nid: str) -> str:
            if nid in fps:
                return fps[nid]
            node = graph.node(nid)
            widgets = {}
            links = {}
            for name, value in node.get("inputs", {}).items():
                if is_link(value):
                    links[name] = (fp(str(value[0])), int(value[1]))
                else:
                    widgets[name] = value
            # ComfyUI's IS_CHANGED convention: the class may report a
            # content mark (e.g. LoadImage's file mtime/size) that is folded
            # into the fingerprint so a re-uploaded file under the same name
            # busts the cross-run cache; a NaN (or a failing hook) means
            # "always changed" and poisons this node AND its downstream.
            mark = self._is_changed_mark(node["class_type"], widgets)
            if mark is _VOLATILE:
                fps[nid] = f"volatile:{nid}:{next(_volatile_seq)}"
                return fps[nid]
            spec = json.dumps(
                [node["class_type"], widgets, links, repr(mark)],
                sort_keys=True, default=repr,
            )
            fps[nid] = hashlib.md5(spec.encode()).hexdigest()
            return fps[nid]

        for nid in graph.node_ids():
            fp(nid)
        return fps

    def execute(self, prompt: dict) -> dict[str, tuple]:
        """Run the graph; returns {node_id: outputs tuple} for OUTPUT_NODEs
        and every executed node."""
        validate_prompt(prompt, self.registry)
        graph = PromptGraph(prompt)
        self.context["current_prompt"] = prompt  # for output metadata
        fps = self._fingerprints(graph)
        next_run_cache: dict[str, tuple] = {}
        cache: dict[str, tuple] = {}

        def eval_node(nid: str) -> tuple:
            if nid in cache:
                return cache[nid]
            node = graph.node(nid)
            cls = self.registry.get(node["class_type"])
            # OUTPUT_NODEs never reuse a cross-run cached result; IS_CHANGED
            # volatility (the USDU node's nan) is handled in the fingerprint
            # itself (a volatile fingerprint never matches across runs), so
            # content-marking nodes like LoadImage can still cache when their
            # file is unchanged
            is_output = bool(getattr(cls, "OUTPUT_NODE", False))
            if not is_output and fps[nid] in self._run_cache:
                out = self._run_cache[fps[nid]]
                cache[nid] = out
                next_run_cache[fps[nid]] = out
                return out
            kwargs = {}
            for name, value in node.get("inputs", {}).items():
                if is_link(value):
                    src_out = eval_node(str(value[0]))
                    kwargs[name] = src_out[int(value[1])]
                else:
                    kwargs[name] = value
            # ComfyUI hidden-input convention: UNIQUE_ID injects the node's
            # own id, EXTRA_PNGINFO the executing workflow (the prompt dict
            # here — it is what SaveImage embeds into output metadata), so
            # nodes like DistributedModelName can write resolved values
            # back (reference nodes/utilities.py:164-224)
            hidden = {}
            try:
                hidden = cls.INPUT_TYPES().get("hidden", {})
            except Exception:
                pass
            for hname, hkind in hidden.items():
                if hname in kwargs:
                    continue
                if hkind == "UNIQUE_ID":
                    kwargs[hname] = nid
                elif hkind == "EXTRA_PNGINFO":
                    kwargs[hname] = {"workflow": prompt}
            inst = cls()
            if hasattr(inst, "set_context"):
                inst.set_context(self.context)
            fn = getattr(inst, getattr(cls, "FUNCTION", "run"))
            out = fn(**kwargs)
            if not isinstance(out, tuple):
                out = (out,)
            cache[nid] = out
            if not is_output:
                next_run_cache[fps[nid]] = out
            return out

        # evaluate every sink (output nodes and nodes nobody consumes)
        consumed: set[str] = set()
        for nid in graph.node_ids():
            for _name, src in graph.input_links(nid):
                consumed.add(src)
        for nid in graph.node_ids():
            cls = self.registry.get(graph.class_of(nid))
            is_output = bool(getattr(cls, "OUTPUT_NODE", False))
            if is_output or nid not in consumed:
                eval_node(nid)
        self._run_cache = next_run_cache
        return cache
