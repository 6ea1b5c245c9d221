"""Property-based tests over random workflow DAGs: the graph transforms
must produce closed, executable-shaped graphs for ANY topology, not just
the example workflows."""

from hypothesis import given, settings
from hypothesis import strategies as st

from comfyui_distributed_amd.graph.prompt import PromptGraph, is_link
from comfyui_distributed_amd.graph.transform import (
    prepare_delegate_master_prompt,
    prune_prompt_for_worker,
)

CLASSES = ["CLIPTextEncode", "KSampler", "VAEDecode", "ImageBatchDivider",
           "PreviewImage", "SaveImage"]


@st.composite
def random_prompt(draw):
    """A random DAG: node i may link only to nodes < i (acyclic by
    construction); one collector placed at a random position; a USDU node
    sometimes."""
    n = draw(st.integers(min_value=2, max_value=12))
    collector_at = draw(st.integers(min_value=1, max_value=n - 1))
    usdu_at = draw(st.one_of(st.none(), st.integers(0, n - 1)))
    prompt = {}
    for i in range(n):
        if i == collector_at:
            cls = "DistributedCollector"
        elif usdu_at is not None and i == usdu_at and i != collector_at:
            cls = "UltimateSDUpscaleDistributed"
        else:
            cls = draw(st.sampled_from(CLASSES))
        inputs = {}
        n_links = draw(st.integers(0, min(i, 3)))
        for j in range(n_links):
            src = draw(st.integers(0, i - 1))
            inputs[f"in{j}"] = [str(src), 0]
        if draw(st.booleans()):
            inputs["widget"] = draw(st.integers(0, 100))
        prompt[str(i)] = {"class_type": cls, "inputs": inputs}
    return prompt


def assert_closed(graph: PromptGraph):
    """Every link inside the graph points at a node inside the graph."""
    for nid in graph.node_ids():
        for _name, src in graph.input_links(nid):
            assert src in graph.raw, f"dangling link {nid} -> {src}"


@given(random_prompt())
@settings(max_examples=150, deadline=None)
def test_worker_prune_is_closed_and_keeps_distributed_nodes(prompt):
    g = PromptGraph(prompt)
    pruned = prune_prompt_for_worker(g)
    assert_closed(pruned)
    dist = g.nodes_of_class("DistributedCollector",
                            "UltimateSDUpscaleDistributed")
    for nid in dist:
        assert nid in pruned.raw, "distributed node pruned away"
        # every distributed node's output is consumed (real sink or the
        # auto-appended PreviewImage)
        consumed = {src for m in pruned.node_ids()
                    for _n, src in pruned.input_links(m)}
        assert nid in consumed
    # original untouched
    assert prompt == {k: v for k, v in g.raw.items()}


@given(random_prompt())
@settings(max_examples=150, deadline=None)
def test_delegate_master_prompt_is_closed(prompt):
    g = PromptGraph(prompt)
    collectors = g.nodes_of_class("DistributedCollector")
    out = prepare_delegate_master_prompt(g, collectors)
    assert_closed(out)
    for cid in collectors:
        assert cid in out.raw
        inputs = out.inputs(cid)
        assert inputs.get("delegate_only") is True
        # the images input, if a link, must point at the placeholder or a
        # kept downstream node — never at a pruned upstream id
        for name, v in inputs.items():
            if is_link(v):
                assert str(v[0]) in out.raw


@given(st.integers(0, 500), st.integers(1, 10))
@settings(max_examples=80, deadline=None)
def test_chunk_bounds_partition(total, parts):
    from comfyui_distributed_amd.nodes.utilities import chunk_bounds

    bounds = chunk_bounds(total, parts)
    assert len(bounds) == parts
    # contiguous partition of [0, total)
    assert bounds[0][0] == 0 and bounds[-1][1] == total
    for (a1, b1), (a2, b2) in zip(bounds, bounds[1:]):
        assert b1 == a2 and a1 <= b1 and a2 <= b2
    sizes = [b - a for a, b in bounds]
    assert max(sizes) - min(sizes) <= 1  # balanced
    assert sum(sizes) == total
