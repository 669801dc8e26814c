"""Strategic merge, template rendering, identity determinism units."""

import pytest

from fma_amd.controller.dualpods import nominal
from fma_amd.controller.dualpods.identity import (instance_id, nominal_hash,
                                                  template_hash)


def test_render_template_substitutes_provider_data():
    out = nominal.render_template(
        "vol: {{.LocalVolume}} on {{ .NodeName }}",
        {"NodeName": "n1", "LocalVolume": "pvc-a"})
    assert out == "vol: pvc-a on n1"


def test_render_template_unknown_field_raises():
    with pytest.raises(nominal.NominalError):
        nominal.render_template("{{.Nope}}", {"NodeName": "n1"})


def test_strategic_merge_named_lists_merge_by_name():
    base = {"containers": [
        {"name": "a", "image": "a:1", "env": [{"name": "X", "value": "1"}]},
        {"name": "b", "image": "b:1"}]}
    patch = {"containers": [
        {"name": "a", "image": "a:2"},
        {"name": "c", "image": "c:1"}]}
    out = nominal.strategic_merge(base, patch)
    by_name = {c["name"]: c for c in out["containers"]}
    assert by_name["a"]["image"] == "a:2"
    assert by_name["a"]["env"] == [{"name": "X", "value": "1"}]  # kept
    assert by_name["b"]["image"] == "b:1"
    assert "c" in by_name
    assert [c["name"] for c in out["containers"]] == ["a", "b", "c"]


def test_strategic_merge_none_deletes():
    assert nominal.strategic_merge({"a": 1, "b": 2}, {"a": None}) == {"b": 2}


def test_strategic_merge_plain_list_replaces():
    assert nominal.strategic_merge({"l": [1, 2]}, {"l": [3]}) == {"l": [3]}


def test_instance_id_deterministic_and_gpu_sensitive():
    msc = {"port": 8000, "options": "--model x"}
    a = instance_id(msc, ["GPU-0"])
    assert a == instance_id(dict(msc), ["GPU-0"])  # dict order irrelevant
    assert a != instance_id(msc, ["GPU-1"])
    assert a != instance_id({**msc, "options": "--model y"}, ["GPU-0"])
    assert a.startswith("I") and a.endswith("i")


def test_nominal_hash_covers_spec_gpus_node():
    spec = {"containers": [{"name": "inference-server"}]}
    h = nominal_hash(spec, ["GPU-0"], "n1")
    assert h == nominal_hash(dict(spec), ["GPU-0"], "n1")
    assert h != nominal_hash(spec, ["GPU-1"], "n1")
    assert h != nominal_hash(spec, ["GPU-0"], "n2")


def test_template_hash_stable_under_canonicalization():
    from fma_amd.controller.populator.podtemplate import canonicalize_template
    t1 = {"spec": {"volumes": [{"name": "b"}, {"name": "a"}],
                   "containers": [{"name": "launcher",
                                   "ports": [{"containerPort": 9},
                                             {"containerPort": 1}]}]}}
    t2 = {"spec": {"volumes": [{"name": "a"}, {"name": "b"}],
                   "containers": [{"name": "launcher",
                                   "ports": [{"containerPort": 1},
                                             {"containerPort": 9}]}]}}
    assert template_hash(canonicalize_template(t1)) == \
        template_hash(canonicalize_template(t2))


# -- property fuzz: instance identity ---------------------------------------

from hypothesis import given, settings, strategies as st  # noqa: E402

_opts = st.text(st.characters(whitelist_categories=("Ll", "Nd"),
                              whitelist_characters=" -"), max_size=30)


@settings(max_examples=150, deadline=None)
@given(opts=_opts, port=st.integers(1024, 65535),
       gpus=st.lists(st.sampled_from(["GPU-0", "GPU-1", "GPU-2"]),
                     min_size=1, max_size=3, unique=True))
def test_instance_id_properties(opts, port, gpus):
    """Identity is a pure function of (ModelServerConfig, GPU list):
    deterministic across calls, I...i shaped (reference
    inference-server.go:1016-1058), and sensitive to every input."""
    from fma_amd.controller.dualpods.identity import instance_id

    msc = {"port": port, "options": opts}
    a = instance_id(msc, gpus)
    assert a == instance_id(dict(msc), list(gpus))
    assert a.startswith("I") and a.endswith("i") and len(a) > 10
    assert a != instance_id({"port": port, "options": opts + "x"}, gpus)
    assert a != instance_id({"port": (port % 65534) + 1, "options": opts},
                            gpus)
    assert a != instance_id(msc, gpus + ["GPU-9"])


def test_strategic_merge_properties_fuzz():
    """Property sweep over random dict/list structures: (1) merging an
    empty patch is identity; (2) patch keys win; (3) named-list merge
    preserves base order and appends new names; (4) the result never
    aliases mutable state of base or patch."""
    import random

    from fma_amd.controller.dualpods import nominal

    rng = random.Random(13)

    def rand_value(depth):
        r = rng.random()
        if depth >= 3 or r < 0.35:
            return rng.choice([1, "x", "y", True, None if depth else 2])
        if r < 0.6:
            return {rng.choice("abcd"): rand_value(depth + 1)
                    for _ in range(rng.randint(0, 3))}
        if r < 0.8:
            return [{"name": f"n{i}", "v": rand_value(depth + 1)}
                    for i in range(rng.randint(0, 3))]
        return [rand_value(depth + 1) for _ in range(rng.randint(0, 3))]

    import copy
    for _ in range(300):
        base = {rng.choice("abcdef"): rand_value(0)
                for _ in range(rng.randint(0, 4))}
        base = {k: v for k, v in base.items() if v is not None}
        base_snapshot = copy.deepcopy(base)

        # (1) empty patch is identity
        assert nominal.strategic_merge(base, {}) == base

        patch = {rng.choice("abcdef"): rand_value(0)
                 for _ in range(rng.randint(0, 4))}
        patch_snapshot = copy.deepcopy(patch)
        out = nominal.strategic_merge(base, patch)
        # inputs never mutated
        assert base == base_snapshot
        assert patch == patch_snapshot
        # (2) scalar patch keys win; None deletes
        for k, v in patch.items():
            if v is None:
                assert k not in out
            elif not isinstance(v, (dict, list)):
                assert out[k] == v
        # (4) no aliasing: mutating the result leaves inputs intact
        for k in list(out):
            if isinstance(out[k], dict):
                out[k]["__mut__"] = 1
            elif isinstance(out[k], list):
                out[k].append("__mut__")
        assert base == base_snapshot
        assert patch == patch_snapshot

    # (3) named-list semantics, explicit
    base = {"containers": [{"name": "a", "x": 1}, {"name": "b", "x": 2}]}
    patch = {"containers": [{"name": "b", "x": 9}, {"name": "c", "x": 3}]}
    out = nominal.strategic_merge(base, patch)
    assert [c["name"] for c in out["containers"]] == ["a", "b", "c"]
    assert out["containers"][1]["x"] == 9
