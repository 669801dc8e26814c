"""CEL evaluator unit tests (the engine behind the shipped VAP YAML)."""

import pytest

from fma_amd.store.cel import CelError, evaluate


def env(user="alice", old=None, new=None):
    return {"request": {"userInfo": {"username": user}},
            "oldObject": old or {"metadata": {}},
            "object": new or {"metadata": {}}}


def test_literals_and_logic():
    assert evaluate("true || false", {}) is True
    assert evaluate("true && false", {}) is False
    assert evaluate("!(1 == 2)", {}) is True
    assert evaluate("'a' != 'b'", {}) is True
    assert evaluate("('x' == 'x') && ('y' == 'y')", {}) is True


def test_member_and_index_access():
    e = {"object": {"metadata": {"labels": {"a": "1"}}}}
    assert evaluate("object.metadata.labels['a']", e) == "1"
    with pytest.raises(CelError):
        evaluate("object.metadata.annotations['a']", e)  # non-optional miss


def test_optional_chaining_and_orvalue():
    e = {"object": {"metadata": {}}}
    assert evaluate("object.metadata.?labels['a'].orValue('d')", e) == "d"
    e = {"object": {"metadata": {"labels": {"a": "v"}}}}
    assert evaluate("object.metadata.?labels['a'].orValue('d')", e) == "v"
    assert evaluate("object.metadata.?labels['zz'].orValue('d')", e) == "d"


def test_string_methods():
    e = env(user="system:serviceaccount:ns:rel-fma-controllers")
    assert evaluate(
        'request.userInfo.username.matches('
        '"^system:serviceaccount:[^:]+:[^:]*-fma-controllers$")', e) is True
    assert evaluate('"hello".startsWith("he")', {}) is True
    assert evaluate('"hello".endsWith("lo")', {}) is True
    assert evaluate('"hello".contains("ell")', {}) is True
    assert evaluate('"hello".size()', {}) == 5


def test_has_macro_and_in():
    e = {"object": {"metadata": {"labels": {"a": "1"}}}}
    assert evaluate("has(object.metadata.labels)", e) is True
    assert evaluate("has(object.metadata.annotations)", e) is False
    assert evaluate("'a' in ['a', 'b']", {}) is True
    assert evaluate("'z' in ['a', 'b']", {}) is False


def test_errors():
    with pytest.raises(CelError):
        evaluate("object.", {"object": {}})
    with pytest.raises(CelError):
        evaluate("unknown_var", {})
    with pytest.raises(CelError):
        evaluate("'str' && true", {})  # non-bool in boolean context
    with pytest.raises(CelError):
        # optional used as value without orValue
        evaluate("object.?missing == 'x'", {"object": {}})


def test_shipped_policy_files_compile():
    """Every expression in the shipped VAP YAML must compile."""
    from fma_amd.store.kubeapiserver import VAP_DIR, AdmissionPolicies
    pols = AdmissionPolicies(VAP_DIR)
    assert set(pols.policies) == {"fma-immutable-fields",
                                  "fma-bound-serverreqpod"}
    assert len(pols.bound) == 2


def test_cel_fuzz_never_crashes_unexpectedly():
    """Property: arbitrary token soup either evaluates or raises CelError
    — never an uncontrolled exception (hypothesis-style fuzz without a
    GPU-worth of runtime)."""
    import random

    from fma_amd.store.cel import CelError

    random.seed(42)
    atoms = ["object", "oldObject", "request", ".", ".?", "(", ")", "[",
             "]", "'k'", '"v"', "==", "!=", "||", "&&", "!", "orValue",
             "matches", "has", "metadata", "annotations", "labels", "in",
             "true", "false", "null", "1", ",", "size"]
    env = {"object": {"metadata": {"labels": {"k": "v"}}},
           "oldObject": {"metadata": {}},
           "request": {"userInfo": {"username": "u"}}}
    crashes = 0
    for _ in range(3000):
        expr = " ".join(random.choice(atoms)
                        for _ in range(random.randint(1, 12)))
        try:
            evaluate(expr, env)
        except CelError:
            pass
        except RecursionError:
            pass
        except Exception as e:  # noqa: BLE001
            crashes += 1
            if crashes <= 3:
                print(f"UNEXPECTED {type(e).__name__} for {expr!r}: {e}")
    assert crashes == 0


def test_cel_valid_expression_fuzz_consistent():
    """Property: generated WELL-FORMED policy-style expressions evaluate
    to the same result as an equivalent Python evaluation."""
    import random

    random.seed(7)
    keys = ["a", "b", "c"]
    for _ in range(500):
        ann = {k: random.choice(["", "x", "y"])
               for k in random.sample(keys, random.randint(0, 3))}
        old_ann = {k: random.choice(["", "x", "y"])
                   for k in random.sample(keys, random.randint(0, 3))}
        k = random.choice(keys)
        expr = (f"oldObject.metadata.?annotations['{k}'].orValue('') == "
                f"object.metadata.?annotations['{k}'].orValue('')")
        env = {"object": {"metadata": {"annotations": ann}},
               "oldObject": {"metadata": {"annotations": old_ann}}}
        expect = old_ann.get(k, "") == ann.get(k, "")
        assert evaluate(expr, env) is expect, (expr, env)
