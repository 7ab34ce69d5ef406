"""Transitive third-party notices generation
(reference analogue: go-licenses over the ./cmd build graph,
tools/generate-notices.sh:16-41)."""

import importlib.util
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

spec = importlib.util.spec_from_file_location(
    "generate_notices", os.path.join(REPO, "tools", "generate_notices.py"))
gen = importlib.util.module_from_spec(spec)
spec.loader.exec_module(gen)


def test_requirement_name_parsing():
    assert gen._requirement_name("typing-extensions>=4.0") == \
        "typing-extensions"
    assert gen._requirement_name(
        "tomli>=1.1; python_version < '3.11'") == "tomli"
    # extra-conditioned requirements are not part of the runtime graph
    assert gen._requirement_name(
        'grpcio-tools; extra == "dev"') is None


def test_closure_walks_transitively():
    """A dep-of-a-dep must land in the closure: protobuf declares
    typing-extensions in this environment, so the closure of the
    declared runtime deps is strictly larger than the root list
    whenever any root has requirements."""
    dists = gen.transitive_closure(gen.RUNTIME_DEPS)
    names = {d.metadata["Name"].lower().replace("_", "-")
             for d in dists}
    assert {"grpcio", "protobuf", "pybind11"} <= names
    roots_requires = set()
    for d in dists:
        if d.metadata["Name"].lower() in ("grpcio", "protobuf",
                                          "pybind11"):
            for spec_ in d.requires or []:
                n = gen._requirement_name(spec_)
                if n:
                    roots_requires.add(n.lower().replace("_", "-"))
    # every first-level requirement made it into the closure
    assert roots_requires <= names


def test_closure_handles_cycles_and_duplicates():
    dists = gen.transitive_closure(["protobuf", "protobuf",
                                    "Protobuf"])
    names = [d.metadata["Name"] for d in dists]
    assert len(names) == len(set(n.lower() for n in names))


def test_notices_file_is_fresh():
    """Same check CI runs: the committed file matches a regeneration."""
    import io
    import sys

    buf = io.StringIO()
    old = sys.stdout
    sys.stdout = buf
    try:
        gen.main()
    finally:
        sys.stdout = old
    with open(os.path.join(REPO, "THIRD_PARTY_NOTICES.md")) as f:
        assert f.read() == buf.getvalue()
