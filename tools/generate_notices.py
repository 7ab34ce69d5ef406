#!/usr/bin/env python3
"""Generate THIRD_PARTY_NOTICES.md from the runtime dependency graph.

Python equivalent of the reference's go-licenses run over the exact
``./cmd`` build graph (reference: tools/generate-notices.sh:16-41):
starting from the declared runtime deps, ``Requires-Dist`` metadata is
walked *recursively* so a future dep that drags in its own tree changes
the notices file (and CI's freshness diff catches staleness).
"""

import sys
from importlib import metadata

try:
    from packaging.requirements import Requirement
except ImportError:  # pragma: no cover - packaging ships with pip
    Requirement = None

RUNTIME_DEPS = ["grpcio", "protobuf", "pybind11"]


def _requirement_name(spec):
    """Distribution name from a Requires-Dist spec, or None when the
    requirement is conditional on an extra / a non-matching marker."""
    if Requirement is None:
        # crude fallback: name is the leading token
        name = spec.split(";")[0].split("[")[0]
        for sep in "<>=!~ (":
            name = name.split(sep)[0]
        return name.strip() or None
    try:
        req = Requirement(spec)
    except Exception:
        return None
    if req.marker is not None:
        try:
            # no extras requested for runtime deps
            if not req.marker.evaluate({"extra": ""}):
                return None
        except Exception:
            return None
    return req.name


def transitive_closure(roots):
    """BFS over Requires-Dist, case-normalized; silently skips
    distributions that are not installed (they cannot ship in the
    image either)."""
    seen, order, queue = set(), [], list(roots)
    while queue:
        name = queue.pop(0)
        key = name.lower().replace("_", "-")
        if key in seen:
            continue
        seen.add(key)
        try:
            dist = metadata.distribution(name)
        except metadata.PackageNotFoundError:
            print("- %s: NOT INSTALLED (skipped)" % name,
                  file=sys.stderr)
            continue
        order.append(dist)
        for spec in dist.requires or []:
            dep = _requirement_name(spec)
            if dep:
                queue.append(dep)
    return order


def main():
    print("# Third-party notices\n")
    print("Runtime dependencies (transitive closure) of "
          "kubevirt-gpu-device-plugin-amd:\n")
    dists = transitive_closure(RUNTIME_DEPS)
    for d in sorted(dists, key=lambda d: d.metadata.get("Name", "")
                    .lower()):
        meta = d.metadata
        info = {
            "name": meta.get("Name", "?"),
            "version": d.version,
            "license": meta.get("License-Expression")
                       or meta.get("License", "unknown"),
            "homepage": meta.get("Home-page", ""),
        }
        print("## %(name)s %(version)s\n" % info)
        print("- License: %(license)s" % info)
        if info["homepage"]:
            print("- Homepage: %(homepage)s" % info)
        print()


if __name__ == "__main__":
    main()
