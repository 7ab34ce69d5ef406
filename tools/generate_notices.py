#!/usr/bin/env python3
"""Generate THIRD_PARTY_NOTICES.md from the runtime dependency graph
(reference analogue: tools/generate-notices.sh runs go-licenses over the
./cmd build graph).  Python equivalent: importlib.metadata over the
declared runtime deps, recursively."""

import sys
from importlib import metadata

RUNTIME_DEPS = ["grpcio", "protobuf", "pybind11"]


def dist_info(name):
    try:
        d = metadata.distribution(name)
    except metadata.PackageNotFoundError:
        return None
    meta = d.metadata
    return {
        "name": meta.get("Name", name),
        "version": d.version,
        "license": meta.get("License-Expression")
                   or meta.get("License", "unknown"),
        "homepage": meta.get("Home-page", ""),
    }


def main():
    print("# Third-party notices\n")
    print("Runtime dependencies of kubevirt-gpu-device-plugin-amd:\n")
    for dep in sorted(RUNTIME_DEPS):
        info = dist_info(dep)
        if info is None:
            print("- %s: NOT INSTALLED" % dep, file=sys.stderr)
            continue
        print("## %(name)s %(version)s\n" % info)
        print("- License: %(license)s" % info)
        if info["homepage"]:
            print("- Homepage: %(homepage)s" % info)
        print()


if __name__ == "__main__":
    main()
