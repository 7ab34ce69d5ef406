"""PCI device-name resolution from the pci.ids database.

Streams the database, seeks the vendor block, and matches the device id
*within that block only* — a device line for another vendor never
matches, preventing cross-vendor id collisions
(reference: getDeviceName/locateVendor, device_plugin.go:371-438, and the
colliding-id ``2331`` test, device_plugin_test.go:421-425).

Sanitization of the marketing name into a resource-name suffix follows
the reference rules exactly (device_plugin.go:404-414): uppercase,
``/`` → ``_``, ``.`` → ``_``, whitespace runs → ``_``, then strip every
char outside ``[a-zA-Z0-9_.]``.
"""

import os
import re

from . import consts

_WS = re.compile(r"\s+")
_BAD = re.compile(r"[^a-zA-Z0-9_.]+")

# Built-in curated table (always present in the package).
BUILTIN_IDS_PATH = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "utils", "amd_pci.ids")


def sanitize_name(name):
    name = name.strip().upper()
    name = name.replace("/", "_").replace(".", "_")
    name = _WS.sub("_", name)
    return _BAD.sub("", name)


def _lookup_in_file(path, device_id, vendor_id):
    in_vendor = False
    prefix = "\t" + device_id
    with open(path, "r", errors="replace") as f:
        for line in f:
            line = line.rstrip("\n")
            if not in_vendor:
                # vendor line = the 4-hex id at column 0 followed by
                # whitespace (a bare startswith would also match a
                # hypothetical longer id sharing the prefix)
                if line.startswith(vendor_id):
                    rest = line[len(vendor_id):]
                    if not rest or rest[0].isspace():
                        in_vendor = True
                continue
            if line.startswith("#"):
                continue
            if not line.startswith("\t"):
                # Next vendor block: the id does not exist for this vendor.
                return ""
            if line.startswith(prefix):
                rest = line[len(prefix):]
                # Guard against prefix-id collisions (e.g. "74a" vs
                # "74a1"): the id must be followed by whitespace.
                if rest and not rest[0].isspace():
                    continue
                return sanitize_name(rest)
    return ""


def get_device_name(device_id, vendor_id=consts.AMD_VENDOR_ID,
                    pci_ids_path=None, search_paths=None):
    """Resolve a sanitized device name for ``device_id``.

    Search order: explicit ``pci_ids_path`` → ``search_paths``
    (default: the system database at ``/usr/pci.ids`` — the full db,
    embedded by the container build — then the curated built-in table).
    Returns "" when not found anywhere; the controller then falls back
    to the raw hex id (reference: device_plugin.go:124-128).
    """
    if pci_ids_path:
        candidates = [pci_ids_path]
    elif search_paths is not None:
        candidates = list(search_paths)
    else:
        candidates = [consts.PCI_IDS_FILE_PATH, BUILTIN_IDS_PATH]
    for path in candidates:
        try:
            name = _lookup_in_file(path, device_id, vendor_id)
        except OSError:
            continue
        if name:
            return name
    return ""
