"""pci.ids resolver tests (reference: device_plugin_test.go:375-426)."""

import textwrap

from kubevirt_gpu_device_plugin_amd.device_plugin import pciids


def _ids_file(tmp_path, content):
    p = tmp_path / "pci.ids"
    p.write_text(textwrap.dedent(content))
    return str(p)


def test_vendor_block_scoped_lookup(tmp_path):
    """A colliding device id under another vendor must not match
    (reference pins NVIDIA id 2331 colliding across vendors,
    device_plugin_test.go:421-425)."""
    path = _ids_file(tmp_path, """\
        10de  NVIDIA Corporation
        \t75a3  Not An AMD Part
        1002  Advanced Micro Devices, Inc. [AMD/ATI]
        \t75a3  Instinct MI355X
        1af4  Red Hat, Inc.
        \t75a3  Virtio colliding id
        """)
    assert pciids.get_device_name("75a3", pci_ids_path=path) \
        == "INSTINCT_MI355X"
    assert pciids.get_device_name(
        "75a3", vendor_id="10de", pci_ids_path=path) == "NOT_AN_AMD_PART"


def test_not_found_returns_empty(tmp_path):
    path = _ids_file(tmp_path, """\
        1002  AMD
        \t75a3  Instinct MI355X
        """)
    assert pciids.get_device_name("beef", pci_ids_path=path) == ""


def test_stops_at_next_vendor(tmp_path):
    path = _ids_file(tmp_path, """\
        1002  AMD
        \t74a1  Instinct MI300X
        8086  Intel
        \t75a3  Some Intel thing
        """)
    assert pciids.get_device_name("75a3", pci_ids_path=path) == ""


def test_sanitization_rules():
    # uppercase, / → _, . → _, whitespace → _, strip other chars
    # (reference: device_plugin.go:404-414)
    assert pciids.sanitize_name("Aqua Vanjaram [Instinct MI300X]") \
        == "AQUA_VANJARAM_INSTINCT_MI300X"
    assert pciids.sanitize_name("GA100 [A100/PCIe 40.GB]") \
        == "GA100_A100_PCIE_40_GB"


def test_comments_ignored(tmp_path):
    path = _ids_file(tmp_path, """\
        1002  AMD
        # a comment line inside the block
        \t75a3  Instinct MI355X
        """)
    assert pciids.get_device_name("75a3", pci_ids_path=path) \
        == "INSTINCT_MI355X"


def test_prefix_id_does_not_match(tmp_path):
    path = _ids_file(tmp_path, """\
        1002  AMD
        \t75a31  Bogus longer id
        """)
    assert pciids.get_device_name("75a3", pci_ids_path=path) == ""


def test_builtin_table_has_mi355x():
    """The curated table must resolve CDNA4 parts so resource names never
    degrade to raw hex ids (SURVEY.md §7.2 hard part)."""
    assert pciids.get_device_name(
        "75a3", pci_ids_path=pciids.BUILTIN_IDS_PATH) == "INSTINCT_MI355X"
    assert pciids.get_device_name(
        "75b3", pci_ids_path=pciids.BUILTIN_IDS_PATH) \
        == "INSTINCT_MI355X_VF"
    assert pciids.get_device_name(
        "74a1", pci_ids_path=pciids.BUILTIN_IDS_PATH) \
        == "AQUA_VANJARAM_INSTINCT_MI300X"


def test_search_path_order(tmp_path):
    """The system db (first path) wins over the built-in table; the
    built-in backfills ids the system db lacks."""
    system = _ids_file(tmp_path, """\
        1002  AMD
        \t74a1  System Db Name For MI300X
        """)
    got = pciids.get_device_name(
        "74a1", search_paths=[system, pciids.BUILTIN_IDS_PATH])
    assert got == "SYSTEM_DB_NAME_FOR_MI300X"
    # 75a3 absent from the fake system db → built-in fallback
    got = pciids.get_device_name(
        "75a3", search_paths=[system, pciids.BUILTIN_IDS_PATH])
    assert got == "INSTINCT_MI355X"


def test_missing_system_db_skipped(tmp_path):
    got = pciids.get_device_name(
        "75a3", search_paths=[str(tmp_path / "nope"),
                              pciids.BUILTIN_IDS_PATH])
    assert got == "INSTINCT_MI355X"
