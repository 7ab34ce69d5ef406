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


# ---- search-order contract + full-public-db hardening ------------------

import os as _os

import pytest as _pytest

REF_FULL_IDS = "/root/reference/utils/pci.ids"


def test_search_order_full_db_miss_falls_to_curated(tmp_path):
    """/usr/pci.ids exists but lacks 75a3 (true of public v2025.07.11):
    the curated built-in table must still resolve it."""
    full = tmp_path / "pci.ids"
    full.write_text("1002  Advanced Micro Devices, Inc. [AMD/ATI]\n"
                    "\t74a1  Aqua Vanjaram [Instinct MI300X]\n")
    name = pciids.get_device_name(
        "75a3", search_paths=[str(full), pciids.BUILTIN_IDS_PATH])
    assert name == "INSTINCT_MI355X"


def test_search_order_full_db_name_overrides_curated(tmp_path):
    """The converse: when the full db DOES carry the id, its name wins
    over the curated table (first hit in search order)."""
    full = tmp_path / "pci.ids"
    full.write_text("1002  Advanced Micro Devices, Inc. [AMD/ATI]\n"
                    "\t75a3  Instinct MI355X rev B\n")
    name = pciids.get_device_name(
        "75a3", search_paths=[str(full), pciids.BUILTIN_IDS_PATH])
    assert name == "INSTINCT_MI355X_REV_B"


def test_vendor_line_requires_word_boundary(tmp_path):
    """A longer id sharing the vendor prefix must not open the block."""
    db = tmp_path / "pci.ids"
    db.write_text("1002b  Not A Real Vendor\n"
                  "\t75a3  Imposter Device\n"
                  "1002  Advanced Micro Devices, Inc. [AMD/ATI]\n"
                  "\t75a3  Instinct MI355X\n")
    assert pciids.get_device_name(
        "75a3", pci_ids_path=str(db)) == "INSTINCT_MI355X"


@_pytest.mark.skipif(not _os.path.exists(REF_FULL_IDS),
                     reason="full public pci.ids not on this host")
class TestAgainstFullPublicDb:
    """Hardening against the real 40k-line public database
    (v2025.07.11, vendored by the reference)."""

    def test_amd_id_resolves(self):
        assert pciids.get_device_name(
            "74a1", pci_ids_path=REF_FULL_IDS) == \
            "AQUA_VANJARAM_INSTINCT_MI300X"

    def test_cross_vendor_id_never_matches(self):
        # 2331 exists under 10de (H100) and 8086-adjacent blocks but
        # not under 1002: vendor-block scoping must return ""
        assert pciids.get_device_name(
            "2331", pci_ids_path=REF_FULL_IDS) == ""
        assert pciids.get_device_name(
            "2331", vendor_id="10de", pci_ids_path=REF_FULL_IDS) == \
            "GH100_H100_PCIE"

    def test_prefix_id_collision_guard(self):
        # "74a" is a strict prefix of 74a0/74a1/…: the guard must not
        # return MI300A's name for it
        assert pciids.get_device_name(
            "74a", pci_ids_path=REF_FULL_IDS) == ""

    def test_mi355x_absent_from_public_db_needs_curated(self):
        # the documented "hard part" (SURVEY §7.2): public v2025.07.11
        # has no 75a3 — proves the curated fallback is load-bearing
        assert pciids.get_device_name(
            "75a3", pci_ids_path=REF_FULL_IDS) == ""
        assert pciids.get_device_name(
            "75a3", search_paths=[REF_FULL_IDS,
                                  pciids.BUILTIN_IDS_PATH]) == \
            "INSTINCT_MI355X"
