"""The suite registry must stay truthful: every referenced file exists,
every test file belongs to a suite (reference tests/suites.py tiering)."""

import pathlib

import suites


def test_all_suite_paths_exist():
    for s in suites.SUITES.values():
        for p in s.paths:
            assert (suites.TESTS_DIR / p).exists(), (s.name, p)


def test_every_test_file_is_registered():
    registered = {p for s in suites.SUITES.values() for p in s.paths}
    on_disk = {p.name for p in suites.TESTS_DIR.glob("test_*.py")}
    missing = on_disk - registered - {"test_suites_registry.py"}
    assert not missing, f"unregistered test files: {sorted(missing)}"


def test_tiers_and_markers_consistent():
    for s in suites.SUITES.values():
        if s.tier == 2:
            assert s.marker == "gpu", s.name
        else:
            assert "gpu" not in s.marker or "not gpu" in s.marker, s.name


def test_cli_list(capsys):
    assert suites.main(["list"]) == 0
    out = capsys.readouterr().out
    assert "codecs" in out and "tier 2" in out
