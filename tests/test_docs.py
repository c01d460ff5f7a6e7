"""Docs integrity (the reference's CI builds its docs as a test —
build.yml:57-70; the analog here: every relative link in docs/ and
README resolves, and every documented CLI command exists)."""

import re
from pathlib import Path

import unionml_amd

REPO = Path(unionml_amd.__file__).parent.parent
LINK_RE = re.compile(r"\[[^\]]*\]\(([^)#\s]+)\)")


def test_docs_relative_links_resolve():
    md_files = list((REPO / "docs").glob("*.md")) + [REPO / "README.md"]
    assert len(md_files) >= 15
    broken = []
    for md in md_files:
        for target in LINK_RE.findall(md.read_text()):
            if target.startswith(("http://", "https://", "mailto:")):
                continue
            resolved = (md.parent / target).resolve()
            if not resolved.exists():
                broken.append(f"{md.name} -> {target}")
    assert not broken, broken


def test_docs_reference_existing_profiles():
    text = "".join(p.read_text() for p in (REPO / "docs").glob("*.md"))
    for ref in re.findall(r"profiles/[\w.]+\.(?:md|json|csv)", text):
        assert (REPO / ref).exists(), ref


def test_documented_cli_commands_exist():
    from typer.testing import CliRunner

    from unionml_amd.cli import app

    result = CliRunner().invoke(app, ["--help"])
    help_text = result.output
    cli_doc = (REPO / "docs" / "cli.md").read_text()
    for cmd in re.findall(r"`(init|deploy|train|predict|serve|activate-schedules|"
                          r"deactivate-schedules|list-model-versions|list-prediction-ids|"
                          r"fetch-model|fetch-predictions|run-scheduler)", cli_doc):
        assert cmd in help_text, cmd
