"""The sphinx doc-build hooks (epytext converter, underscores mover)
are importable and convert correctly without a sphinx install."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "docs"))


def test_epytext_fields():
    import epytext
    assert epytext._convert_line("@param np: procs") == ":param np: procs"
    assert epytext._convert_line("@return: value") == ":returns: value"
    assert epytext._convert_line("@rtype: int") == ":rtype: int"
    assert epytext._convert_line("@raise ValueError: bad") == \
        ":raises ValueError: bad"
    assert epytext._convert_line("C{x} and L{y}") == "``x`` and :any:`y`"


def test_underscores_rewrite(tmp_path):
    import underscores
    html = tmp_path / "index.html"
    html.write_text('<link href="_static/a.css"><a href="_sources/x">')
    underscores._rewrite_html(str(tmp_path))
    out = html.read_text()
    assert "_static" not in out and "static/a.css" in out
    assert "sources/x" in out
