"""Sphinx autodoc hook translating epytext-style docstring markup to
reST on the fly (reference parity surface: docs/epytext.py — the
reference repo carries legacy epytext docstrings; this framework's own
docstrings are reST, but migrated user code may still use the epytext
forms, so the converter keeps autodoc rendering them correctly).

Handled forms:
  @param name: text   ->  :param name: text
  @type name: text    ->  :type name: text
  @return: text       ->  :returns: text
  @rtype: text        ->  :rtype: text
  @raise Exc: text    ->  :raises Exc: text
  C{code}             ->  ``code``
  L{target}           ->  :any:`target`
"""

import re

_FIELD_MAP = {
    "param": "param",
    "type": "type",
    "keyword": "keyword",
    "return": "returns",
    "returns": "returns",
    "rtype": "rtype",
    "raise": "raises",
    "raises": "raises",
    "ivar": "ivar",
    "cvar": "cvar",
}

_FIELD_RE = re.compile(
    r"@(%s)(\s+[\w.*]+)?\s*:" % "|".join(_FIELD_MAP))
_INLINE_CODE_RE = re.compile(r"C\{([^}]*)\}")
_INLINE_LINK_RE = re.compile(r"L\{([^}]*)\}")


def _convert_line(line):
    def field(m):
        name = _FIELD_MAP[m.group(1)]
        arg = m.group(2) or ""
        return ":%s%s:" % (name, arg)

    line = _FIELD_RE.sub(field, line)
    line = _INLINE_CODE_RE.sub(r"``\1``", line)
    line = _INLINE_LINK_RE.sub(r":any:`\1`", line)
    return line


def _process_docstring(app, what, name, obj, options, lines):
    for i, line in enumerate(lines):
        lines[i] = _convert_line(line)


def setup(app):
    app.connect("autodoc-process-docstring", _process_docstring)
    return {"parallel_read_safe": True}
