"""Post-build hook renaming the generated ``_static``/``_sources``
directories (GitHub Pages serves with Jekyll by default, which ignores
underscore-prefixed paths) — reference parity surface:
docs/underscores.py. Original implementation: rewrite the directory
names and patch every generated HTML page's references.
"""

import os
import re
import shutil

_DIRS = ("_static", "_sources", "_images")


def _rewrite_html(root):
    pat = re.compile(r"(%s)/" % "|".join(_DIRS))
    for dirpath, _dirnames, filenames in os.walk(root):
        for fn in filenames:
            if not fn.endswith((".html", ".js", ".css")):
                continue
            path = os.path.join(dirpath, fn)
            with open(path, encoding="utf-8", errors="ignore") as f:
                text = f.read()
            new = pat.sub(lambda m: m.group(1).lstrip("_") + "/", text)
            if new != text:
                with open(path, "w", encoding="utf-8") as f:
                    f.write(new)


def _move_dirs(app, exception):
    if exception is not None or app.builder.name != "html":
        return
    out = app.outdir
    for d in _DIRS:
        src = os.path.join(out, d)
        if os.path.isdir(src):
            dst = os.path.join(out, d.lstrip("_"))
            if os.path.isdir(dst):
                shutil.rmtree(dst)
            shutil.move(src, dst)
    _rewrite_html(out)


def setup(app):
    app.connect("build-finished", _move_dirs)
    return {"parallel_read_safe": True}
