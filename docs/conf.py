# Sphinx configuration for the sparkdl API docs
# (reference parity: docs build over the public modules, C10 in
# SURVEY.md §2.1; build with `make html` when sphinx is installed).

import os
import sys

sys.path.insert(0, os.path.abspath(".."))
sys.path.insert(0, os.path.abspath("."))  # local sphinx hooks

project = "sparkdl (MI355X-native)"
author = "sparkdl contributors"
release = "2.2.0-db1"

extensions = [
    "sphinx.ext.autodoc",
    "sphinx.ext.viewcode",
    "sphinx.ext.napoleon",
    "epytext",      # legacy epytext docstring markup -> reST
    "underscores",  # GH-Pages-safe _static/_sources renaming
]

autodoc_member_order = "bysource"
autodoc_mock_imports = ["tensorflow"]

exclude_patterns = ["_build"]
templates_path = ["_templates"]
html_static_path = ["static"]
html_theme = "classic"
html_theme_options = {"stickysidebar": "true"}
