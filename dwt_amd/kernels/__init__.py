"""HIP/CDNA4 kernel extension for dwt_amd (gfx950, built in-tree).

``build.py`` compiles the extension with hipcc (cross-compiles fine on a
CPU-only box); ``dispatch.py`` is the python-side dispatch layer the ops in
``dwt_amd/ops`` call into.
"""
