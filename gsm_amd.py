"""Import alias: the package directory is `go-spacemesh_amd/` (hyphenated to
mirror the reference repo name), which Python cannot import literally.
`import gsm_amd` re-exports it."""
import importlib
import sys

_pkg = importlib.import_module("go-spacemesh_amd")
sys.modules[__name__] = _pkg
