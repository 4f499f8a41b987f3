"""Single version constant.

The reference keeps two divergent VERSION consts (cmd/kube-copilot/server.go:29
"v1.0.2" vs pkg/handlers/version.go:8 "v1.0.18"); we keep exactly one.
"""

VERSION = "v1.0.0"
__version__ = VERSION
