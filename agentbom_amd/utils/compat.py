"""Small stdlib compatibility shims (Python 3.10 base image)."""

try:
    import tomllib  # Python 3.11+
except ModuleNotFoundError:  # 3.10: the tomli wheel is API-identical
    import tomli as tomllib  # noqa: F401
