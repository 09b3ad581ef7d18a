from . import v1alpha1  # noqa: F401
