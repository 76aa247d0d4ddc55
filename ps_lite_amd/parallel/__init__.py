from .local import LocalCluster, launch_local  # noqa: F401
