from .launch import LaunchSpec, RankResult, launch_ranks  # noqa: F401
