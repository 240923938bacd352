from xotorch_amd.orchestration.node import Node  # noqa: F401
